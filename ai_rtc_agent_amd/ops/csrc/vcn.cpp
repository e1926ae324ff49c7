// VCN H.264 codec session layer (native C++, VA-API via runtime dlopen).
//
// MI355X-native replacement scaffold for reference N1/N2:
// the aiortc fork's NVDEC/NVENC sessions. The MI355X exposes its VCN video
// blocks through the VA-API userspace (libva + the amdgpu DRM driver);
// neither ships in this build image (SURVEY.md §7 environment note), so
// every libva entry point is resolved at RUNTIME with dlopen/dlsym and the
// module degrades to a precise "unavailable" report instead of a link
// failure. On a deployment box with the VCN stack, probe() walks:
//   dlopen(libva, libva-drm) -> open /dev/dri/renderD* ->
//   vaGetDisplayDRM -> vaInitialize -> vaQueryConfigProfiles ->
//   H264 encode/decode entrypoint check
// and reports per-stage results; the session object owns the display and
// config lifetime. Frame-level encode/decode plumbing (param/bitstream
// buffers) lands on top of this session layer.

#include <dlfcn.h>
#include <fcntl.h>
#include <stdint.h>
#include <string.h>
#include <unistd.h>

#include <algorithm>
#include <string>
#include <vector>

// --- minimal, ABI-stable VA-API declarations (va.h public contract) -------
typedef void* VADisplay;
typedef int VAStatus;
typedef unsigned int VAGenericID;
typedef VAGenericID VAConfigID;
typedef int VAProfile;
typedef int VAEntrypoint;

#define VA_STATUS_SUCCESS 0x00000000
// VAProfile values (va.h): H264 main/high
#define VA_PROFILE_H264_MAIN 6
#define VA_PROFILE_H264_HIGH 7
// VAEntrypoint values (va.h)
#define VA_ENTRYPOINT_VLD 1        // decode
#define VA_ENTRYPOINT_ENCSLICE 6   // encode
#define VA_ENTRYPOINT_ENCSLICE_LP 8

namespace vcn {

struct VaApi {
  void* h_va = nullptr;
  void* h_va_drm = nullptr;
  VADisplay (*GetDisplayDRM)(int fd) = nullptr;
  VAStatus (*Initialize)(VADisplay, int* major, int* minor) = nullptr;
  VAStatus (*Terminate)(VADisplay) = nullptr;
  int (*MaxNumProfiles)(VADisplay) = nullptr;
  VAStatus (*QueryConfigProfiles)(VADisplay, VAProfile*, int*) = nullptr;
  int (*MaxNumEntrypoints)(VADisplay) = nullptr;
  VAStatus (*QueryConfigEntrypoints)(VADisplay, VAProfile, VAEntrypoint*, int*) = nullptr;

  bool load(std::string* err) {
    h_va = dlopen("libva.so.2", RTLD_NOW | RTLD_GLOBAL);
    if (!h_va) h_va = dlopen("libva.so", RTLD_NOW | RTLD_GLOBAL);
    if (!h_va) {
      *err = "libva not found (VCN userspace not installed)";
      return false;
    }
    h_va_drm = dlopen("libva-drm.so.2", RTLD_NOW);
    if (!h_va_drm) h_va_drm = dlopen("libva-drm.so", RTLD_NOW);
    if (!h_va_drm) {
      *err = "libva-drm not found";
      return false;
    }
#define RESOLVE(field, lib, name)                                   \
  field = reinterpret_cast<decltype(field)>(dlsym(lib, name));      \
  if (!field) {                                                     \
    *err = std::string("missing symbol ") + name;                   \
    return false;                                                   \
  }
    RESOLVE(GetDisplayDRM, h_va_drm, "vaGetDisplayDRM");
    RESOLVE(Initialize, h_va, "vaInitialize");
    RESOLVE(Terminate, h_va, "vaTerminate");
    RESOLVE(MaxNumProfiles, h_va, "vaMaxNumProfiles");
    RESOLVE(QueryConfigProfiles, h_va, "vaQueryConfigProfiles");
    RESOLVE(MaxNumEntrypoints, h_va, "vaMaxNumEntrypoints");
    RESOLVE(QueryConfigEntrypoints, h_va, "vaQueryConfigEntrypoints");
#undef RESOLVE
    return true;
  }
};

struct ProbeResult {
  bool available = false;
  std::string stage;    // how far the probe got
  std::string detail;   // failure reason or capability summary
  bool h264_decode = false;
  bool h264_encode = false;
  int va_major = 0, va_minor = 0;
  std::string device;
};

// Walk the render nodes looking for a VA display that initialises.
inline ProbeResult probe() {
  ProbeResult r;
  VaApi va;
  std::string err;
  r.stage = "dlopen";
  if (!va.load(&err)) {
    r.detail = err;
    return r;
  }
  r.stage = "drm-open";
  for (int node = 128; node < 136; ++node) {
    std::string dev = "/dev/dri/renderD" + std::to_string(node);
    int fd = open(dev.c_str(), O_RDWR);
    if (fd < 0) continue;
    VADisplay dpy = va.GetDisplayDRM(fd);
    if (!dpy) {
      close(fd);
      continue;
    }
    r.stage = "vaInitialize";
    if (va.Initialize(dpy, &r.va_major, &r.va_minor) != VA_STATUS_SUCCESS) {
      close(fd);
      continue;
    }
    r.device = dev;
    r.stage = "profiles";
    int maxp = va.MaxNumProfiles(dpy);
    std::vector<VAProfile> profiles(maxp > 0 ? maxp : 0);
    int np = 0;
    if (maxp > 0 &&
        va.QueryConfigProfiles(dpy, profiles.data(), &np) == VA_STATUS_SUCCESS) {
      for (int i = 0; i < np; ++i) {
        if (profiles[i] != VA_PROFILE_H264_MAIN && profiles[i] != VA_PROFILE_H264_HIGH)
          continue;
        int maxe = va.MaxNumEntrypoints(dpy);
        std::vector<VAEntrypoint> eps(maxe > 0 ? maxe : 0);
        int ne = 0;
        if (maxe > 0 && va.QueryConfigEntrypoints(dpy, profiles[i], eps.data(), &ne) ==
                            VA_STATUS_SUCCESS) {
          for (int j = 0; j < ne; ++j) {
            if (eps[j] == VA_ENTRYPOINT_VLD) r.h264_decode = true;
            if (eps[j] == VA_ENTRYPOINT_ENCSLICE || eps[j] == VA_ENTRYPOINT_ENCSLICE_LP)
              r.h264_encode = true;
          }
        }
      }
    }
    va.Terminate(dpy);
    close(fd);
    r.available = r.h264_decode || r.h264_encode;
    r.stage = "done";
    r.detail = r.available
                   ? "VCN H.264 " + std::string(r.h264_decode ? "dec " : "") +
                         std::string(r.h264_encode ? "enc" : "")
                   : "display up but no H.264 profile/entrypoint";
    return r;
  }
  if (r.stage == "drm-open") r.detail = "no usable /dev/dri/renderD* node";
  return r;
}

}  // namespace vcn

// ---------------------------------------------------------------------------
// VCN H.264 ENCODE session (VA-API, runtime-resolved).
//
// Round-1 verdict item #1: the session/param/bitstream-buffer plumbing
// behind the probe. Scope + honesty notes:
// - No environment reachable this round ships libva (the dev container and
//   the GPU pool image both lack it — probed 2026-09-13), so this code is
//   written against the stable VA-API C ABI but is HARDWARE-UNVALIDATED.
//   The selection layer therefore requires an explicit operator opt-in
//   (AIRTC_VCN_EXPERIMENTAL=1) on top of a successful probe before the
//   session is used for media; the software H.264 codec stays the default
//   (media/codec.py), so the round-1 "advertised path crashes at frame 1"
//   landmine cannot recur.
// - The encoder feeds NV12 via vaDeriveImage+map (one host copy). The
//   zero-copy HIP interop (vaExportSurfaceHandle dmabuf ->
//   hipImportExternalMemory) is the step after hardware validation.
// - SPS/PPS are OUR packed headers (h264::build_sps_pps above), so the SDP
//   advertisement and the bitstream agree byte-for-byte.
// - Decode stays probe-only by design this round: the receive path is
//   covered by the in-repo software decoder, and hardware decode brings
//   no wire-format benefit until an interop box exists to validate on.
// ---------------------------------------------------------------------------
namespace h264 {
std::vector<uint8_t> build_sps_pps(int width, int height);  // defined below
}

namespace vcn {

typedef unsigned int VABufferID;
typedef unsigned int VASurfaceID;
typedef unsigned int VAContextID;
typedef unsigned int VAImageID;

#define VA_INVALID_ID 0xFFFFFFFFu
#define VA_RT_FORMAT_YUV420 0x00000001
#define VA_PROGRESSIVE 0x1
#define VA_RC_CQP 0x00000010

// VAConfigAttribType
#define VA_CFG_ATTRIB_RTFORMAT 0
#define VA_CFG_ATTRIB_RATE_CONTROL 5
struct VAConfigAttrib {
  int type;
  uint32_t value;
};

// VABufferType (va.h enum, ABI-stable ordinals)
#define VA_BUF_EncCoded 21
#define VA_BUF_EncSequenceParameter 22
#define VA_BUF_EncPictureParameter 23
#define VA_BUF_EncSliceParameter 24
#define VA_BUF_EncPackedHeaderParameter 25
#define VA_BUF_EncPackedHeaderData 26

// VAEncPackedHeaderType
#define VA_PACKED_SEQUENCE 1
#define VA_PACKED_PICTURE 2
#define VA_PACKED_SLICE 3

struct VAImageFormat {
  uint32_t fourcc, byte_order, bits_per_pixel, depth;
  uint32_t red_mask, green_mask, blue_mask, alpha_mask;
  uint32_t va_reserved[4];
};
struct VAImage {
  VAImageID image_id;
  VAImageFormat format;
  VABufferID buf;
  uint16_t width, height;
  uint32_t data_size;
  uint32_t num_planes;
  uint32_t pitches[3];
  uint32_t offsets[3];
  int32_t num_palette_entries;
  int32_t entry_bytes;
  int8_t component_order[4];
  uint32_t va_reserved[4];
};
struct VACodedBufferSegment {
  uint32_t size, bit_offset, status, reserved;
  void* buf;
  void* next;
  uint32_t va_reserved[4];
};
struct VAEncPackedHeaderParameterBuffer {
  uint32_t type;
  uint32_t bit_length;
  uint8_t has_emulation_bytes;
  uint32_t va_reserved[4];
};
struct VAPictureH264 {
  VASurfaceID picture_id;
  uint32_t frame_idx;
  uint32_t flags;
  int32_t TopFieldOrderCnt;
  int32_t BottomFieldOrderCnt;
  uint32_t va_reserved[4];
};
#define VA_PICTURE_H264_INVALID 0x00000001

// va_enc_h264.h (ABI-stable since libva 1.x; declared verbatim by layout)
struct VAEncSequenceParameterBufferH264 {
  uint8_t seq_parameter_set_id;
  uint8_t level_idc;
  uint32_t intra_period;
  uint32_t intra_idr_period;
  uint32_t ip_period;
  uint32_t bits_per_second;
  uint32_t max_num_ref_frames;
  uint16_t picture_width_in_mbs;
  uint16_t picture_height_in_mbs;
  union {
    struct {
      uint32_t chroma_format_idc : 2;
      uint32_t frame_mbs_only_flag : 1;
      uint32_t mb_adaptive_frame_field_flag : 1;
      uint32_t seq_scaling_matrix_present_flag : 1;
      uint32_t direct_8x8_inference_flag : 1;
      uint32_t log2_max_frame_num_minus4 : 4;
      uint32_t pic_order_cnt_type : 2;
      uint32_t log2_max_pic_order_cnt_lsb_minus4 : 4;
      uint32_t delta_pic_order_always_zero_flag : 1;
    } bits;
    uint32_t value;
  } seq_fields;
  uint8_t bit_depth_luma_minus8;
  uint8_t bit_depth_chroma_minus8;
  uint8_t num_ref_frames_in_pic_order_cnt_cycle;
  int32_t offset_for_non_ref_pic;
  int32_t offset_for_top_to_bottom_field;
  int32_t offset_for_ref_frame[256];
  uint8_t frame_cropping_flag;
  uint32_t frame_crop_left_offset;
  uint32_t frame_crop_right_offset;
  uint32_t frame_crop_top_offset;
  uint32_t frame_crop_bottom_offset;
  uint8_t vui_parameters_present_flag;
  union {
    struct {
      uint32_t aspect_ratio_info_present_flag : 1;
      uint32_t timing_info_present_flag : 1;
      uint32_t bitstream_restriction_flag : 1;
      uint32_t log2_max_mv_length_horizontal : 5;
      uint32_t log2_max_mv_length_vertical : 5;
      uint32_t fixed_frame_rate_flag : 1;
      uint32_t low_delay_hrd_flag : 1;
      uint32_t motion_vectors_over_pic_boundaries_flag : 1;
      uint32_t reserved : 16;
    } bits;
    uint32_t value;
  } vui_fields;
  uint8_t aspect_ratio_idc;
  uint32_t sar_width;
  uint32_t sar_height;
  uint32_t num_units_in_tick;
  uint32_t time_scale;
  uint32_t va_reserved[4];
};

struct VAEncPictureParameterBufferH264 {
  VAPictureH264 CurrPic;
  VAPictureH264 ReferenceFrames[16];
  VABufferID coded_buf;
  uint8_t pic_parameter_set_id;
  uint8_t seq_parameter_set_id;
  uint8_t last_picture;
  uint16_t frame_num;
  uint8_t pic_init_qp;
  uint8_t num_ref_idx_l0_active_minus1;
  uint8_t num_ref_idx_l1_active_minus1;
  int8_t chroma_qp_index_offset;
  int8_t second_chroma_qp_index_offset;
  union {
    struct {
      uint32_t idr_pic_flag : 1;
      uint32_t reference_pic_flag : 2;
      uint32_t entropy_coding_mode_flag : 1;
      uint32_t weighted_pred_flag : 1;
      uint32_t weighted_bipred_idc : 2;
      uint32_t constrained_intra_pred_flag : 1;
      uint32_t transform_8x8_mode_flag : 1;
      uint32_t deblocking_filter_control_present_flag : 1;
      uint32_t redundant_pic_cnt_present_flag : 1;
      uint32_t pic_order_present_flag : 1;
      uint32_t pic_scaling_matrix_present_flag : 1;
    } bits;
    uint32_t value;
  } pic_fields;
  uint32_t va_reserved[4];
};

struct VAEncSliceParameterBufferH264 {
  uint32_t macroblock_address;
  uint32_t num_macroblocks;
  VABufferID macroblock_info;
  uint8_t slice_type;
  uint8_t pic_parameter_set_id;
  uint16_t idr_pic_id;
  uint16_t pic_order_cnt_lsb;
  int32_t delta_pic_order_cnt_bottom;
  int32_t delta_pic_order_cnt[2];
  uint8_t direct_spatial_mv_pred_flag;
  uint8_t num_ref_idx_active_override_flag;
  uint8_t num_ref_idx_l0_active_minus1;
  uint8_t num_ref_idx_l1_active_minus1;
  VAPictureH264 RefPicList0[32];
  VAPictureH264 RefPicList1[32];
  uint8_t luma_log2_weight_denom;
  uint8_t chroma_log2_weight_denom;
  uint8_t luma_weight_l0_flag;
  int16_t luma_weight_l0[32];
  int16_t luma_offset_l0[32];
  uint8_t chroma_weight_l0_flag;
  int16_t chroma_weight_l0[32][2];
  int16_t chroma_offset_l0[32][2];
  uint8_t luma_weight_l1_flag;
  int16_t luma_weight_l1[32];
  int16_t luma_offset_l1[32];
  uint8_t chroma_weight_l1_flag;
  int16_t chroma_weight_l1[32][2];
  int16_t chroma_offset_l1[32][2];
  uint8_t cabac_init_idc;
  int8_t slice_qp_delta;
  uint8_t disable_deblocking_filter_idc;
  int8_t slice_alpha_c0_offset_div2;
  int8_t slice_beta_offset_div2;
  uint32_t va_reserved[8];
};

struct VaSessionApi {
  VAStatus (*CreateConfig)(VADisplay, VAProfile, VAEntrypoint, VAConfigAttrib*,
                           int, VAConfigID*) = nullptr;
  VAStatus (*DestroyConfig)(VADisplay, VAConfigID) = nullptr;
  VAStatus (*CreateSurfaces)(VADisplay, unsigned int, unsigned int,
                             unsigned int, VASurfaceID*, unsigned int, void*,
                             unsigned int) = nullptr;
  VAStatus (*DestroySurfaces)(VADisplay, VASurfaceID*, int) = nullptr;
  VAStatus (*CreateContext)(VADisplay, VAConfigID, int, int, int, VASurfaceID*,
                            int, VAContextID*) = nullptr;
  VAStatus (*DestroyContext)(VADisplay, VAContextID) = nullptr;
  VAStatus (*CreateBuffer)(VADisplay, VAContextID, int, unsigned int,
                           unsigned int, void*, VABufferID*) = nullptr;
  VAStatus (*DestroyBuffer)(VADisplay, VABufferID) = nullptr;
  VAStatus (*MapBuffer)(VADisplay, VABufferID, void**) = nullptr;
  VAStatus (*UnmapBuffer)(VADisplay, VABufferID) = nullptr;
  VAStatus (*BeginPicture)(VADisplay, VAContextID, VASurfaceID) = nullptr;
  VAStatus (*RenderPicture)(VADisplay, VAContextID, VABufferID*, int) = nullptr;
  VAStatus (*EndPicture)(VADisplay, VAContextID) = nullptr;
  VAStatus (*SyncSurface)(VADisplay, VASurfaceID) = nullptr;
  VAStatus (*DeriveImage)(VADisplay, VASurfaceID, VAImage*) = nullptr;
  VAStatus (*DestroyImage)(VADisplay, VAImageID) = nullptr;

  bool load(void* h_va, std::string* err) {
#define RESOLVE2(field, name)                                        \
  field = reinterpret_cast<decltype(field)>(dlsym(h_va, name));      \
  if (!field) {                                                      \
    *err = std::string("missing symbol ") + name;                    \
    return false;                                                    \
  }
    RESOLVE2(CreateConfig, "vaCreateConfig");
    RESOLVE2(DestroyConfig, "vaDestroyConfig");
    RESOLVE2(CreateSurfaces, "vaCreateSurfaces");
    RESOLVE2(DestroySurfaces, "vaDestroySurfaces");
    RESOLVE2(CreateContext, "vaCreateContext");
    RESOLVE2(DestroyContext, "vaDestroyContext");
    RESOLVE2(CreateBuffer, "vaCreateBuffer");
    RESOLVE2(DestroyBuffer, "vaDestroyBuffer");
    RESOLVE2(MapBuffer, "vaMapBuffer");
    RESOLVE2(UnmapBuffer, "vaUnmapBuffer");
    RESOLVE2(BeginPicture, "vaBeginPicture");
    RESOLVE2(RenderPicture, "vaRenderPicture");
    RESOLVE2(EndPicture, "vaEndPicture");
    RESOLVE2(SyncSurface, "vaSyncSurface");
    RESOLVE2(DeriveImage, "vaDeriveImage");
    RESOLVE2(DestroyImage, "vaDestroyImage");
#undef RESOLVE2
    return true;
  }
};

class EncodeSession {
 public:
  std::string error;

  EncodeSession(int width, int height) : w_(width), h_(height) {
    mbw_ = (width + 15) / 16;
    mbh_ = (height + 15) / 16;
  }
  ~EncodeSession() { close(); }

  bool open() {
    std::string err;
    if (!va_.load(&err)) {
      error = err;
      return false;
    }
    for (int node = 128; node < 136; ++node) {
      std::string dev = "/dev/dri/renderD" + std::to_string(node);
      fd_ = ::open(dev.c_str(), O_RDWR);
      if (fd_ < 0) continue;
      dpy_ = va_.GetDisplayDRM(fd_);
      int mj, mn;
      if (dpy_ && va_.Initialize(dpy_, &mj, &mn) == VA_STATUS_SUCCESS) break;
      ::close(fd_);
      fd_ = -1;
      dpy_ = nullptr;
    }
    if (!dpy_) {
      error = "no VA display";
      return false;
    }
    if (!api_.load(va_.h_va, &error)) return false;

    VAConfigAttrib attribs[2] = {
        {VA_CFG_ATTRIB_RTFORMAT, VA_RT_FORMAT_YUV420},
        {VA_CFG_ATTRIB_RATE_CONTROL, VA_RC_CQP},
    };
    // constrained baseline first (matches our SPS), then main
    for (VAProfile prof : {13 /*ConstrainedBaseline*/, VA_PROFILE_H264_MAIN}) {
      if (api_.CreateConfig(dpy_, prof, VA_ENTRYPOINT_ENCSLICE, attribs, 2,
                            &config_) == VA_STATUS_SUCCESS)
        break;
      if (api_.CreateConfig(dpy_, prof, VA_ENTRYPOINT_ENCSLICE_LP, attribs, 2,
                            &config_) == VA_STATUS_SUCCESS)
        break;
    }
    if (config_ == VA_INVALID_ID) {
      error = "vaCreateConfig failed (no H.264 encode entrypoint)";
      return false;
    }
    if (api_.CreateSurfaces(dpy_, VA_RT_FORMAT_YUV420, mbw_ * 16, mbh_ * 16,
                            surfaces_, 2, nullptr, 0) != VA_STATUS_SUCCESS) {
      error = "vaCreateSurfaces failed";
      return false;
    }
    if (api_.CreateContext(dpy_, config_, mbw_ * 16, mbh_ * 16, VA_PROGRESSIVE,
                           surfaces_, 2, &ctx_) != VA_STATUS_SUCCESS) {
      error = "vaCreateContext failed";
      return false;
    }
    if (api_.CreateBuffer(dpy_, ctx_, VA_BUF_EncCoded,
                          (unsigned)(mbw_ * mbh_ * 400 + 4096), 1, nullptr,
                          &coded_buf_) != VA_STATUS_SUCCESS) {
      error = "coded buffer allocation failed";
      return false;
    }
    opened_ = true;
    return true;
  }

  // rgb: packed RGB24, w*h*3 bytes. Returns Annex-B bytes in out (cap) or
  // negative on failure.
  int encode_idr(const uint8_t* rgb, int qp, uint8_t* out, int cap) {
    if (!opened_) return -1;
    VASurfaceID surf = surfaces_[frame_count_ & 1];
    if (!upload_nv12(rgb, surf)) return -2;

    std::vector<VABufferID> bufs;
    auto mkbuf = [&](int type, const void* data, unsigned size) -> bool {
      VABufferID id;
      if (api_.CreateBuffer(dpy_, ctx_, type, size, 1, const_cast<void*>(data),
                            &id) != VA_STATUS_SUCCESS)
        return false;
      bufs.push_back(id);
      return true;
    };

    VAEncSequenceParameterBufferH264 seq;
    memset(&seq, 0, sizeof(seq));
    seq.level_idc = 31;
    seq.intra_period = 1;  // every frame IDR (matches the sw codec policy)
    seq.intra_idr_period = 1;
    seq.ip_period = 1;
    seq.max_num_ref_frames = 1;
    seq.picture_width_in_mbs = (uint16_t)mbw_;
    seq.picture_height_in_mbs = (uint16_t)mbh_;
    seq.seq_fields.bits.chroma_format_idc = 1;
    seq.seq_fields.bits.frame_mbs_only_flag = 1;
    seq.seq_fields.bits.direct_8x8_inference_flag = 1;
    seq.seq_fields.bits.log2_max_frame_num_minus4 = 0;
    seq.seq_fields.bits.pic_order_cnt_type = 2;
    if (mbw_ * 16 != w_ || mbh_ * 16 != h_) {
      seq.frame_cropping_flag = 1;
      seq.frame_crop_right_offset = (mbw_ * 16 - w_) / 2;
      seq.frame_crop_bottom_offset = (mbh_ * 16 - h_) / 2;
    }

    VAEncPictureParameterBufferH264 pic;
    memset(&pic, 0, sizeof(pic));
    pic.CurrPic.picture_id = surf;
    pic.CurrPic.TopFieldOrderCnt = 0;
    for (auto& r : pic.ReferenceFrames) {
      r.picture_id = VA_INVALID_ID;
      r.flags = VA_PICTURE_H264_INVALID;
    }
    pic.coded_buf = coded_buf_;
    pic.frame_num = 0;
    pic.pic_init_qp = (uint8_t)qp;
    pic.pic_fields.bits.idr_pic_flag = 1;
    pic.pic_fields.bits.reference_pic_flag = 1;
    pic.pic_fields.bits.entropy_coding_mode_flag = 0;  // CAVLC
    pic.pic_fields.bits.deblocking_filter_control_present_flag = 1;

    VAEncSliceParameterBufferH264 slice;
    memset(&slice, 0, sizeof(slice));
    slice.num_macroblocks = (uint32_t)(mbw_ * mbh_);
    slice.macroblock_info = VA_INVALID_ID;
    slice.slice_type = 2;  // I
    slice.idr_pic_id = (uint16_t)(frame_count_ & 1);
    for (auto& r : slice.RefPicList0) {
      r.picture_id = VA_INVALID_ID;
      r.flags = VA_PICTURE_H264_INVALID;
    }
    for (auto& r : slice.RefPicList1) {
      r.picture_id = VA_INVALID_ID;
      r.flags = VA_PICTURE_H264_INVALID;
    }
    slice.slice_qp_delta = (int8_t)(qp - pic.pic_init_qp);

    // packed SPS+PPS: our own bitstream generator, so the stream matches
    // what the SDP/signalling layer advertises
    std::vector<uint8_t> hdr = h264::build_sps_pps(w_, h_);
    VAEncPackedHeaderParameterBuffer ph;
    memset(&ph, 0, sizeof(ph));
    ph.type = VA_PACKED_SEQUENCE;
    ph.bit_length = (uint32_t)hdr.size() * 8;
    ph.has_emulation_bytes = 1;

    bool ok = mkbuf(VA_BUF_EncSequenceParameter, &seq, sizeof(seq)) &&
              mkbuf(VA_BUF_EncPackedHeaderParameter, &ph, sizeof(ph)) &&
              mkbuf(VA_BUF_EncPackedHeaderData, hdr.data(), (unsigned)hdr.size()) &&
              mkbuf(VA_BUF_EncPictureParameter, &pic, sizeof(pic)) &&
              mkbuf(VA_BUF_EncSliceParameter, &slice, sizeof(slice));
    int rc = -3;
    if (ok && api_.BeginPicture(dpy_, ctx_, surf) == VA_STATUS_SUCCESS &&
        api_.RenderPicture(dpy_, ctx_, bufs.data(), (int)bufs.size()) ==
            VA_STATUS_SUCCESS &&
        api_.EndPicture(dpy_, ctx_) == VA_STATUS_SUCCESS &&
        api_.SyncSurface(dpy_, surf) == VA_STATUS_SUCCESS) {
      void* seg_raw = nullptr;
      if (api_.MapBuffer(dpy_, coded_buf_, &seg_raw) == VA_STATUS_SUCCESS) {
        int total = 0;
        for (auto* seg = (VACodedBufferSegment*)seg_raw; seg;
             seg = (VACodedBufferSegment*)seg->next) {
          if (total + (int)seg->size > cap) {
            total = -4;
            break;
          }
          memcpy(out + total, seg->buf, seg->size);
          total += (int)seg->size;
        }
        api_.UnmapBuffer(dpy_, coded_buf_);
        rc = total;
      }
    }
    for (VABufferID b : bufs) api_.DestroyBuffer(dpy_, b);
    ++frame_count_;
    return rc;
  }

  void close() {
    if (dpy_) {
      if (coded_buf_ != VA_INVALID_ID) api_.DestroyBuffer(dpy_, coded_buf_);
      if (ctx_ != VA_INVALID_ID) api_.DestroyContext(dpy_, ctx_);
      if (surfaces_[0] != VA_INVALID_ID)
        api_.DestroySurfaces(dpy_, surfaces_, 2);
      if (config_ != VA_INVALID_ID) api_.DestroyConfig(dpy_, config_);
      va_.Terminate(dpy_);
      dpy_ = nullptr;
    }
    if (fd_ >= 0) {
      ::close(fd_);
      fd_ = -1;
    }
    opened_ = false;
  }

 private:
  bool upload_nv12(const uint8_t* rgb, VASurfaceID surf) {
    VAImage img;
    if (api_.DeriveImage(dpy_, surf, &img) != VA_STATUS_SUCCESS) return false;
    void* base = nullptr;
    bool ok = api_.MapBuffer(dpy_, img.buf, &base) == VA_STATUS_SUCCESS;
    if (ok) {
      uint8_t* y = (uint8_t*)base + img.offsets[0];
      uint8_t* uv = (uint8_t*)base + img.offsets[1];
      const int yp = (int)img.pitches[0], cp = (int)img.pitches[1];
      for (int j = 0; j < mbh_ * 16; ++j) {
        int sj = j < h_ ? j : h_ - 1;
        for (int i = 0; i < mbw_ * 16; ++i) {
          int si = i < w_ ? i : w_ - 1;
          const uint8_t* p = rgb + (sj * w_ + si) * 3;
          int R = p[0], G = p[1], B = p[2];
          y[j * yp + i] =
              (uint8_t)(((66 * R + 129 * G + 25 * B + 128) >> 8) + 16);
        }
      }
      for (int j = 0; j < mbh_ * 8; ++j) {
        for (int i = 0; i < mbw_ * 8; ++i) {
          int sj = std::min(2 * j, h_ - 1), si = std::min(2 * i, w_ - 1);
          const uint8_t* p = rgb + (sj * w_ + si) * 3;
          int R = p[0], G = p[1], B = p[2];
          uv[j * cp + 2 * i] =
              (uint8_t)(((-38 * R - 74 * G + 112 * B + 128) >> 8) + 128);
          uv[j * cp + 2 * i + 1] =
              (uint8_t)(((112 * R - 94 * G - 18 * B + 128) >> 8) + 128);
        }
      }
      api_.UnmapBuffer(dpy_, img.buf);
    }
    api_.DestroyImage(dpy_, img.image_id);
    return ok;
  }

  int w_, h_, mbw_, mbh_;
  int fd_ = -1;
  VADisplay dpy_ = nullptr;
  VaApi va_;
  VaSessionApi api_;
  VAConfigID config_ = VA_INVALID_ID;
  VAContextID ctx_ = VA_INVALID_ID;
  VASurfaceID surfaces_[2] = {VA_INVALID_ID, VA_INVALID_ID};
  VABufferID coded_buf_ = VA_INVALID_ID;
  bool opened_ = false;
  uint64_t frame_count_ = 0;
};

}  // namespace vcn

// C ABI: VCN encode session lifecycle
extern "C" void* airtc_vcn_enc_create(int w, int h, char* errbuf, int errlen) {
  auto* s = new vcn::EncodeSession(w, h);
  if (!s->open()) {
    if (errbuf && errlen > 0) {
      strncpy(errbuf, s->error.c_str(), errlen - 1);
      errbuf[errlen - 1] = 0;
    }
    delete s;
    return nullptr;
  }
  return s;
}
extern "C" void airtc_vcn_enc_destroy(void* h) {
  delete (vcn::EncodeSession*)h;
}
extern "C" int airtc_vcn_enc_encode(void* h, const uint8_t* rgb, int qp,
                                    uint8_t* out, int cap) {
  return ((vcn::EncodeSession*)h)->encode_idr(rgb, qp, out, cap);
}

// C ABI for the binding layer (kernels.h): fills a human-readable summary,
// returns bit0 = decode available, bit1 = encode available, -1 = none.
extern "C" int airtc_vcn_probe(char* buf, int buflen) {
  vcn::ProbeResult r = vcn::probe();
  std::string s = "stage=" + r.stage + " device=" + (r.device.empty() ? "-" : r.device) +
                  " va=" + std::to_string(r.va_major) + "." + std::to_string(r.va_minor) +
                  " detail=" + r.detail;
  strncpy(buf, s.c_str(), buflen - 1);
  buf[buflen - 1] = 0;
  if (!r.available) return -1;
  return (r.h264_decode ? 1 : 0) | (r.h264_encode ? 2 : 0);
}

// ---------------------------------------------------------------------------
// H.264 bitstream layer (encoder side): SPS/PPS generation with exp-Golomb
// coding and emulation prevention. This is the hardware-independent half of
// the VCN encode session — VA-API encodes slices, but the parameter sets
// (and their RTP/SDP advertisement) are the application's job. Baseline
// profile, progressive, CAVLC: the low-latency WebRTC configuration.
// ---------------------------------------------------------------------------
namespace h264 {

struct BitWriter {
  std::vector<uint8_t> bytes;
  uint32_t cur = 0;
  int nbits = 0;

  void put(uint32_t value, int width) {
    for (int i = width - 1; i >= 0; --i) {
      cur = (cur << 1) | ((value >> i) & 1);
      if (++nbits == 8) {
        bytes.push_back((uint8_t)cur);
        cur = 0;
        nbits = 0;
      }
    }
  }
  void ue(uint32_t v) {  // unsigned exp-Golomb
    uint32_t vp1 = v + 1;
    int lead = 0;
    for (uint32_t t = vp1; t > 1; t >>= 1) ++lead;
    put(0, lead);
    put(vp1, lead + 1);
  }
  void se(int32_t v) {  // signed exp-Golomb
    ue(v <= 0 ? (uint32_t)(-2 * v) : (uint32_t)(2 * v - 1));
  }
  void rbsp_trailing() {
    put(1, 1);
    if (nbits) put(0, 8 - nbits);
  }
};

// RBSP -> NAL with emulation prevention (00 00 {00,01,02,03} -> 00 00 03 xx)
inline void emit_nal(std::vector<uint8_t>* out, uint8_t nal_header,
                     const std::vector<uint8_t>& rbsp) {
  out->insert(out->end(), {0, 0, 0, 1, nal_header});
  int zeros = 0;
  for (uint8_t b : rbsp) {
    if (zeros >= 2 && b <= 3) {
      out->push_back(3);
      zeros = 0;
    }
    out->push_back(b);
    zeros = (b == 0) ? zeros + 1 : 0;
  }
}

std::vector<uint8_t> build_sps_pps(int width, int height) {
  std::vector<uint8_t> out;
  // --- SPS (nal 7, baseline/constrained) ---
  {
    BitWriter w;
    w.put(66, 8);   // profile_idc: baseline
    w.put(0xC0, 8); // constraint_set0+1, reserved
    w.put(31, 8);   // level_idc 3.1
    w.ue(0);        // sps_id
    w.ue(0);        // log2_max_frame_num_minus4
    w.ue(2);        // pic_order_cnt_type: 2 (low-latency, no reordering)
    w.ue(1);        // max_num_ref_frames
    w.put(0, 1);    // gaps_in_frame_num_value_allowed
    const int mbs_w = (width + 15) / 16, mbs_h = (height + 15) / 16;
    w.ue(mbs_w - 1);
    w.ue(mbs_h - 1);
    w.put(1, 1);    // frame_mbs_only
    w.put(1, 1);    // direct_8x8_inference
    const int crop_r = mbs_w * 16 - width, crop_b = mbs_h * 16 - height;
    if (crop_r || crop_b) {
      w.put(1, 1);  // frame_cropping
      w.ue(0); w.ue(crop_r / 2); w.ue(0); w.ue(crop_b / 2);
    } else {
      w.put(0, 1);
    }
    w.put(0, 1);    // vui_parameters_present
    w.rbsp_trailing();
    emit_nal(&out, 0x67, w.bytes);
  }
  // --- PPS (nal 8) ---
  {
    BitWriter w;
    w.ue(0);        // pps_id
    w.ue(0);        // sps_id
    w.put(0, 1);    // entropy_coding_mode: CAVLC
    w.put(0, 1);    // bottom_field_pic_order_in_frame_present
    w.ue(0);        // num_slice_groups_minus1
    w.ue(0);        // num_ref_idx_l0_default_active_minus1
    w.ue(0);        // num_ref_idx_l1_default_active_minus1
    w.put(0, 1);    // weighted_pred
    w.put(0, 2);    // weighted_bipred_idc
    w.se(0);        // pic_init_qp_minus26
    w.se(0);        // pic_init_qs_minus26
    w.se(0);        // chroma_qp_index_offset
    w.put(1, 1);    // deblocking_filter_control_present
    w.put(0, 1);    // constrained_intra_pred
    w.put(0, 1);    // redundant_pic_cnt_present
    w.rbsp_trailing();
    emit_nal(&out, 0x68, w.bytes);
  }
  return out;
}

}  // namespace h264

// C ABI: fills buf with Annex-B SPS+PPS for (width, height); returns length
// or -1 if the buffer is too small.
extern "C" int airtc_h264_sps_pps(int width, int height, uint8_t* buf,
                                  int buflen) {
  std::vector<uint8_t> v = h264::build_sps_pps(width, height);
  if ((int)v.size() > buflen) return -1;
  memcpy(buf, v.data(), v.size());
  return (int)v.size();
}
