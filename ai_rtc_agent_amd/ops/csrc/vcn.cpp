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

inline std::vector<uint8_t> build_sps_pps(int width, int height) {
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
