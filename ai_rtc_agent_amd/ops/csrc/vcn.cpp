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
