// Torch bindings for the MI355X HIP kernels (host-side translation unit).
//
// Thin layer: validates tensors, allocates outputs through the torch caching
// allocator (hipGraph-capture safe), and calls the extern "C" launchers from
// kernels.h on the current stream. No compute logic lives here.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "kernels.h"

#define CHECK_IN(x)                                                     \
  TORCH_CHECK(x.is_cuda(), #x " must be on GPU");                       \
  TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

namespace {

hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

const uint16_t* h_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const uint16_t*>(t.data_ptr<at::Half>());
}
uint16_t* h_ptr_mut(torch::Tensor& t) {
  return reinterpret_cast<uint16_t*>(t.data_ptr<at::Half>());
}

torch::Tensor conv2d(torch::Tensor x, torch::Tensor w_perm,
                     c10::optional<torch::Tensor> bias,
                     c10::optional<torch::Tensor> cbias,
                     c10::optional<torch::Tensor> residual, int64_t R,
                     int64_t S, int64_t stride, int64_t pad, int64_t act,
                     c10::optional<torch::Tensor> in_affine, int64_t in_act,
                     c10::optional<torch::Tensor> counters) {
  CHECK_IN(x);
  CHECK_IN(w_perm);
  TORCH_CHECK(x.dtype() == torch::kHalf && w_perm.dtype() == torch::kHalf);
  const int B = x.size(0), H = x.size(1), W = x.size(2), IC = x.size(3);
  const int OC = w_perm.size(0);
  const int HO = (H + 2 * (int)pad - (int)R) / (int)stride + 1;
  const int WO = (W + 2 * (int)pad - (int)S) / (int)stride + 1;
  auto out = torch::empty({B, HO, WO, OC}, x.options());
  const float* bp = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->dtype() == torch::kFloat && bias->is_contiguous());
    bp = bias->data_ptr<float>();
  }
  const uint16_t* cb = nullptr;
  if (cbias.has_value()) {
    CHECK_IN((*cbias));
    TORCH_CHECK(cbias->dtype() == torch::kHalf);
    cb = h_ptr(*cbias);
  }
  const uint16_t* res = nullptr;
  if (residual.has_value()) {
    CHECK_IN((*residual));
    TORCH_CHECK(residual->dtype() == torch::kHalf);
    TORCH_CHECK(residual->numel() == out.numel(), "residual shape mismatch");
    res = h_ptr(*residual);
  }
  const float* aff = nullptr;
  if (in_affine.has_value()) {
    CHECK_IN((*in_affine));
    TORCH_CHECK(in_affine->dtype() == torch::kFloat &&
                    in_affine->numel() == (long)B * IC * 2,
                "in_affine must be (B, IC, 2) f32");
    aff = in_affine->data_ptr<float>();
  }
  const int path = airtc_conv2d_splitk_for(B, HO, WO, OC, IC);
  if (path == 0) {
    airtc_conv2d_direct(h_ptr(x), h_ptr(w_perm), bp, cb, res, h_ptr_mut(out),
                        B, H, W, IC, HO, WO, OC, (int)R, (int)S, (int)stride,
                        (int)pad, (int)act, aff, (int)in_act, cur_stream());
    return out;
  }
  float* wsp = nullptr;
  torch::Tensor ws;
  const int splitk = path == 100 ? 1 : (path > 0 ? path : -path);
  if (splitk > 1) {
    ws = torch::empty({(long)B * splitk, (long)HO * WO, OC},
                      x.options().dtype(torch::kFloat));
    wsp = ws.data_ptr<float>();
  }
  int* cnt = nullptr;
  if (counters.has_value() && splitk > 1) {
    CHECK_IN((*counters));
    TORCH_CHECK(counters->dtype() == torch::kInt, "counters must be int32");
    TORCH_CHECK(counters->numel() >= (long)B * ((HO * WO + 63) / 64) *
                                         ((OC + 63) / 64),
                "counters buffer too small");
    cnt = counters->data_ptr<int>();
  }
  airtc_conv2d_mfma(h_ptr(x), h_ptr(w_perm), bp, cb, res, h_ptr_mut(out), wsp,
                    B, H, W, IC, HO, WO, OC, (int)R, (int)S, (int)stride,
                    (int)pad, (int)act, path, aff, (int)in_act, cnt,
                    cur_stream());
  return out;
}

torch::Tensor group_norm_coeffs(torch::Tensor x, int64_t groups,
                                torch::Tensor gamma, torch::Tensor beta,
                                double eps) {
  CHECK_IN(x);
  const int B = x.size(0);
  const int C = x.size(-1);
  const long HW = x.numel() / ((long)B * C);
  auto coeffs = torch::empty({B, C, 2}, x.options().dtype(torch::kFloat));
  const int nchunk = airtc_group_norm_nchunk(B, (int)groups);
  auto ws = torch::empty({(long)B * groups * nchunk * 2},
                         x.options().dtype(torch::kFloat));
  airtc_group_norm_coeffs(h_ptr(x), gamma.data_ptr<float>(),
                          beta.data_ptr<float>(), coeffs.data_ptr<float>(),
                          ws.data_ptr<float>(), B, (int)HW, C, (int)groups,
                          (float)eps, cur_stream());
  return coeffs;
}

torch::Tensor group_norm_silu_fp8(torch::Tensor x, int64_t groups,
                                  torch::Tensor gamma, torch::Tensor beta,
                                  double eps, int64_t act, double a_scale) {
  CHECK_IN(x);
  TORCH_CHECK(x.dtype() == torch::kHalf);
  const int B = x.size(0);
  const int C = x.size(-1);
  const long HW = x.numel() / ((long)B * C);
  auto out = torch::empty_like(x, x.options().dtype(torch::kUInt8));
  const int nchunk = airtc_group_norm_nchunk(B, (int)groups);
  auto ws = torch::empty({(long)B * groups * nchunk * 2},
                         x.options().dtype(torch::kFloat));
  airtc_group_norm_silu_fp8(h_ptr(x), gamma.data_ptr<float>(),
                            beta.data_ptr<float>(), out.data_ptr<uint8_t>(),
                            ws.data_ptr<float>(), B, (int)HW, C, (int)groups,
                            (float)eps, (int)act, (float)a_scale,
                            cur_stream());
  return out;
}

torch::Tensor group_norm_silu(torch::Tensor x, int64_t groups,
                              torch::Tensor gamma, torch::Tensor beta,
                              double eps, int64_t act) {
  CHECK_IN(x);
  const int B = x.size(0);
  const int C = x.size(-1);
  const long HW = x.numel() / ((long)B * C);
  auto out = torch::empty_like(x);
  const int nchunk = airtc_group_norm_nchunk(B, (int)groups);
  auto ws = torch::empty({(long)B * groups * nchunk * 2},
                         x.options().dtype(torch::kFloat));
  airtc_group_norm_silu(h_ptr(x), gamma.data_ptr<float>(),
                        beta.data_ptr<float>(), h_ptr_mut(out),
                        ws.data_ptr<float>(), B, (int)HW, C, (int)groups,
                        (float)eps, (int)act, cur_stream());
  return out;
}

torch::Tensor layer_norm(torch::Tensor x, torch::Tensor gamma,
                         torch::Tensor beta, double eps) {
  CHECK_IN(x);
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  auto out = torch::empty_like(x);
  airtc_layer_norm(h_ptr(x), gamma.data_ptr<float>(), beta.data_ptr<float>(),
                   h_ptr_mut(out), rows, C, (float)eps, cur_stream());
  return out;
}

torch::Tensor attention_bhlc(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                             int64_t heads, double scale) {
  // q/k/v: (B, L, C) with unit channel stride; row/batch strides are free so
  // chunk() views of a fused QKV projection run zero-copy.
  TORCH_CHECK(q.is_cuda() && q.stride(2) == 1, "q: unit channel stride");
  TORCH_CHECK(k.stride(2) == 1 && v.stride(2) == 1, "k/v: unit channel stride");
  TORCH_CHECK(k.strides() == v.strides(), "k/v must share layout");
  const int B = q.size(0), Lq = q.size(1), C = q.size(2);
  const int Lk = k.size(1), Ck = k.size(2);
  const int d = C / (int)heads;
  TORCH_CHECK(Ck == C, "q/k channel mismatch");
  TORCH_CHECK(d % 32 == 0 && d <= 160, "head_dim must be padded to one of 32/64/96/128/160");
  auto out = torch::empty({B, Lq, C}, q.options());
  airtc_attention(h_ptr(q), h_ptr(k), h_ptr(v), h_ptr_mut(out), B, (int)heads,
                  Lq, Lk, d, q.stride(0), d, q.stride(1), k.stride(0), d,
                  k.stride(1), (long)Lq * C, d, C, (float)scale, cur_stream());
  return out;
}

torch::Tensor silu(torch::Tensor x) {
  CHECK_IN(x);
  auto out = torch::empty_like(x);
  airtc_silu_f16(h_ptr(x), h_ptr_mut(out), x.numel(), cur_stream());
  return out;
}

torch::Tensor geglu(torch::Tensor x) {
  CHECK_IN(x);
  const long inner = x.size(-1) / 2;
  const long rows = x.numel() / (2 * inner);
  auto sizes = x.sizes().vec();
  sizes.back() = inner;
  auto out = torch::empty(sizes, x.options());
  airtc_geglu_f16(h_ptr(x), h_ptr_mut(out), rows, inner, cur_stream());
  return out;
}

torch::Tensor add_act(torch::Tensor a, torch::Tensor b, int64_t act) {
  CHECK_IN(a);
  CHECK_IN(b);
  auto out = torch::empty_like(a);
  airtc_add_act_f16(h_ptr(a), h_ptr(b), h_ptr_mut(out), a.numel(), (int)act,
                    cur_stream());
  return out;
}

torch::Tensor upsample2x(torch::Tensor x) {
  CHECK_IN(x);
  const int B = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  auto out = torch::empty({B, 2 * H, 2 * W, C}, x.options());
  airtc_upsample2x_f16(h_ptr(x), h_ptr_mut(out), B, H, W, C, cur_stream());
  return out;
}

torch::Tensor preprocess_u8(torch::Tensor frame) {
  CHECK_IN(frame);
  TORCH_CHECK(frame.dtype() == torch::kUInt8);
  auto out = torch::empty(frame.sizes(),
                          frame.options().dtype(torch::kHalf));
  airtc_preprocess_u8(frame.data_ptr<uint8_t>(),
                      reinterpret_cast<uint16_t*>(out.data_ptr<at::Half>()),
                      frame.numel(), cur_stream());
  return out;
}

torch::Tensor postprocess_u8(torch::Tensor img) {
  CHECK_IN(img);
  TORCH_CHECK(img.dtype() == torch::kHalf);
  auto out = torch::empty(img.sizes(), img.options().dtype(torch::kUInt8));
  airtc_postprocess_u8(h_ptr(img), out.data_ptr<uint8_t>(), img.numel(),
                       cur_stream());
  return out;
}

// --- software H.264 baseline-intra codec (h264sw.cpp) -----------------
extern "C" {
void* airtc_h264enc_create(int w, int h);
void airtc_h264enc_set_slices(void*, int n);
void airtc_h264enc_destroy(void*);
int airtc_h264enc_encode(void*, const uint8_t*, int qp, uint8_t*, int cap);
int airtc_h264enc_encode_ex(void*, const uint8_t*, int qp, int force_idr,
                            uint8_t*, int cap);
void airtc_h264enc_set_mb_mode(void*, int m);
void airtc_h264_pred4(int, const uint8_t*, const uint8_t*, uint8_t, int, int,
                      int, uint8_t*);
void* airtc_h264dec_create();
void airtc_h264dec_destroy(void*);
int airtc_h264dec_decode(void*, const uint8_t*, int, uint8_t*, int, int*, int*);
void airtc_h264dec_dims(void*, int*, int*);
int airtc_h264sw_table_check();
}

class H264SwEncoder {
 public:
  H264SwEncoder(int w, int h, int slices = 4, int mb_mode = 0)
      : w_(w), h_(h) {
    handle_ = airtc_h264enc_create(w, h);
    TORCH_CHECK(handle_, "invalid encoder dimensions");
    airtc_h264enc_set_slices(handle_, slices);
    airtc_h264enc_set_mb_mode(handle_, mb_mode);
  }
  ~H264SwEncoder() {
    if (handle_) airtc_h264enc_destroy(handle_);
  }
  H264SwEncoder(const H264SwEncoder&) = delete;
  // keyframe=false emits a P frame (P_Skip + intra refresh) once a
  // reference exists; true (default) emits an IDR with in-band SPS/PPS
  pybind11::bytes encode(pybind11::bytes rgb, int qp, bool keyframe = true) {
    std::string s(rgb);
    TORCH_CHECK((int)s.size() == w_ * h_ * 3, "rgb buffer size mismatch");
    std::vector<uint8_t> out((size_t)w_ * h_ * 6 + 4096);
    int n;
    {
      pybind11::gil_scoped_release nogil;
      n = airtc_h264enc_encode_ex(handle_, (const uint8_t*)s.data(), qp,
                                  keyframe ? 1 : 0, out.data(),
                                  (int)out.size());
    }
    TORCH_CHECK(n > 0, "h264 encode failed rc=", n);
    return pybind11::bytes((const char*)out.data(), n);
  }

 private:
  void* handle_;
  int w_, h_;
};

class H264SwDecoder {
 public:
  H264SwDecoder() { handle_ = airtc_h264dec_create(); }
  ~H264SwDecoder() {
    if (handle_) airtc_h264dec_destroy(handle_);
  }
  H264SwDecoder(const H264SwDecoder&) = delete;
  // returns (rgb_bytes, w, h) or None (no frame in this AU / undecodable)
  pybind11::object decode(pybind11::bytes data) {
    std::string s(data);
    int w = 0, h = 0, rc;
    {
      pybind11::gil_scoped_release nogil;
      rc = airtc_h264dec_decode(handle_, (const uint8_t*)s.data(),
                                (int)s.size(), buf_.data(), (int)buf_.size(),
                                &w, &h);
      if (rc == -8) {  // output buffer too small: size from the parsed SPS
        int dw = 0, dh = 0;
        airtc_h264dec_dims(handle_, &dw, &dh);
        if (dw > 0 && dh > 0) {
          buf_.resize((size_t)dw * dh * 3);
          rc = airtc_h264dec_decode(handle_, (const uint8_t*)s.data(),
                                    (int)s.size(), buf_.data(),
                                    (int)buf_.size(), &w, &h);
        }
      }
    }
    if (rc != 1) return pybind11::none();
    return pybind11::make_tuple(
        pybind11::bytes((const char*)buf_.data(), (size_t)w * h * 3), w, h);
  }

 private:
  void* handle_;
  std::vector<uint8_t> buf_ = std::vector<uint8_t>((size_t)1024 * 1024 * 3);
};

// --- VCN hardware encode session (vcn.cpp; hardware-unvalidated, see the
// scope note there — gated behind AIRTC_VCN_EXPERIMENTAL in media/codec.py)
extern "C" {
void* airtc_vcn_enc_create(int w, int h, char* errbuf, int errlen);
void airtc_vcn_enc_destroy(void*);
int airtc_vcn_enc_encode(void*, const uint8_t*, int qp, uint8_t*, int cap);
}

class VcnEncoder {
 public:
  VcnEncoder(int w, int h) : w_(w), h_(h) {
    char err[256] = {0};
    handle_ = airtc_vcn_enc_create(w, h, err, sizeof(err));
    if (!handle_) throw std::runtime_error(std::string("VCN encode: ") + err);
  }
  ~VcnEncoder() {
    if (handle_) airtc_vcn_enc_destroy(handle_);
  }
  VcnEncoder(const VcnEncoder&) = delete;
  pybind11::bytes encode(pybind11::bytes rgb, int qp) {
    std::string s(rgb);
    TORCH_CHECK((int)s.size() == w_ * h_ * 3, "rgb buffer size mismatch");
    std::vector<uint8_t> out((size_t)w_ * h_ * 2 + 65536);
    int n;
    {
      pybind11::gil_scoped_release nogil;
      n = airtc_vcn_enc_encode(handle_, (const uint8_t*)s.data(), qp,
                               out.data(), (int)out.size());
    }
    TORCH_CHECK(n > 0, "VCN encode failed rc=", n);
    return pybind11::bytes((const char*)out.data(), n);
  }

 private:
  void* handle_;
  int w_, h_;
};

pybind11::bytes h264_sps_pps(int width, int height) {
  uint8_t buf[256];
  int n = airtc_h264_sps_pps(width, height, buf, sizeof(buf));
  TORCH_CHECK(n > 0, "sps/pps generation failed");
  return pybind11::bytes(reinterpret_cast<const char*>(buf), n);
}

pybind11::dict vcn_probe() {
  char buf[512];
  int rc = airtc_vcn_probe(buf, sizeof(buf));
  pybind11::dict d;
  d["available"] = rc >= 0;
  d["h264_decode"] = rc >= 0 && (rc & 1);
  d["h264_encode"] = rc >= 0 && (rc & 2);
  d["detail"] = std::string(buf);
  return d;
}

// fused LCM scheduler math (elementwise.hip): one kernel per scheduler op
torch::Tensor sched_add_noise(torch::Tensor x0, torch::Tensor noise,
                              torch::Tensor a, torch::Tensor bt) {
  CHECK_IN(x0);
  CHECK_IN(noise);
  TORCH_CHECK(x0.dtype() == torch::kHalf && noise.dtype() == torch::kHalf);
  TORCH_CHECK(a.dtype() == torch::kFloat && bt.dtype() == torch::kFloat);
  const long B = x0.size(0);
  TORCH_CHECK(a.numel() == B && bt.numel() == B, "coeffs must be (B,)");
  const long per_b = x0.numel() / B;
  TORCH_CHECK(per_b % 8 == 0, "row size must be a multiple of 8");
  auto out = torch::empty_like(x0);
  airtc_sched_add_noise(h_ptr(x0), h_ptr(noise), a.data_ptr<float>(),
                        bt.data_ptr<float>(), h_ptr_mut(out), per_b,
                        x0.numel(), cur_stream());
  return out;
}

torch::Tensor sched_blend(torch::Tensor xt, torch::Tensor eps,
                          torch::Tensor a, torch::Tensor bt,
                          torch::Tensor c_out, torch::Tensor c_skip) {
  CHECK_IN(xt);
  CHECK_IN(eps);
  TORCH_CHECK(xt.dtype() == torch::kHalf && eps.dtype() == torch::kHalf);
  const long B = xt.size(0);
  TORCH_CHECK(a.numel() == B && bt.numel() == B && c_out.numel() == B &&
              c_skip.numel() == B, "coeffs must be (B,)");
  const long per_b = xt.numel() / B;
  TORCH_CHECK(per_b % 8 == 0, "row size must be a multiple of 8");
  auto out = torch::empty_like(xt);
  airtc_sched_blend(h_ptr(xt), h_ptr(eps), a.data_ptr<float>(),
                    bt.data_ptr<float>(), c_out.data_ptr<float>(),
                    c_skip.data_ptr<float>(), h_ptr_mut(out), per_b,
                    xt.numel(), cur_stream());
  return out;
}

// fp8 conv: MX-scaled MFMA path (conv2d_fp8.hip). Same surface as conv2d
// but weights are pre-quantized e4m3 bytes + per-OC dequant scales, and the
// activation scale rides along as a scalar.
torch::Tensor conv2d_fp8(torch::Tensor x, torch::Tensor w_fp8,
                         torch::Tensor dq, double a_scale,
                         c10::optional<torch::Tensor> bias,
                         c10::optional<torch::Tensor> cbias,
                         c10::optional<torch::Tensor> residual, int64_t R,
                         int64_t S, int64_t stride, int64_t pad, int64_t act,
                         c10::optional<torch::Tensor> in_affine,
                         int64_t in_act, double out_scale) {
  CHECK_IN(x);
  CHECK_IN(w_fp8);
  CHECK_IN(dq);
  const bool x_q8 = x.dtype() == torch::kUInt8;
  TORCH_CHECK(x_q8 || x.dtype() == torch::kHalf,
              "x must be f16 or u8 e4m3 codes");
  TORCH_CHECK(w_fp8.dtype() == torch::kUInt8, "w_fp8 must be e4m3 bytes");
  TORCH_CHECK(dq.dtype() == torch::kFloat, "dq must be f32[OC]");
  const int B = x.size(0), H = x.size(1), W = x.size(2), IC = x.size(3);
  const int OC = w_fp8.size(0);
  TORCH_CHECK(IC % 64 == 0, "fp8 conv path requires IC % 64 == 0");
  TORCH_CHECK(dq.numel() == OC, "dq must have OC entries");
  TORCH_CHECK(w_fp8.size(1) == (long)R * S * IC, "w_fp8 must be (OC, R*S*IC)");
  TORCH_CHECK(!(x_q8 && in_affine.has_value()),
              "pre-quantized input cannot take a fused input affine");
  const int HO = (H + 2 * (int)pad - (int)R) / (int)stride + 1;
  const int WO = (W + 2 * (int)pad - (int)S) / (int)stride + 1;
  int path0 = airtc_conv2d_splitk_for(B, HO, WO, OC, IC);
  if (path0 == 0 || path0 == 100) path0 = HO * WO >= 2048 ? 1 : -1;
  const int splitk0 = path0 > 0 ? path0 : -path0;
  // q8 output only on the split-K-free path (the finalize pass stays
  // fp8-agnostic); callers get f16 back otherwise and must dispatch on
  // the returned dtype
  const bool want_q8 = out_scale > 0 && splitk0 == 1;
  auto out = torch::empty({B, HO, WO, OC},
                          x.options().dtype(want_q8 ? torch::kUInt8
                                                    : torch::kHalf));
  const float* bp = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->dtype() == torch::kFloat && bias->is_contiguous());
    bp = bias->data_ptr<float>();
  }
  const uint16_t* cb = nullptr;
  if (cbias.has_value()) {
    CHECK_IN((*cbias));
    TORCH_CHECK(cbias->dtype() == torch::kHalf);
    cb = h_ptr(*cbias);
  }
  const uint16_t* res = nullptr;
  if (residual.has_value()) {
    CHECK_IN((*residual));
    TORCH_CHECK(residual->dtype() == torch::kHalf);
    TORCH_CHECK(residual->numel() == out.numel(), "residual shape mismatch");
    res = h_ptr(*residual);
  }
  const float* aff = nullptr;
  if (in_affine.has_value()) {
    CHECK_IN((*in_affine));
    TORCH_CHECK(in_affine->dtype() == torch::kFloat &&
                    in_affine->numel() == (long)B * IC * 2,
                "in_affine must be (B, IC, 2) f32");
    aff = in_affine->data_ptr<float>();
  }
  const int path = path0;
  const int splitk = splitk0;
  float* wsp = nullptr;
  torch::Tensor ws;
  if (splitk > 1) {
    ws = torch::empty({(long)B * splitk, (long)HO * WO, OC},
                      x.options().dtype(torch::kFloat));
    wsp = ws.data_ptr<float>();
  }
  const uint16_t* xp = x_q8
                           ? reinterpret_cast<const uint16_t*>(
                                 x.data_ptr<uint8_t>())
                           : h_ptr(x);
  uint16_t* op = want_q8 ? nullptr : h_ptr_mut(out);
  uint8_t* oq = want_q8 ? out.data_ptr<uint8_t>() : nullptr;
  airtc_conv2d_fp8_mfma(xp, w_fp8.data_ptr<uint8_t>(),
                        dq.data_ptr<float>(), bp, cb, res, op, wsp,
                        B, H, W, IC, HO, WO, OC, (int)R, (int)S, (int)stride,
                        (int)pad, (int)act, path, aff, (int)in_act,
                        (float)a_scale, x_q8 ? 1 : 0, oq,
                        want_q8 ? (float)(1.0 / out_scale) : 0.f,
                        cur_stream());
  return out;
}

// fp8 hardware probes (fp8.hip): raw MX MFMA tile + fused scale-converts
torch::Tensor fp8_mx_probe(torch::Tensor A, torch::Tensor B, int64_t sa,
                           int64_t sb) {
  CHECK_IN(A);
  CHECK_IN(B);
  TORCH_CHECK(A.dtype() == torch::kUInt8 && B.dtype() == torch::kUInt8);
  TORCH_CHECK(A.numel() == 2048 && B.numel() == 2048, "fragments are 2048B");
  auto draw = torch::empty({256}, A.options().dtype(torch::kFloat));
  airtc_fp8_mx_probe(A.data_ptr<uint8_t>(), B.data_ptr<uint8_t>(),
                     draw.data_ptr<float>(), (int)sa, (int)sb, cur_stream());
  return draw;
}

torch::Tensor fp8_quant_probe(torch::Tensor in16, double sa, int64_t variant) {
  CHECK_IN(in16);
  TORCH_CHECK(in16.dtype() == torch::kHalf && in16.numel() == 16);
  auto out = torch::empty({16}, in16.options().dtype(torch::kUInt8));
  airtc_fp8_quant_probe(h_ptr(in16), (float)sa, out.data_ptr<uint8_t>(),
                        (int)variant, cur_stream());
  return out;
}

pybind11::tuple fp8_cvt_probe(torch::Tensor fin, double scale,
                              torch::Tensor enc_in) {
  CHECK_IN(fin);
  CHECK_IN(enc_in);
  TORCH_CHECK(fin.dtype() == torch::kHalf && fin.numel() == 2);
  TORCH_CHECK(enc_in.dtype() == torch::kUInt8 && enc_in.numel() == 2);
  auto enc_out = torch::empty({2}, enc_in.options());
  auto dec_out = torch::empty({2}, fin.options());
  airtc_fp8_cvt_probe(h_ptr(fin), (float)scale, enc_out.data_ptr<uint8_t>(),
                      enc_in.data_ptr<uint8_t>(), h_ptr_mut(dec_out),
                      cur_stream());
  return pybind11::make_tuple(enc_out, dec_out);
}

}  // namespace

void airtc_register_dtls(pybind11::module_& m);  // dtls.cpp

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  airtc_register_dtls(m);
  m.def("conv2d", &conv2d, "implicit-GEMM MFMA conv2d (NHWC)",
        pybind11::arg("x"), pybind11::arg("w_perm"), pybind11::arg("bias"),
        pybind11::arg("cbias"), pybind11::arg("residual"), pybind11::arg("R"),
        pybind11::arg("S"), pybind11::arg("stride"), pybind11::arg("pad"),
        pybind11::arg("act"), pybind11::arg("in_affine") = pybind11::none(),
        pybind11::arg("in_act") = 0,
        pybind11::arg("counters") = pybind11::none());
  m.def("group_norm_silu", &group_norm_silu);
  m.def("group_norm_silu_fp8", &group_norm_silu_fp8,
        "GN+act writing e4m3 codes (producer-side fp8 quantization)");
  m.def("group_norm_coeffs", &group_norm_coeffs,
        "(B,C,2) f32 affine pairs for the fused GN->conv input transform");
  m.def("layer_norm", &layer_norm);
  m.def("attention_bhlc", &attention_bhlc);
  m.def("silu", &silu);
  m.def("geglu", &geglu);
  m.def("add_act", &add_act);
  m.def("upsample2x", &upsample2x);
  m.def("sched_add_noise", &sched_add_noise,
        "fused q(x_t|x0) sample: a*x0 + b*noise (per-batch-row coeffs)");
  m.def("sched_blend", &sched_blend,
        "fused LCM step: c_out*(x_t - b*eps)/a + c_skip*x_t");
  m.def("preprocess_u8", &preprocess_u8);
  m.def("postprocess_u8", &postprocess_u8);
  m.def("conv2d_fp8", &conv2d_fp8,
        "fp8 e4m3 implicit-GEMM conv2d on the MX-scaled MFMA (NHWC)",
        pybind11::arg("x"), pybind11::arg("w_fp8"), pybind11::arg("dq"),
        pybind11::arg("a_scale"), pybind11::arg("bias"),
        pybind11::arg("cbias"), pybind11::arg("residual"), pybind11::arg("R"),
        pybind11::arg("S"), pybind11::arg("stride"), pybind11::arg("pad"),
        pybind11::arg("act"), pybind11::arg("in_affine") = pybind11::none(),
        pybind11::arg("in_act") = 0, pybind11::arg("out_scale") = 0.0);
  m.def("fp8_mx_probe", &fp8_mx_probe,
        "raw-fragment v_mfma_scale_f32_16x16x128_f8f6f4 tile (layout probe)");
  m.def("fp8_quant_probe", &fp8_quant_probe,
        "conv kernel's exact 16-value activation encode (variant 0/1)");
  m.def("fp8_cvt_probe", &fp8_cvt_probe,
        "v_cvt_scalef32_pk_{fp8_f16,f16_fp8} semantics probe");
  m.def("vcn_probe", &vcn_probe, "probe the VCN VA-API stack");
  m.def("h264_sps_pps", &h264_sps_pps, "Annex-B SPS+PPS for (w, h)");
  m.def("h264sw_table_check", []() { return airtc_h264sw_table_check(); },
        "0 iff all CAVLC tables are prefix-free");
  pybind11::class_<H264SwEncoder>(m, "H264SwEncoder")
      .def(pybind11::init<int, int, int, int>(), pybind11::arg("w"),
           pybind11::arg("h"), pybind11::arg("slices") = 4,
           pybind11::arg("mb_mode") = 0)
      .def("encode", &H264SwEncoder::encode, pybind11::arg("rgb"),
           pybind11::arg("qp"), pybind11::arg("keyframe") = true,
           "RGB24 bytes + QP -> Annex-B (IDR, or P when keyframe=false)");
  m.def("h264_pred4",
        [](int mode, pybind11::bytes top, pybind11::bytes left, int tl,
           bool ht, bool hl, bool htl) {
          std::string t(top), l(left);
          TORCH_CHECK(t.size() == 8 && l.size() == 4, "t8/l4 sizes");
          uint8_t out[16];
          airtc_h264_pred4(mode, (const uint8_t*)t.data(),
                           (const uint8_t*)l.data(), (uint8_t)tl, ht, hl, htl,
                           out);
          return pybind11::bytes((const char*)out, 16);
        },
        "one 4x4 intra prediction (decoder path) for golden tests");
  pybind11::class_<H264SwDecoder>(m, "H264SwDecoder")
      .def(pybind11::init<>())
      .def("decode", &H264SwDecoder::decode,
           "Annex-B AU -> (rgb bytes, w, h) | None");
  pybind11::class_<VcnEncoder>(m, "VcnEncoder")
      .def(pybind11::init<int, int>())
      .def("encode", &VcnEncoder::encode,
           "RGB24 bytes + QP -> Annex-B IDR via the VCN hardware encoder");
}
