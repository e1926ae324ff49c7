// Host-visible launcher declarations for the MI355X kernels.
// Interfaces use uint16_t* for f16 storage so the host translation unit
// (compiled by g++) never needs _Float16.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

extern "C" {

// elementwise ---------------------------------------------------------------
void airtc_preprocess_u8(const uint8_t* in, uint16_t* out, long n, hipStream_t s);
void airtc_postprocess_u8(const uint16_t* in, uint8_t* out, long n, hipStream_t s);
void airtc_silu_f16(const uint16_t* in, uint16_t* out, long n, hipStream_t s);
void airtc_geglu_f16(const uint16_t* in, uint16_t* out, long rows, long inner,
                     hipStream_t s);
void airtc_add_act_f16(const uint16_t* a, const uint16_t* b, uint16_t* out,
                       long n, int act, hipStream_t s);
void airtc_upsample2x_f16(const uint16_t* in, uint16_t* out, int B, int H,
                          int W, int C, hipStream_t s);
// fused LCM scheduler math (one kernel each; coefficients are per-batch-row
// f32 arrays of B entries, per_b = elements per batch row)
void airtc_sched_add_noise(const uint16_t* x0, const uint16_t* noise,
                           const float* a, const float* bt, uint16_t* out,
                           long per_b, long n, hipStream_t s);
void airtc_sched_blend(const uint16_t* xt, const uint16_t* eps,
                       const float* a, const float* bt, const float* c_out,
                       const float* c_skip, uint16_t* out, long per_b, long n,
                       hipStream_t s);

// norms ---------------------------------------------------------------------
int airtc_group_norm_nchunk(int B, int G);
void airtc_group_norm_silu(const uint16_t* x, const float* gamma,
                           const float* beta, uint16_t* out, float* ws, int B,
                           int HW, int C, int G, float eps, int act,
                           hipStream_t s);
void airtc_layer_norm(const uint16_t* x, const float* gamma, const float* beta,
                      uint16_t* out, long rows, int C, float eps,
                      hipStream_t s);
// fp8-output GN: apply pass writes e4m3 codes q = clamp(act(gn(x))/a_scale)
// (producer-side quantization for the fp8 conv path)
void airtc_group_norm_silu_fp8(const uint16_t* x, const float* gamma,
                               const float* beta, uint8_t* out, float* ws,
                               int B, int HW, int C, int G, float eps,
                               int act, float a_scale, hipStream_t s);
// (B, C, 2) f32 affine pairs for the fused GN->conv input transform
void airtc_group_norm_coeffs(const uint16_t* x, const float* gamma,
                             const float* beta, float* coeffs, float* ws,
                             int B, int HW, int C, int G, float eps,
                             hipStream_t s);

// conv ----------------------------------------------------------------------
// x   : (B, H, W, IC) NHWC f16 (zero-padding handled inline)
// w   : (OC, R*S*IC) f16, k order = (r, s, ic)
// cbias: optional per-(batch, out-channel) f16 bias (B, OC) — time-emb add
// residual: optional f16 tensor with out's shape, added pre-activation
// out : (B, HO, WO, OC) f16
// path: from airtc_conv2d_splitk_for (0=direct, 1=BM128, -k=BM64 splitk k);
// ws  : f32 workspace (B*k, HO*WO, OC) required when path < -1
int airtc_conv2d_splitk_for(int B, int HO, int WO, int OC, int IC);
// in_aff: optional (B, IC, 2) f32 per-channel input affine (fused GN) with
// in_act applied after — transforms x AT LOAD TIME (padding stays zero)
void airtc_conv2d_mfma(const uint16_t* x, const uint16_t* w, const float* bias,
                       const uint16_t* cbias, const uint16_t* residual,
                       uint16_t* out, float* ws, int B, int H, int W, int IC,
                       int HO, int WO, int OC, int R, int S, int stride,
                       int pad, int act, int path, const float* in_aff,
                       int in_act, int* counters, hipStream_t s);
void airtc_conv2d_direct(const uint16_t* x, const uint16_t* w,
                         const float* bias, const uint16_t* cbias,
                         const uint16_t* residual, uint16_t* out, int B, int H,
                         int W, int IC, int HO, int WO, int OC, int R, int S,
                         int stride, int pad, int act, const float* in_aff,
                         int in_act, hipStream_t s);

// fp8 (MX-scaled MFMA path) -------------------------------------------------
// conv2d on v_mfma_scale_f32_16x16x128_f8f6f4 (conv2d_fp8.hip). w_fp8 is
// [OC][K] OCP e4m3 bytes (per-OC host quantization); dq[oc] = a_scale *
// w_scale[oc] dequantizes in the epilogue; activations quantize in the
// staging loads (a_scale). Requires IC % 64 == 0; path from
// airtc_conv2d_splitk_for (nonzero).
// x_is_q8: x already holds e4m3 codes (u8, producer-quantized by e.g.
// airtc_group_norm_silu_fp8) — staging is a raw byte copy, in_aff unused.
// out_q8/out_inv: when non-null and split-K == 1, the epilogue writes e4m3
// codes q = e4m3(clamp(v / out_scale)) instead of f16 (chained fp8 layers).
void airtc_conv2d_fp8_mfma(const uint16_t* x, const uint8_t* w_fp8,
                           const float* dq, const float* bias,
                           const uint16_t* cbias, const uint16_t* residual,
                           uint16_t* out, float* ws, int B, int H, int W,
                           int IC, int HO, int WO, int OC, int R, int S,
                           int stride, int pad, int act, int path,
                           const float* in_aff, int in_act, float a_scale,
                           int x_is_q8, uint8_t* out_q8, float out_inv,
                           hipStream_t s);
// hardware probes: raw-fragment MX MFMA tile and the fused scale-converts
// (layout/semantics verified on hardware before the fp8 conv relies on them)
void airtc_fp8_mx_probe(const uint8_t* A, const uint8_t* B, float* draw,
                        int sa, int sb, hipStream_t s);
void airtc_fp8_cvt_probe(const uint16_t* fin, float scale, uint8_t* enc_out,
                         const uint8_t* enc_in, uint16_t* dec_out,
                         hipStream_t s);
void airtc_fp8_quant_probe(const uint16_t* in16, float sa, uint8_t* out16,
                           int variant, hipStream_t s);

// attention -----------------------------------------------------------------
// q: base+strides address (B,H) heads; row stride in elements.
// all of q/k/v/out share the (b,h) base law: base = b*sb + h*sh
void airtc_attention(const uint16_t* q, const uint16_t* k, const uint16_t* v,
                     uint16_t* out, int B, int H, int Lq, int Lk, int d,
                     long q_sb, long q_sh, long q_row, long k_sb, long k_sh,
                     long k_row, long o_sb, long o_sh, long o_row, float scale,
                     hipStream_t s);

}  // extern "C"

// vcn ------------------------------------------------------------------------
// Probe the VCN video block's VA-API userspace (runtime dlopen). Returns
// -1 = unavailable, else bit0 = H.264 decode, bit1 = H.264 encode; buf gets
// a stage/detail summary. Implemented in vcn.cpp.
extern "C" int airtc_vcn_probe(char* buf, int buflen);

// H.264 parameter-set generation (vcn.cpp): Annex-B SPS+PPS for (w, h).
extern "C" int airtc_h264_sps_pps(int width, int height, uint8_t* buf,
                                  int buflen);
