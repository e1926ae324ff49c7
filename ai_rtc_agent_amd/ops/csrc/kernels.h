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

// norms ---------------------------------------------------------------------
void airtc_group_norm_silu(const uint16_t* x, const float* gamma,
                           const float* beta, uint16_t* out, int B, int HW,
                           int C, int G, float eps, int act, hipStream_t s);
void airtc_layer_norm(const uint16_t* x, const float* gamma, const float* beta,
                      uint16_t* out, long rows, int C, float eps,
                      hipStream_t s);

// conv ----------------------------------------------------------------------
// x_pad: (B, Hp, Wp, IC) NHWC f16 (already zero-padded when padding=1)
// w    : (OC, R*S*IC) f16, k order = (r, s, ic)
// out  : (B, HO, WO, OC) f16
void airtc_conv2d_mfma(const uint16_t* x_pad, const uint16_t* w,
                       const float* bias, uint16_t* out, int B, int Hp, int Wp,
                       int IC, int HO, int WO, int OC, int R, int S,
                       int stride, int act, hipStream_t s);
void airtc_conv2d_direct(const uint16_t* x_pad, const uint16_t* w,
                         const float* bias, uint16_t* out, int B, int Hp,
                         int Wp, int IC, int HO, int WO, int OC, int R, int S,
                         int stride, int act, hipStream_t s);

// attention -----------------------------------------------------------------
// q: base+strides address (B,H) heads; row stride in elements.
// all of q/k/v/out share the (b,h) base law: base = b*sb + h*sh
void airtc_attention(const uint16_t* q, const uint16_t* k, const uint16_t* v,
                     uint16_t* out, int B, int H, int Lq, int Lk, int d,
                     long q_sb, long q_sh, long q_row, long k_sb, long k_sh,
                     long k_row, long o_sb, long o_sh, long o_row, float scale,
                     hipStream_t s);

}  // extern "C"
