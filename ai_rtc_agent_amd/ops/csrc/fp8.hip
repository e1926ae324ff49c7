// FP8 (OCP e4m3) foundations — MX-scaled MFMA path for gfx950.
//
// CDNA4's only 2x-rate low-precision matmul is the block-scaled
// v_mfma_scale_f32_16x16x128_f8f6f4 (per-32-element e8m0 block scales,
// HW-fused dequant; the non-scaled fp8 MFMAs run at the bf16 rate).
// This file holds:
//   - layout/semantics PROBE kernels (fp8_mx_probe / fp8_cvt_probe): the
//     lane->element maps and the cvt_scalef32 scale direction are verified
//     ON HARDWARE by tests/tools before the conv kernel relies on them
//     (guide: "Always A=I-check with ASYMMETRIC B").
//   - the fp8 quantization helpers shared with conv2d_fp8.hip.
//
// Replaces nothing in the reference (it has no fp8 path): this is the
// MI355X-native reduced-precision serving tier (SURVEY.md §6 perf goals).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(2))) short s16x2;
typedef __attribute__((ext_vector_type(2))) _Float16 hf16x2;

// ---------------------------------------------------------------------------
// Probe 1: one MX MFMA tile, raw fragments in / raw accumulators out.
// Lane l contributes bytes A[l*32 .. l*32+32) and B[l*32 .. +32); the raw
// accumulator dump draw[l*4+j] lets the host infer every mapping without
// assuming any of them. sa/sb are the (uniform) e8m0 scale bytes.
// ---------------------------------------------------------------------------
__global__ void fp8_mx_probe_kernel(const uint8_t* __restrict__ A,
                                    const uint8_t* __restrict__ B,
                                    float* __restrict__ draw, int sa, int sb) {
  const int lane = threadIdx.x & 63;
  i32x8 a = *reinterpret_cast<const i32x8*>(A + lane * 32);
  i32x8 b = *reinterpret_cast<const i32x8*>(B + lane * 32);
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  // cbsz=0, blgp=0 -> both operands fp8 e4m3; opsel 0 -> scale byte 0
  c = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(a, b, c, 0, 0, 0, sa, 0,
                                                       sb);
#pragma unroll
  for (int j = 0; j < 4; ++j) draw[lane * 4 + j] = c[j];
}

// ---------------------------------------------------------------------------
// Probe 2: the gfx950 fused scale-converts.
//   enc: 2 f16 -> 2 fp8 bytes (one v_cvt_scalef32_pk_fp8_f16)
//   dec: 2 fp8 bytes -> 2 f16 (one v_cvt_scalef32_pk_f16_fp8)
// The host test establishes whether the scale multiplies or divides on each
// direction (the ISA doc is not in-image; measured truth goes in the test).
// ---------------------------------------------------------------------------
__global__ void fp8_cvt_probe_kernel(const _Float16* __restrict__ fin,
                                     float scale, uint8_t* __restrict__ enc_out,
                                     const uint8_t* __restrict__ enc_in,
                                     _Float16* __restrict__ dec_out) {
  if (threadIdx.x != 0) return;
  hf16x2 v;
  v[0] = fin[0];
  v[1] = fin[1];
  s16x2 packed = {0, 0};
  packed = __builtin_amdgcn_cvt_scalef32_pk_fp8_f16(packed, v, scale, false);
  enc_out[0] = (uint8_t)(packed[0] & 0xFF);
  enc_out[1] = (uint8_t)((packed[0] >> 8) & 0xFF);
  const int src = (int)enc_in[0] | ((int)enc_in[1] << 8);
  hf16x2 dec = __builtin_amdgcn_cvt_scalef32_pk_f16_fp8(src, scale, false);
  dec_out[0] = dec[0];
  dec_out[1] = dec[1];
}

// ---------------------------------------------------------------------------
// Probe 3: the conv kernel's exact activation-encode path (clamp to
// +-448*sa, two pk_fp8 calls packing 4 bytes per i32) on 16 f16 values —
// variant 0 = cvt_scalef32_pk_fp8_f16 (divide-by-sa in the instruction),
// variant 1 = v_mul by 1/sa then non-scaled cvt_pk_fp8_f32. Isolates the
// encode from the rest of the conv when numerics disagree.
// ---------------------------------------------------------------------------
__global__ void fp8_quant_probe_kernel(const _Float16* __restrict__ in16,
                                       float sa, uint8_t* __restrict__ out16,
                                       int variant) {
  if (threadIdx.x != 0) return;
  const float clampv = 448.0f * sa;
  const f16 lim = (f16)clampv;
  const float inv = 1.0f / sa;
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    s16x2 packed = {0, 0};
    const int j = q * 4;
    if (variant == 0) {
      hf16x2 limv = {lim, lim}, nlimv = {(f16)(-lim), (f16)(-lim)};
      hf16x2 v0 = {in16[j], in16[j + 1]};
      hf16x2 v1 = {in16[j + 2], in16[j + 3]};
      v0 = __builtin_elementwise_min(__builtin_elementwise_max(v0, nlimv), limv);
      v1 = __builtin_elementwise_min(__builtin_elementwise_max(v1, nlimv), limv);
      packed = __builtin_amdgcn_cvt_scalef32_pk_fp8_f16(packed, v0, sa, false);
      packed = __builtin_amdgcn_cvt_scalef32_pk_fp8_f16(packed, v1, sa, true);
    } else {
      float a[4];
#pragma unroll
      for (int h = 0; h < 4; ++h)
        a[h] = fminf(fmaxf((float)in16[j + h], -clampv), clampv) * inv;
      int p2 = 0;
      p2 = __builtin_amdgcn_cvt_pk_fp8_f32(a[0], a[1], p2, false);
      p2 = __builtin_amdgcn_cvt_pk_fp8_f32(a[2], a[3], p2, true);
      packed[0] = (short)(p2 & 0xFFFF);
      packed[1] = (short)((p2 >> 16) & 0xFFFF);
    }
    out16[q * 4 + 0] = (uint8_t)(packed[0] & 0xFF);
    out16[q * 4 + 1] = (uint8_t)((packed[0] >> 8) & 0xFF);
    out16[q * 4 + 2] = (uint8_t)(packed[1] & 0xFF);
    out16[q * 4 + 3] = (uint8_t)((packed[1] >> 8) & 0xFF);
  }
}

extern "C" void airtc_fp8_quant_probe(const uint16_t* in16, float sa,
                                      uint8_t* out16, int variant,
                                      hipStream_t s) {
  hipLaunchKernelGGL(fp8_quant_probe_kernel, dim3(1), dim3(64), 0, s,
                     reinterpret_cast<const _Float16*>(in16), sa, out16,
                     variant);
}

extern "C" void airtc_fp8_mx_probe(const uint8_t* A, const uint8_t* B,
                                   float* draw, int sa, int sb,
                                   hipStream_t s) {
  hipLaunchKernelGGL(fp8_mx_probe_kernel, dim3(1), dim3(64), 0, s, A, B, draw,
                     sa, sb);
}

extern "C" void airtc_fp8_cvt_probe(const uint16_t* fin, float scale,
                                    uint8_t* enc_out, const uint8_t* enc_in,
                                    uint16_t* dec_out, hipStream_t s) {
  hipLaunchKernelGGL(fp8_cvt_probe_kernel, dim3(1), dim3(64), 0, s,
                     reinterpret_cast<const _Float16*>(fin), scale, enc_out,
                     enc_in, reinterpret_cast<_Float16*>(dec_out));
}
