// Common device-side helpers for the MI355X (gfx950 / CDNA4) kernels.
//
// Conventions:
//  - wavefront = 64 lanes (CDNA), blocks are multiples of 64 threads
//  - f16 storage, f32 accumulation; MFMA shape 16x16x32 (f16 in, f32 out)
//  - activations NHWC; channel runs are the fast (contiguous) dimension
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64

typedef _Float16 f16;
typedef __attribute__((__vector_size__(8 * sizeof(_Float16)))) _Float16 f16x8;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4;
typedef __attribute__((__vector_size__(2 * sizeof(float)))) float f32x2;
typedef __attribute__((__vector_size__(4 * sizeof(uint32_t)))) uint32_t u32x4;

// D = A(16x32) @ B(32x16) + C  — per-wave MFMA, f16 inputs, f32 accum.
// A fragment: lane l holds A[l%16][(l/16)*8 + j], j=0..7  (one ds_read_b128)
// B fragment: lane l holds B[(l/16)*8 + j][l%16]
// C/D       : lane l holds D[(l/16)*4 + r][l%16], r=0..3
__device__ __forceinline__ f32x4 mfma16x16x32(f16x8 a, f16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ float sigmoidf_dev(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

__device__ __forceinline__ float siluf(float x) { return x * sigmoidf_dev(x); }

__device__ __forceinline__ float geluf(float x) {
  // erf-based GELU (matches torch default)
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752f));
}

// activation codes shared with the host side
enum ActCode { ACT_NONE = 0, ACT_SILU = 1, ACT_RELU = 2 };

__device__ __forceinline__ float apply_act(float v, int act) {
  if (act == ACT_SILU) return siluf(v);
  if (act == ACT_RELU) return v > 0.0f ? v : 0.0f;
  return v;
}

// butterfly reduce over all 64 lanes of a wave (defined below, after the
// DPP helpers: the first 4 steps run on the VALU pipe, only the two
// cross-row steps pay the ds_bpermute cost)
template <typename Op>
__device__ __forceinline__ float wave_reduce(float v, Op op);

// DPP cross-lane move (VALU-pipe, ~free) — __shfl_xor lowers to
// ds_bpermute (LDS pipe), which contends with staging/fragment traffic in
// LDS-heavy kernels. ctrl: 0x00-0xFF quad_perm, 0x140 row_mirror,
// 0x141 row_half_mirror (row = 16 lanes on CDNA).
template <int CTRL>
__device__ __forceinline__ float dpp_mov_f32(float v) {
  int i = __builtin_bit_cast(int, v);
  i = __builtin_amdgcn_update_dpp(0, i, CTRL, 0xF, 0xF, true);
  return __builtin_bit_cast(float, i);
}

// reduce over each 16-lane quarter-wave entirely on the VALU pipe:
// quad xor1, quad xor2, half-mirror (joins the quads of each 8-group),
// row-mirror (joins the two 8-groups of the 16-row)
template <typename Op>
__device__ __forceinline__ float quarter_reduce(float v, Op op) {
  v = op(v, dpp_mov_f32<0xB1>(v));   // quad_perm [1,0,3,2] = xor 1
  v = op(v, dpp_mov_f32<0x4E>(v));   // quad_perm [2,3,0,1] = xor 2
  v = op(v, dpp_mov_f32<0x141>(v));  // row_half_mirror
  v = op(v, dpp_mov_f32<0x140>(v));  // row_mirror
  return v;
}

template <typename Op>
__device__ __forceinline__ float wave_reduce(float v, Op op) {
  v = quarter_reduce(v, op);  // lanes within each 16-row: VALU-pipe DPP
  v = op(v, __shfl_xor(v, 16, 64));
  v = op(v, __shfl_xor(v, 32, 64));
  return v;
}

struct SumOp {
  __device__ float operator()(float a, float b) const { return a + b; }
};
struct MaxOp {
  __device__ float operator()(float a, float b) const { return fmaxf(a, b); }
};

__device__ __forceinline__ int ceil_div_dev(int a, int b) { return (a + b - 1) / b; }
static inline int ceil_div(int a, int b) { return (a + b - 1) / b; }
