// Elementwise kernels: frame pre/post-processing, activations, resampling.
//
// Replaces (MI355X-natively) reference N3/N4 — CV-CUDA convertto + reformat
// (reference lib/pipeline.py:61-63) — and the wrapper's tensor post-process
// (reference lib/pipeline.py:72-74). All kernels vectorize to 8-16 B/lane
// (guide G13: scalar f16 ~2-2.5x slower) and grid-stride over capped grids
// (guide G11).

#include "common.h"

#define EW_BLOCK 256
#define EW_MAX_BLOCKS 2048

// ---------------------------------------------------------------------------
// u8 RGB -> f16 in [-1, 1]    (n = total element count, multiple of 8)
// ---------------------------------------------------------------------------
__global__ void preprocess_u8_kernel(const uint8_t* __restrict__ in,
                                     f16* __restrict__ out, long n8) {
  const float inv = 1.0f / 127.5f;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * blockDim.x) {
    // 8 u8 in, 8 f16 out per lane
    uint2 raw = reinterpret_cast<const uint2*>(in)[i];
    f16x8 o;
    const uint8_t* b = reinterpret_cast<const uint8_t*>(&raw);
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (f16)((float)b[j] * inv - 1.0f);
    reinterpret_cast<f16x8*>(out)[i] = o;
  }
}

extern "C" void airtc_preprocess_u8(const uint8_t* in, uint16_t* out, long n,
                                    hipStream_t s) {
  long n8 = n / 8;
  int blocks = (int)min((long)EW_MAX_BLOCKS, (n8 + EW_BLOCK - 1) / EW_BLOCK);
  hipLaunchKernelGGL(preprocess_u8_kernel, dim3(blocks), dim3(EW_BLOCK), 0, s,
                     in, reinterpret_cast<f16*>(out), n8);
}

// ---------------------------------------------------------------------------
// f16 in [-1,1] -> u8 with round+clamp (reference lib/pipeline.py:74)
// ---------------------------------------------------------------------------
__global__ void postprocess_u8_kernel(const f16* __restrict__ in,
                                      uint8_t* __restrict__ out, long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * blockDim.x) {
    f16x8 v = reinterpret_cast<const f16x8*>(in)[i];
    uint2 packed;
    uint8_t* b = reinterpret_cast<uint8_t*>(&packed);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = ((float)v[j] + 1.0f) * 127.5f;
      b[j] = (uint8_t)min(255.0f, max(0.0f, nearbyintf(f)));
    }
    reinterpret_cast<uint2*>(out)[i] = packed;
  }
}

extern "C" void airtc_postprocess_u8(const uint16_t* in, uint8_t* out, long n,
                                     hipStream_t s) {
  long n8 = n / 8;
  int blocks = (int)min((long)EW_MAX_BLOCKS, (n8 + EW_BLOCK - 1) / EW_BLOCK);
  hipLaunchKernelGGL(postprocess_u8_kernel, dim3(blocks), dim3(EW_BLOCK), 0, s,
                     reinterpret_cast<const f16*>(in), out, n8);
}

// ---------------------------------------------------------------------------
// silu (f16, vec8)
// ---------------------------------------------------------------------------
__global__ void silu_kernel(const f16* __restrict__ in, f16* __restrict__ out,
                            long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * blockDim.x) {
    f16x8 v = reinterpret_cast<const f16x8*>(in)[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = (f16)siluf((float)v[j]);
    reinterpret_cast<f16x8*>(out)[i] = v;
  }
}

extern "C" void airtc_silu_f16(const uint16_t* in, uint16_t* out, long n,
                               hipStream_t s) {
  long n8 = n / 8;
  int blocks = (int)min((long)EW_MAX_BLOCKS, (n8 + EW_BLOCK - 1) / EW_BLOCK);
  hipLaunchKernelGGL(silu_kernel, dim3(blocks), dim3(EW_BLOCK), 0, s,
                     reinterpret_cast<const f16*>(in),
                     reinterpret_cast<f16*>(out), n8);
}

// ---------------------------------------------------------------------------
// GEGLU: x (rows, 2*inner) -> a * gelu(b), out (rows, inner)
// ---------------------------------------------------------------------------
__global__ void geglu_kernel(const f16* __restrict__ in, f16* __restrict__ out,
                             long rows, long inner8) {
  long total = rows * inner8;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    long r = i / inner8, c8 = i % inner8;
    const f16x8* row = reinterpret_cast<const f16x8*>(in + r * inner8 * 16);
    f16x8 a = row[c8];
    f16x8 b = row[c8 + inner8];
    f16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (f16)((float)a[j] * geluf((float)b[j]));
    reinterpret_cast<f16x8*>(out)[i] = o;
  }
}

extern "C" void airtc_geglu_f16(const uint16_t* in, uint16_t* out, long rows,
                                long inner, hipStream_t s) {
  long inner8 = inner / 8;
  long total = rows * inner8;
  int blocks = (int)min((long)EW_MAX_BLOCKS, (total + EW_BLOCK - 1) / EW_BLOCK);
  hipLaunchKernelGGL(geglu_kernel, dim3(blocks), dim3(EW_BLOCK), 0, s,
                     reinterpret_cast<const f16*>(in),
                     reinterpret_cast<f16*>(out), rows, inner8);
}

// ---------------------------------------------------------------------------
// fused add (+ optional activation): residual adds / TAESD add+relu
// ---------------------------------------------------------------------------
__global__ void add_act_kernel(const f16* __restrict__ a,
                               const f16* __restrict__ b, f16* __restrict__ out,
                               long n8, int act) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * blockDim.x) {
    f16x8 va = reinterpret_cast<const f16x8*>(a)[i];
    f16x8 vb = reinterpret_cast<const f16x8*>(b)[i];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      va[j] = (f16)apply_act((float)va[j] + (float)vb[j], act);
    reinterpret_cast<f16x8*>(out)[i] = va;
  }
}

extern "C" void airtc_add_act_f16(const uint16_t* a, const uint16_t* b,
                                  uint16_t* out, long n, int act,
                                  hipStream_t s) {
  long n8 = n / 8;
  int blocks = (int)min((long)EW_MAX_BLOCKS, (n8 + EW_BLOCK - 1) / EW_BLOCK);
  hipLaunchKernelGGL(add_act_kernel, dim3(blocks), dim3(EW_BLOCK), 0, s,
                     reinterpret_cast<const f16*>(a),
                     reinterpret_cast<const f16*>(b),
                     reinterpret_cast<f16*>(out), n8, act);
}

// ---------------------------------------------------------------------------
// nearest-2x upsample, NHWC (C % 8 == 0)
// ---------------------------------------------------------------------------
__global__ void upsample2x_kernel(const f16* __restrict__ in,
                                  f16* __restrict__ out, int B, int H, int W,
                                  int C8) {
  long total = (long)B * 2 * H * 2 * W * C8;
  int W2 = 2 * W;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * blockDim.x) {
    long c8 = i % C8;
    long rest = i / C8;
    int wo = rest % W2;
    rest /= W2;
    int ho = rest % (2 * H);
    long b = rest / (2 * H);
    long src = ((b * H + (ho >> 1)) * W + (wo >> 1)) * C8 + c8;
    reinterpret_cast<f16x8*>(out)[i] =
        reinterpret_cast<const f16x8*>(in)[src];
  }
}

extern "C" void airtc_upsample2x_f16(const uint16_t* in, uint16_t* out, int B,
                                     int H, int W, int C, hipStream_t s) {
  int C8 = C / 8;
  long total = (long)B * 2 * H * 2 * W * C8;
  int blocks = (int)min((long)EW_MAX_BLOCKS, (total + EW_BLOCK - 1) / EW_BLOCK);
  hipLaunchKernelGGL(upsample2x_kernel, dim3(blocks), dim3(EW_BLOCK), 0, s,
                     reinterpret_cast<const f16*>(in),
                     reinterpret_cast<f16*>(out), B, H, W, C8);
}

// ---------------------------------------------------------------------------
// Fused scheduler math (ROADMAP micro-fusion tier): the per-frame LCM
// scheduler steps were ~6 small aten broadcast kernels; each becomes ONE
// vectorized kernel with the per-batch-row coefficients read from f32
// arrays. per_b8 = elements-per-batch-row / 8.
//   add_noise:  out = a[b]*x0 + bt[b]*noise
//   blend:      out = c_out[b]*(x_t - bt[b]*eps)/a[b] + c_skip[b]*x_t
// ---------------------------------------------------------------------------
__global__ void sched_add_noise_kernel(const f16* __restrict__ x0,
                                       const f16* __restrict__ noise,
                                       const float* __restrict__ a,
                                       const float* __restrict__ bt,
                                       f16* __restrict__ out, long per_b8,
                                       long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * blockDim.x) {
    const long b = i / per_b8;
    const float av = a[b], bv = bt[b];
    f16x8 x = reinterpret_cast<const f16x8*>(x0)[i];
    f16x8 nz = reinterpret_cast<const f16x8*>(noise)[i];
    f16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (f16)(av * (float)x[j] + bv * (float)nz[j]);
    reinterpret_cast<f16x8*>(out)[i] = o;
  }
}

__global__ void sched_blend_kernel(const f16* __restrict__ xt,
                                   const f16* __restrict__ eps,
                                   const float* __restrict__ a,
                                   const float* __restrict__ bt,
                                   const float* __restrict__ c_out,
                                   const float* __restrict__ c_skip,
                                   f16* __restrict__ out, long per_b8,
                                   long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += gridDim.x * blockDim.x) {
    const long b = i / per_b8;
    const float inv_a = 1.0f / a[b], bv = bt[b];
    const float co = c_out[b], cs = c_skip[b];
    f16x8 x = reinterpret_cast<const f16x8*>(xt)[i];
    f16x8 e = reinterpret_cast<const f16x8*>(eps)[i];
    f16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float xf = (float)x[j];
      const float x0 = (xf - bv * (float)e[j]) * inv_a;
      o[j] = (f16)(co * x0 + cs * xf);
    }
    reinterpret_cast<f16x8*>(out)[i] = o;
  }
}

extern "C" void airtc_sched_add_noise(const uint16_t* x0,
                                      const uint16_t* noise, const float* a,
                                      const float* bt, uint16_t* out,
                                      long per_b, long n, hipStream_t s) {
  const long n8 = n / 8;
  int blocks = (int)min((long)EW_MAX_BLOCKS, (n8 + EW_BLOCK - 1) / EW_BLOCK);
  hipLaunchKernelGGL(sched_add_noise_kernel, dim3(blocks), dim3(EW_BLOCK), 0,
                     s, reinterpret_cast<const f16*>(x0),
                     reinterpret_cast<const f16*>(noise), a, bt,
                     reinterpret_cast<f16*>(out), per_b / 8, n8);
}

extern "C" void airtc_sched_blend(const uint16_t* xt, const uint16_t* eps,
                                  const float* a, const float* bt,
                                  const float* c_out, const float* c_skip,
                                  uint16_t* out, long per_b, long n,
                                  hipStream_t s) {
  const long n8 = n / 8;
  int blocks = (int)min((long)EW_MAX_BLOCKS, (n8 + EW_BLOCK - 1) / EW_BLOCK);
  hipLaunchKernelGGL(sched_blend_kernel, dim3(blocks), dim3(EW_BLOCK), 0, s,
                     reinterpret_cast<const f16*>(xt),
                     reinterpret_cast<const f16*>(eps), a, bt, c_out, c_skip,
                     reinterpret_cast<f16*>(out), per_b / 8, n8);
}
