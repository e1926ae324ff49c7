// Fused normalisation kernels (NHWC).
//
// group_norm_silu: the UNet/ResNet GroupNorm(32)+SiLU pair fused into one
// kernel (reference runs these inside its TRT engine).
// One workgroup per (batch, group); two passes over the group's slice
// (sum/sumsq reduce, then normalise+affine+activation) with half2 loads.
//
// layer_norm: one wave per row (transformer blocks), f16x8 loads, f32 stats.

#include "common.h"

// ---------------------------------------------------------------------------
// GroupNorm(+act) over NHWC: group g covers channels [g*Cg, (g+1)*Cg)
// ---------------------------------------------------------------------------
typedef __attribute__((__vector_size__(2 * sizeof(_Float16)))) _Float16 f16x2;

__global__ void group_norm_silu_kernel(const f16* __restrict__ x,
                                       const float* __restrict__ gamma,
                                       const float* __restrict__ beta,
                                       f16* __restrict__ out, int HW, int C,
                                       int G, float eps, int act) {
  const int b = blockIdx.x / G;
  const int g = blockIdx.x % G;
  const int Cg = C / G;
  const int Cg2 = Cg / 2;
  const long base = (long)b * HW * C + (long)g * Cg;
  const long n2 = (long)HW * Cg2;  // half2 elements in this group slice

  float sum = 0.f, sumsq = 0.f;
  for (long i = threadIdx.x; i < n2; i += blockDim.x) {
    long p = i / Cg2, c2 = i % Cg2;
    f16x2 v = *reinterpret_cast<const f16x2*>(&x[base + p * C + c2 * 2]);
    float a = (float)v[0], c = (float)v[1];
    sum += a + c;
    sumsq += a * a + c * c;
  }
  __shared__ float red[2][16];  // up to 16 waves
  int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  sum = wave_reduce(sum, SumOp());
  sumsq = wave_reduce(sumsq, SumOp());
  if (lane == 0) { red[0][wid] = sum; red[1][wid] = sumsq; }
  __syncthreads();
  int nw = blockDim.x >> 6;
  if (wid == 0) {
    sum = lane < nw ? red[0][lane] : 0.f;
    sumsq = lane < nw ? red[1][lane] : 0.f;
    sum = wave_reduce(sum, SumOp());
    sumsq = wave_reduce(sumsq, SumOp());
    if (lane == 0) {
      float n = (float)HW * Cg;
      float mean = sum / n;
      float var = sumsq / n - mean * mean;
      red[0][0] = mean;
      red[1][0] = rsqrtf(var + eps);
    }
  }
  __syncthreads();
  const float mean = red[0][0], rstd = red[1][0];

  for (long i = threadIdx.x; i < n2; i += blockDim.x) {
    long p = i / Cg2, c2 = i % Cg2;
    long idx = base + p * C + c2 * 2;
    f16x2 v = *reinterpret_cast<const f16x2*>(&x[idx]);
    int ch = g * Cg + (int)c2 * 2;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      float t = ((float)v[j] - mean) * rstd * gamma[ch + j] + beta[ch + j];
      v[j] = (f16)apply_act(t, act);
    }
    *reinterpret_cast<f16x2*>(&out[idx]) = v;
  }
}

extern "C" void airtc_group_norm_silu(const uint16_t* x, const float* gamma,
                                      const float* beta, uint16_t* out, int B,
                                      int HW, int C, int G, float eps, int act,
                                      hipStream_t s) {
  hipLaunchKernelGGL(group_norm_silu_kernel, dim3(B * G), dim3(256), 0, s,
                     reinterpret_cast<const f16*>(x), gamma, beta,
                     reinterpret_cast<f16*>(out), HW, C, G, eps, act);
}

// ---------------------------------------------------------------------------
// LayerNorm: rows x C, one wave per row (C % 8 == 0)
// ---------------------------------------------------------------------------
__global__ void layer_norm_kernel(const f16* __restrict__ x,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ beta,
                                  f16* __restrict__ out, long rows, int C8,
                                  float eps) {
  const int lane = threadIdx.x & 63;
  const long row = (long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= rows) return;
  const f16x8* xr = reinterpret_cast<const f16x8*>(x) + row * C8;

  float sum = 0.f, sumsq = 0.f;
  for (int i = lane; i < C8; i += WAVE) {
    f16x8 v = xr[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)v[j];
      sum += f;
      sumsq += f * f;
    }
  }
  sum = wave_reduce(sum, SumOp());
  sumsq = wave_reduce(sumsq, SumOp());
  float n = (float)C8 * 8.0f;
  float mean = sum / n;
  float rstd = rsqrtf(sumsq / n - mean * mean + eps);

  f16x8* orow = reinterpret_cast<f16x8*>(out) + row * C8;
  for (int i = lane; i < C8; i += WAVE) {
    f16x8 v = xr[i];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v[j] = (f16)(((float)v[j] - mean) * rstd * gamma[i * 8 + j] + beta[i * 8 + j]);
    orow[i] = v;
  }
}

extern "C" void airtc_layer_norm(const uint16_t* x, const float* gamma,
                                 const float* beta, uint16_t* out, long rows,
                                 int C, float eps, hipStream_t s) {
  int waves_per_block = 4;
  long blocks = (rows + waves_per_block - 1) / waves_per_block;
  hipLaunchKernelGGL(layer_norm_kernel, dim3((uint32_t)blocks),
                     dim3(waves_per_block * WAVE), 0, s,
                     reinterpret_cast<const f16*>(x), gamma, beta,
                     reinterpret_cast<f16*>(out), rows, C / 8, eps);
}
