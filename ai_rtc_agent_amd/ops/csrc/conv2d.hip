// Conv2d NHWC — implicit-GEMM on MFMA (CDNA4 16x16x32 f16 tiles).
//
// This is the MI355X-native replacement for the reference's TensorRT
// UNet/VAE conv engines (reference
// lib/wrapper.py:409-512). Not a port: designed for gfx950 per
// /opt/skills/guides/cdna_hip_programming.md —
//  - GEMM view M=(B·HO·WO) pixels, N=OC, K=R·S·IC with the gather
//    on-the-fly (im2col never materialised)
//  - inputs are zero-PADDED NHWC so the K-gather has no bounds checks;
//    channel runs are 16B-contiguous (IC%32==0 on this path), so each lane
//    stages 8 f16 per load (guide G13)
//  - LDS tiles padded by one 16B access width (guide G4) for
//    conflict-reduced ds_read_b128 fragment reads
//  - T14 async-stage split: next K-tile's global loads issue before the
//    current tile's MFMAs so HBM latency hides under compute
//  - bias + activation fused in the epilogue
//
// Tile: BM=128 pixels x BN=64 out-channels x BK=32, 4 waves (2x2), each wave
// a 64x32 sub-tile = 4x2 fragments of 16x16, 8 MFMA per K-step.

#include "common.h"

#define BM 128
#define BN 64
#define BK 32
#define APITCH (BK + 8)  // f16 elements per LDS row (+16B pad)

__global__ __launch_bounds__(256) void conv2d_mfma_kernel(
    const f16* __restrict__ x, const f16* __restrict__ w,
    const float* __restrict__ bias, f16* __restrict__ out, int Hp, int Wp,
    int IC, int HO, int WO, int OC, int R, int S, int stride, int act,
    int K) {
  __shared__ f16 ldsA[BM * APITCH];
  __shared__ f16 ldsB[BN * APITCH];

  const int M = HO * WO;
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const long b = blockIdx.z;
  const f16* xb = x + b * (long)Hp * Wp * IC;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid & 1;   // wave row (2 waves over M)
  const int wn = wid >> 1;  // wave col (2 waves over N)

  // --- staging assignments (per K-step) ---
  // A: 512 x 16B loads; thread t does flats {t, t+256}
  int a_row[2], a_hi[2], a_wi[2], a_k8[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int flat = tid + i * 256;
    a_row[i] = flat >> 2;
    a_k8[i] = (flat & 3) * 8;
    int m = m0 + a_row[i];
    if (m >= M) m = M - 1;  // clamp: duplicate loads, stores masked
    a_hi[i] = (m / WO) * stride;
    a_wi[i] = (m % WO) * stride;
  }
  // B: 256 x 16B loads
  const int b_row = tid >> 2;
  const int b_k8 = (tid & 3) * 8;
  const int b_oc = min(n0 + b_row, OC - 1);
  const f16* wrow = w + (long)b_oc * K + b_k8;

  f32x4 acc[4][2];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK;

  // prologue: load K-tile 0 into registers
  f16x8 regA[2], regB;
  {
    const int rs = 0, ic0 = 0, r = 0, sc = 0;
    (void)rs; (void)ic0; (void)r; (void)sc;
#pragma unroll
    for (int i = 0; i < 2; ++i)
      regA[i] = *reinterpret_cast<const f16x8*>(
          &xb[((long)a_hi[i] * Wp + a_wi[i]) * IC + a_k8[i]]);
    regB = *reinterpret_cast<const f16x8*>(wrow);
  }

  for (int kt = 0; kt < nk; ++kt) {
    __syncthreads();  // previous tile's fragment reads done
    // write staged registers to LDS
#pragma unroll
    for (int i = 0; i < 2; ++i)
      *reinterpret_cast<f16x8*>(&ldsA[a_row[i] * APITCH + a_k8[i]]) = regA[i];
    *reinterpret_cast<f16x8*>(&ldsB[b_row * APITCH + b_k8]) = regB;
    __syncthreads();

    // T14: issue NEXT tile's global loads before this tile's MFMAs
    if (kt + 1 < nk) {
      const int k0 = (kt + 1) * BK;
      const int rs = k0 / IC;  // BK | IC, so one (r,s) per K-step
      const int ic0 = k0 - rs * IC;
      const int r = rs / S;
      const int sc = rs - r * S;
#pragma unroll
      for (int i = 0; i < 2; ++i)
        regA[i] = *reinterpret_cast<const f16x8*>(
            &xb[((long)(a_hi[i] + r) * Wp + (a_wi[i] + sc)) * IC + ic0 +
                a_k8[i]]);
      regB = *reinterpret_cast<const f16x8*>(wrow + k0);
    }

    // fragments + MFMA
    const int arow_base = wm * 64 + (lane & 15);
    const int fcol = (lane >> 4) * 8;
    f16x8 bfrag[2];
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
      bfrag[ni] = *reinterpret_cast<const f16x8*>(
          &ldsB[(wn * 32 + ni * 16 + (lane & 15)) * APITCH + fcol]);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      f16x8 afrag = *reinterpret_cast<const f16x8*>(
          &ldsA[(arow_base + mi * 16) * APITCH + fcol]);
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        acc[mi][ni] = mfma16x16x32(afrag, bfrag[ni], acc[mi][ni]);
    }
  }

  // epilogue: bias + activation + masked f16 stores
  f16* ob = out + b * (long)M * OC;
#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int col = n0 + wn * 32 + ni * 16 + (lane & 15);
    if (col >= OC) continue;
    const float bv = bias ? bias[col] : 0.0f;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int m = m0 + wm * 64 + mi * 16 + (lane >> 4) * 4 + j;
        if (m < M)
          ob[(long)m * OC + col] = (f16)apply_act(acc[mi][ni][j] + bv, act);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Direct conv for small/ragged IC (conv_in with IC=3/4, final RGB convs):
// K is tiny (<= 9*31), one thread per output element, f32 accumulate.
// ---------------------------------------------------------------------------
__global__ void conv2d_direct_kernel(const f16* __restrict__ x,
                                     const f16* __restrict__ w,
                                     const float* __restrict__ bias,
                                     f16* __restrict__ out, int Hp, int Wp,
                                     int IC, int HO, int WO, int OC, int R,
                                     int S, int stride, int act, int K,
                                     long total) {
  const int M = HO * WO;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * (long)blockDim.x) {
    const int oc = i % OC;
    long rest = i / OC;
    const int m = rest % M;
    const long b = rest / M;
    const int ho = m / WO, wo = m % WO;
    const f16* xb = x + b * (long)Hp * Wp * IC;
    const f16* wk = w + (long)oc * K;
    float a = bias ? bias[oc] : 0.0f;
    for (int r = 0; r < R; ++r)
      for (int s = 0; s < S; ++s) {
        const f16* xr =
            &xb[((long)(ho * stride + r) * Wp + (wo * stride + s)) * IC];
        const f16* wr = &wk[(r * S + s) * IC];
        for (int c = 0; c < IC; ++c) a += (float)xr[c] * (float)wr[c];
      }
    out[i] = (f16)apply_act(a, act);
  }
}

extern "C" void airtc_conv2d_mfma(const uint16_t* x_pad, const uint16_t* w,
                                  const float* bias, uint16_t* out, int B,
                                  int Hp, int Wp, int IC, int HO, int WO,
                                  int OC, int R, int S, int stride, int act,
                                  hipStream_t s) {
  const int K = R * S * IC;
  dim3 grid(ceil_div(HO * WO, BM), ceil_div(OC, BN), B);
  hipLaunchKernelGGL(conv2d_mfma_kernel, grid, dim3(256), 0, s,
                     reinterpret_cast<const f16*>(x_pad),
                     reinterpret_cast<const f16*>(w), bias,
                     reinterpret_cast<f16*>(out), Hp, Wp, IC, HO, WO, OC, R, S,
                     stride, act, K);
}

extern "C" void airtc_conv2d_direct(const uint16_t* x_pad, const uint16_t* w,
                                    const float* bias, uint16_t* out, int B,
                                    int Hp, int Wp, int IC, int HO, int WO,
                                    int OC, int R, int S, int stride, int act,
                                    hipStream_t s) {
  const int K = R * S * IC;
  long total = (long)B * HO * WO * OC;
  int blocks = (int)min((long)4096, (total + 255) / 256);
  hipLaunchKernelGGL(conv2d_direct_kernel, dim3(blocks), dim3(256), 0, s,
                     reinterpret_cast<const f16*>(x_pad),
                     reinterpret_cast<const f16*>(w), bias,
                     reinterpret_cast<f16*>(out), Hp, Wp, IC, HO, WO, OC, R, S,
                     stride, act, K, total);
}
