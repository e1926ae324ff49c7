// Conv2d NHWC — implicit-GEMM on MFMA (CDNA4 16x16x32 f16 tiles).
//
// MI355X-native replacement for the reference's TensorRT UNet/VAE conv
// engines (reference lib/wrapper.py:409-512).
// Designed for gfx950 per /opt/skills/guides/cdna_hip_programming.md:
//
//  - GEMM view M=(HO·WO) pixels x N=OC x K=R·S·IC; the im2col gather runs
//    on the fly with INLINE zero-padding (per-load bounds predicate — no
//    padded input copy, no aten F.pad fill+copy kernels)
//  - channel runs are 16B-contiguous (IC%32==0 on the MFMA path): each lane
//    stages 8 f16 per load (guide G13)
//  - LDS tiles padded by one 16B access width (guide G4) for ds_read_b128
//  - T14 async-stage split: next K-tile's global loads issue before the
//    current tile's MFMAs
//  - EPILOGUE FUSION: bias + per-(batch,channel) bias (time-embedding add)
//    + residual add + activation, all in the conv store
//  - two tile geometries + SPLIT-K: the UNet's small-spatial wide-channel
//    layers (8²x1280: M=64, K=11520) would otherwise launch 20 workgroups
//    on a 256-CU chip (measured 235us each, 56% of frame time); split-K
//    over the K loop with f32 slab partials + a finalize pass fills the
//    chip (profiles/ has the before/after)
//
// Geometry A (large M): BM=128 BN=64, 4 waves (2x2), wave=64x32, 8 MFMA/step
// Geometry B (small M): BM=64  BN=64, 4 waves (2x2), wave=32x32, 4 MFMA/step,
//                       optional split-K over blockIdx.z

#include <stdlib.h>

#include "common.h"

#define BN 64

struct KPos {
  int r, s, ic0;
};

__device__ __forceinline__ KPos kpos_at(int k0, int IC, int S) {
  const int rs = k0 / IC;
  return {rs / S, rs - (rs / S) * S, k0 - rs * IC};
}

// one staged 16B A-load with inline zero-padding. aff (optional,
// per-channel float pairs pre-offset for the batch) applies the fused
// GroupNorm affine + activation AT LOAD TIME — the producing GN-apply
// kernel and a full activation HBM round-trip disappear. Padding lanes
// stay zero (ok-gated): the transform models act(gn(x)) of the unfused
// pipeline, which the conv then zero-pads.
__device__ __forceinline__ f16x8 load_a(const f16* xb, int ho_s, int wo_s,
                                        int r, int s, int pad, int H, int W,
                                        int IC, int ic, const float* aff,
                                        int in_act) {
  const int hi = ho_s + r - pad;
  const int wi = wo_s + s - pad;
  const bool ok = (unsigned)hi < (unsigned)H && (unsigned)wi < (unsigned)W;
  const int hc = ok ? hi : 0, wc = ok ? wi : 0;
  f16x8 v = *reinterpret_cast<const f16x8*>(&xb[((long)hc * W + wc) * IC + ic]);
  if (!ok) {
    v = f16x8{0, 0, 0, 0, 0, 0, 0, 0};
  } else if (aff) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)v[j] * aff[(ic + j) * 2] + aff[(ic + j) * 2 + 1];
      v[j] = (f16)apply_act(f, in_act);
    }
  }
  return v;
}

// fused epilogue value: acc + bias + cbias, + residual, then act
__device__ __forceinline__ f16 epilogue(float acc, const float* bias,
                                        const f16* cbias, long cb_off,
                                        const f16* residual, long idx, int oc,
                                        int act) {
  float v = acc;
  if (bias) v += bias[oc];
  if (cbias) v += (float)cbias[cb_off + oc];
  if (residual) v += (float)residual[idx];
  return (f16)apply_act(v, act);
}

// ---------------------------------------------------------------------------
// GLDS variant: K-tiles stream HBM -> LDS via buffer_load...lds (async DMA,
// no VGPR round-trip, no ds_write pass — guide §5 "glds, 2 LDS buffers,
// BK=64, vmcnt(0) + plain __syncthreads()"). Zero-padding comes FREE from
// the buffer descriptor's bounds check (OOB voffset -> 0 written to LDS).
// The LDS image is lane-linear (glds writes base + lane*16), so the
// bank-conflict fix moves to the SOURCE address (guide rule 21): 16B unit
// u of row r holds global unit u ^ ((r>>1)&7), and fragment reads apply
// the same XOR. Schedule: 3-deep buffer ring with COUNTED vmcnt + raw
// barriers (one tile stays in flight across every barrier). Measured: ties
// the register-staged default within noise at SD shapes (ladder) — kept
// env-gated (AIRTC_CONV_GLDS=1) as the starting point for deeper pipelines.
// ---------------------------------------------------------------------------
typedef __attribute__((address_space(3))) f16 lds_f16;

__device__ __forceinline__ unsigned row_swz(int row) {
  return (unsigned)((row >> 1) & 7);
}

// XCD-aware tile mapping (guide T1, bijective form): the dispatcher places
// workgroup b on XCD b%8 with a PRIVATE L2 per XCD. The flat (mtile,ntile)
// grid is remapped so each XCD walks a CONTIGUOUS id range, with the
// OC-tile index fastest: the n-tiles sharing one activation panel run on
// one XCD back-to-back, so the panel (e.g. 737 KB per 128-pixel m-tile)
// stays L2-resident across its OC re-reads instead of re-fetching from
// LLC/HBM (PMC showed the conv at 4-5% MfmaUtil, memory-stalled).
__device__ __forceinline__ void xcd_tile_map(int wg, int nwg, int n_tiles,
                                             int* mt, int* nt) {
  const int x = wg & 7, pos = wg >> 3;
  const int q = nwg >> 3, r = nwg & 7;
  const int wgid = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + pos;
  *nt = wgid % n_tiles;
  *mt = wgid / n_tiles;
}

// ---------------------------------------------------------------------------
// templated MFMA conv: MFRAG = M-fragments per wave (4 -> BM=128, 2 -> BM=64)
// BK = K-tile depth: 64 when IC%64==0 (all SD/TAESD layers — halves the
// barrier count per K element, 2 MFMA K-chunks per stage), else 32.
// SPLITK > 1: partials go to ws (f32), finalize pass reduces.
// DBUF: double-buffered single-barrier register-staged schedule (A/B'd
// slower at SD shapes; kept for experiments). GLDS: the DMA schedule above.
// ---------------------------------------------------------------------------
template <int MFRAG, int NFRAG, int BK, bool DBUF>
__global__ __launch_bounds__(256) void conv2d_mfma_kernel(
    const f16* __restrict__ x, const f16* __restrict__ w,
    const float* __restrict__ bias, const f16* __restrict__ cbias,
    const f16* __restrict__ residual, f16* __restrict__ out,
    float* __restrict__ ws, int H, int W, int IC, int HO, int WO, int OC,
    int R, int S, int stride, int pad, int act, int K, int splitk,
    const float* __restrict__ in_aff, int in_act,
    int* __restrict__ counters) {
  constexpr int BM = MFRAG * 32;              // 128 or 64
  constexpr int BNK = NFRAG * 32;             // 64 or 128 out-channels
  constexpr int KPITCH = BK + 8;              // +16B row pad (guide G4)
  constexpr int ALOADS = MFRAG * BK / 64;     // staged 16B A-loads per thread
  constexpr int BLOADS = NFRAG * BK / 64;     // staged 16B B-loads per thread
  constexpr int KSH = BK == 64 ? 3 : 2;       // flat -> (row, k8) shifts
  constexpr int KMSK = BK / 8 - 1;
  constexpr int NBUF = DBUF ? 2 : 1;
  __shared__ f16 ldsA[NBUF * BM * KPITCH];
  __shared__ f16 ldsB[NBUF * BNK * KPITCH];

  const int M = HO * WO;
  const int n_tiles = ceil_div_dev(OC, BNK);
  int mt, nt;
  if (splitk > 0)
    xcd_tile_map(blockIdx.x, gridDim.x, n_tiles, &mt, &nt);
  else {  // splitk<0 encodes "plain mapping" for A/B (|splitk| used below)
    mt = blockIdx.x / n_tiles;
    nt = blockIdx.x - mt * n_tiles;
  }
  const int spk = splitk > 0 ? splitk : -splitk;
  const int m0 = mt * BM;
  const int n0 = nt * BNK;
  const int b = blockIdx.z / spk;
  const int split = blockIdx.z - b * spk;
  const f16* xb = x + (long)b * H * W * IC;
  const float* affb = in_aff ? in_aff + (long)b * IC * 2 : nullptr;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid & 1;
  const int wn = wid >> 1;

  // staging map
  int a_row[ALOADS], a_ho[ALOADS], a_wo[ALOADS], a_k8[ALOADS];
#pragma unroll
  for (int i = 0; i < ALOADS; ++i) {
    int flat = tid + i * 256;           // [0, BM*BK/8)
    a_row[i] = flat >> KSH;
    a_k8[i] = (flat & KMSK) * 8;
    int m = min(m0 + a_row[i], M - 1);
    a_ho[i] = (m / WO) * stride;
    a_wo[i] = (m % WO) * stride;
  }
  int b_row[BLOADS], b_k8[BLOADS];
  const f16* wrow[BLOADS];
#pragma unroll
  for (int i = 0; i < BLOADS; ++i) {
    int flat = tid + i * 256;           // [0, BN*BK/8)
    b_row[i] = flat >> KSH;
    b_k8[i] = (flat & KMSK) * 8;
    wrow[i] = w + (long)min(n0 + b_row[i], OC - 1) * K + b_k8[i];
  }

  f32x4 acc[MFRAG][NFRAG];
#pragma unroll
  for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
    for (int ni = 0; ni < NFRAG; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  // K-step range for this split
  const int nk = K / BK;
  const int per = (nk + spk - 1) / spk;
  const int k_lo = split * per;
  const int k_hi = min(nk, k_lo + per);
  // NOTE: an empty split (uneven tail) still stores its zero slab below —
  // the finalize pass reads every slab.

  // staging helpers
  f16x8 regA[ALOADS], regB[BLOADS];
  auto load_tile = [&](int kt) {
    KPos p = kpos_at(kt * BK, IC, S);
#pragma unroll
    for (int i = 0; i < ALOADS; ++i)
      regA[i] = load_a(xb, a_ho[i], a_wo[i], p.r, p.s, pad, H, W, IC,
                       p.ic0 + a_k8[i], affb, in_act);
#pragma unroll
    for (int i = 0; i < BLOADS; ++i)
      regB[i] = *reinterpret_cast<const f16x8*>(wrow[i] + kt * BK);
  };
  auto write_tile = [&](int buf) {
    f16* la = &ldsA[buf * BM * KPITCH];
    f16* lb = &ldsB[buf * BN * KPITCH];
#pragma unroll
    for (int i = 0; i < ALOADS; ++i)
      *reinterpret_cast<f16x8*>(&la[a_row[i] * KPITCH + a_k8[i]]) = regA[i];
#pragma unroll
    for (int i = 0; i < BLOADS; ++i)
      *reinterpret_cast<f16x8*>(&lb[b_row[i] * KPITCH + b_k8[i]]) = regB[i];
  };
  const int arow_base = wm * (MFRAG * 16) + (lane & 15);
  auto compute_tile = [&](int buf) {
    const f16* la = &ldsA[buf * BM * KPITCH];
    const f16* lb = &ldsB[buf * BNK * KPITCH];
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      const int fcol = kk * 32 + (lane >> 4) * 8;
      f16x8 bfrag[NFRAG];
#pragma unroll
      for (int ni = 0; ni < NFRAG; ++ni)
        bfrag[ni] = *reinterpret_cast<const f16x8*>(
            &lb[(wn * (NFRAG * 16) + ni * 16 + (lane & 15)) * KPITCH + fcol]);
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi) {
        f16x8 afrag = *reinterpret_cast<const f16x8*>(
            &la[(arow_base + mi * 16) * KPITCH + fcol]);
#pragma unroll
        for (int ni = 0; ni < NFRAG; ++ni)
          acc[mi][ni] = mfma16x16x32(afrag, bfrag[ni], acc[mi][ni]);
      }
    }
  };

  if constexpr (!DBUF) {
    // 2-barrier single-buffer loop (T14 prefetch only)
    if (k_lo < k_hi) load_tile(k_lo);
    for (int kt = k_lo; kt < k_hi; ++kt) {
      __syncthreads();
      write_tile(0);
      __syncthreads();
      if (kt + 1 < k_hi) load_tile(kt + 1);
      compute_tile(0);
    }
  } else {
    // ONE barrier per K-step: double-buffered LDS, write tile t+1 AFTER the
    // barrier, re-issue the loads for t+2 immediately (guide T14/G15 form)
    if (k_lo < k_hi) {
      load_tile(k_lo);
      write_tile(0);
      if (k_lo + 1 < k_hi) load_tile(k_lo + 1);
    }
    __syncthreads();
    for (int kt = k_lo; kt < k_hi; ++kt) {
      const int cur = (kt - k_lo) & 1;
      if (kt + 1 < k_hi) {
        write_tile(cur ^ 1);
        if (kt + 2 < k_hi) load_tile(kt + 2);
      }
      compute_tile(cur);
      __syncthreads();
    }
  }

  if (spk == 1) {
    f16* ob = out + (long)b * M * OC;
    const long cb_off = (long)b * OC;
#pragma unroll
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int col = n0 + wn * (NFRAG * 16) + ni * 16 + (lane & 15);
      if (col >= OC) continue;
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int m = m0 + wm * (MFRAG * 16) + mi * 16 + (lane >> 4) * 4 + j;
          if (m < M) {
            const long idx = (long)b * M * OC + (long)m * OC + col;
            ob[(long)m * OC + col] = epilogue(acc[mi][ni][j], bias, cbias,
                                              cb_off, residual, idx, col, act);
          }
        }
    }
  } else {
    // f32 slab store; with `counters` the LAST split block for this
    // (b, mt, nt) tile reduces every slab and applies the epilogue IN
    // KERNEL (threadfence-reduction pattern) — the separate finalize
    // launch disappears (it was 0.66 ms/frame across ~105 launches).
    // Counters are self-cleaning: the last block resets its slot to 0,
    // so one zeroed persistent buffer serves every launch/graph replay.
    float* wsb = ws + ((long)b * spk + split) * M * OC;
#pragma unroll
    for (int ni = 0; ni < NFRAG; ++ni) {
      const int col = n0 + wn * (NFRAG * 16) + ni * 16 + (lane & 15);
      if (col >= OC) continue;
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int m = m0 + wm * (MFRAG * 16) + mi * 16 + (lane >> 4) * 4 + j;
          if (m < M) wsb[(long)m * OC + col] = acc[mi][ni][j];
        }
    }
    if (counters) {
      __threadfence();
      __shared__ int is_last;
      if (tid == 0) {
        const long slot = (long)b * gridDim.x + blockIdx.x;
        const int old = atomicAdd(&counters[slot], 1);
        is_last = (old == spk - 1);
        if (is_last) counters[slot] = 0;
      }
      __syncthreads();
      if (!is_last) return;
      f16* ob = out + (long)b * M * OC;
      const long cb_off = (long)b * OC;
      const float* slab0 = ws + (long)b * spk * M * OC;
      for (int i = tid; i < BM * BNK; i += 256) {
        const int m = m0 + i / BNK;
        const int col = n0 + i % BNK;
        if (m >= M || col >= OC) continue;
        const float* p = slab0 + (long)m * OC + col;
        float a = 0.f;
        for (int sp = 0; sp < spk; ++sp) a += p[(long)sp * M * OC];
        const long oidx = (long)b * M * OC + (long)m * OC + col;
        ob[(long)m * OC + col] =
            epilogue(a, bias, cbias, cb_off, residual, oidx, col, act);
      }
    }
  }
}

template <int MFRAG>
__global__ __launch_bounds__(256) void conv2d_mfma_glds_kernel(
    const f16* __restrict__ x, const f16* __restrict__ w,
    const float* __restrict__ bias, const f16* __restrict__ cbias,
    const f16* __restrict__ residual, f16* __restrict__ out,
    float* __restrict__ ws, int H, int W, int IC, int HO, int WO, int OC,
    int R, int S, int stride, int pad, int act, int K, int splitk) {
  constexpr int BK = 64;
  constexpr int BM = MFRAG * 32;
  constexpr int ALOADS = MFRAG;  // 16B-per-lane glds issues per thread (A)
  constexpr int NBUF = 3;        // 3-deep ring: 2 tiles in flight across barriers
  constexpr int PER_TILE = ALOADS + 2;  // glds instructions per wave per tile
  __shared__ f16 ldsA[NBUF * BM * BK];
  __shared__ f16 ldsB[NBUF * BN * BK];

  const int M = HO * WO;
  const int n_tiles = ceil_div_dev(OC, BN);
  int mt, nt;
  if (splitk > 0)
    xcd_tile_map(blockIdx.x, gridDim.x, n_tiles, &mt, &nt);
  else {
    mt = blockIdx.x / n_tiles;
    nt = blockIdx.x - mt * n_tiles;
  }
  const int spk = splitk > 0 ? splitk : -splitk;
  const int m0 = mt * BM;
  const int n0 = nt * BN;
  const int b = blockIdx.z / spk;
  const int split = blockIdx.z - b * spk;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid & 1;
  const int wn = wid >> 1;

  // buffer descriptors: hardware bounds check gives OOB -> 0 (= the conv's
  // zero padding, free)
  const long img_bytes = (long)H * W * IC * 2;
  auto rsrcA = __builtin_amdgcn_make_buffer_rsrc(
      (void*)(x + (long)b * H * W * IC), (short)0, (int)img_bytes, 0x00020000);
  auto rsrcB = __builtin_amdgcn_make_buffer_rsrc(
      (void*)w, (short)0, (int)((long)OC * K * 2), 0x00020000);
  constexpr unsigned OOB = 0x7ffffff0u;

  // staging map: lane's LDS slot is (row, u_log); it LOADS global unit
  // u_log ^ row_swz(row) (source-side swizzle, guide rule 21)
  int a_ho[ALOADS], a_wo[ALOADS], a_src8[ALOADS];
  bool a_mok[ALOADS];
#pragma unroll
  for (int i = 0; i < ALOADS; ++i) {
    const int flat = tid + i * 256;
    const int row = flat >> 3, u = flat & 7;
    const int m = m0 + row;
    a_mok[i] = m < M;
    const int mm = min(m, M - 1);
    a_ho[i] = (mm / WO) * stride;
    a_wo[i] = (mm % WO) * stride;
    a_src8[i] = (int)(u ^ row_swz(row)) * 8;
  }
  int b_off[2];  // BLOADS = 2 at BK=64
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int flat = tid + i * 256;
    const int row = flat >> 3, u = flat & 7;
    b_off[i] = (min(n0 + row, OC - 1) * K + (int)(u ^ row_swz(row)) * 8) * 2;
  }

  f32x4 acc[MFRAG][2];
#pragma unroll
  for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK;
  const int per = (nk + spk - 1) / spk;
  const int k_lo = split * per;
  const int k_hi = min(nk, k_lo + per);

  auto issue = [&](int buf, int kt) {
    KPos p = kpos_at(kt * BK, IC, S);
#pragma unroll
    for (int i = 0; i < ALOADS; ++i) {
      const int hi = a_ho[i] + p.r - pad;
      const int wi = a_wo[i] + p.s - pad;
      const bool ok = a_mok[i] && (unsigned)hi < (unsigned)H &&
                      (unsigned)wi < (unsigned)W;
      const unsigned voff =
          ok ? (unsigned)((((long)hi * W + wi) * IC + p.ic0 + a_src8[i]) * 2)
             : OOB;
      const int base = __builtin_amdgcn_readfirstlane(
          buf * BM * BK + (i * 256 + wid * 64) * 8);
      __builtin_amdgcn_raw_ptr_buffer_load_lds(
          rsrcA, (__attribute__((address_space(3))) void*)&ldsA[base], 16,
          (int)voff, 0, 0, 0);
    }
    const int kbyte = kt * BK * 2;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int base = __builtin_amdgcn_readfirstlane(
          buf * BN * BK + (i * 256 + wid * 64) * 8);
      __builtin_amdgcn_raw_ptr_buffer_load_lds(
          rsrcB, (__attribute__((address_space(3))) void*)&ldsB[base], 16,
          b_off[i] + kbyte, 0, 0, 0);
    }
  };

  const int arow_base = wm * (MFRAG * 16) + (lane & 15);
  auto compute = [&](int buf) {
    const f16* la = &ldsA[buf * BM * BK];
    const f16* lb = &ldsB[buf * BN * BK];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int u = kk * 4 + (lane >> 4);
      f16x8 bfrag[2];
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int row = wn * 32 + ni * 16 + (lane & 15);
        bfrag[ni] = *reinterpret_cast<const f16x8*>(
            &lb[row * BK + (u ^ row_swz(row)) * 8]);
      }
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi) {
        const int row = arow_base + mi * 16;
        f16x8 afrag = *reinterpret_cast<const f16x8*>(
            &la[row * BK + (u ^ row_swz(row)) * 8]);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = mfma16x16x32(afrag, bfrag[ni], acc[mi][ni]);
      }
    }
  };

  // 3-buffer pipeline with COUNTED vmcnt + raw barriers (guide §5
  // 'Pipelining across barriers'): one tile stays in flight across every
  // barrier; __syncthreads() would drain it (hipcc emits vmcnt(0) inside),
  // so the barrier is the raw s_barrier and the waits are hand-counted.
  // Safety: each wave's counted vmcnt covers its OWN tile-t glds before the
  // barrier; after the barrier every wave's tile t is complete. A buffer is
  // re-issued 3 tiles later — one full barrier after its last ds_read
  // (lgkmcnt(0) drains reads before the barrier).
  if (k_lo < k_hi) issue(0, k_lo);
  if (k_lo + 1 < k_hi) issue(1, k_lo + 1);
  for (int kt = k_lo; kt < k_hi; ++kt) {
    if (kt + 1 < k_hi) {
      // wait tile kt; leave tile kt+1 in flight
      if constexpr (PER_TILE == 6)
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    // the barrier also separates iter t-1's reads (each wave drained its
    // lgkm below before arriving) from this issue's overwrite of buf(t-1+3)
    if (kt + 2 < k_hi) issue((kt + 2 - k_lo) % NBUF, kt + 2);
    compute((kt - k_lo) % NBUF);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }

  // epilogue (same as the register-staged kernel)
  if (spk == 1) {
    f16* ob = out + (long)b * M * OC;
    const long cb_off = (long)b * OC;
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int col = n0 + wn * 32 + ni * 16 + (lane & 15);
      if (col >= OC) continue;
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int m = m0 + wm * (MFRAG * 16) + mi * 16 + (lane >> 4) * 4 + j;
          if (m < M) {
            const long idx = (long)b * M * OC + (long)m * OC + col;
            ob[(long)m * OC + col] = epilogue(acc[mi][ni][j], bias, cbias,
                                              cb_off, residual, idx, col, act);
          }
        }
    }
  } else {
    float* wsb = ws + ((long)b * spk + split) * M * OC;
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int col = n0 + wn * 32 + ni * 16 + (lane & 15);
      if (col >= OC) continue;
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int m = m0 + wm * (MFRAG * 16) + mi * 16 + (lane >> 4) * 4 + j;
          if (m < M) wsb[(long)m * OC + col] = acc[mi][ni][j];
        }
    }
  }
}

// ---------------------------------------------------------------------------
// Halo-tiled 3x3 stride-1 conv: the im2col kernels above re-read every
// activation element ~9x through L2 (the 3x3 gather) — measured as the
// remaining conv bottleneck (TAESD 512²x64ch conv: 302 TF-equivalent but
// ~5x the minimal activation traffic). Here each block owns an 8x8 output
// tile and stages the 10x10 input PATCH in LDS once per 64-deep IC chunk;
// the nine (r,s) taps re-read it from LDS. Weights stream straight from
// global into register B-fragments (every block reads the same lines ->
// L2-hot), so the only barriers are the IC/64 patch swaps.
// Block: 64 px x 64 oc, 4 waves (2 px x 2 oc), wave = 32px x 32oc.
// ---------------------------------------------------------------------------
#define PPITCH_C 72  // patch channel pitch (f16): 64 + 8 pad (bank spread)

__global__ __launch_bounds__(256) void conv3x3_tiled_kernel(
    const f16* __restrict__ x, const f16* __restrict__ w,
    const float* __restrict__ bias, const f16* __restrict__ cbias,
    const f16* __restrict__ residual, f16* __restrict__ out, int H, int W,
    int IC, int OC, int act, int K) {
  __shared__ f16 patch[10 * 10 * PPITCH_C];

  const int px_tiles_w = W >> 3;
  const int pt = blockIdx.x;
  const int tr0 = (pt / px_tiles_w) << 3;
  const int tc0 = (pt - (pt / px_tiles_w) * px_tiles_w) << 3;
  const int n0 = blockIdx.y * BN;
  const long b = blockIdx.z;
  const f16* xb = x + b * (long)H * W * IC;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid & 1;   // px half (32 px)
  const int wn = wid >> 1;  // oc half (32 oc)

  // patch staging map: 10*10 cells x 8 ic-octets = 800 16B loads
  f32x4 acc[2][2];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int nic = IC >> 6;  // 64-deep ic chunks
  for (int icc = 0; icc < nic; ++icc) {
    if (icc) __syncthreads();  // previous chunk's reads complete
    const int ic0 = icc << 6;
    for (int i = tid; i < 800; i += 256) {
      const int cell = i >> 3, icq = i & 7;
      const int pr = cell / 10, pc = cell - pr * 10;
      const int hi = tr0 + pr - 1, wi = tc0 + pc - 1;
      const bool ok = (unsigned)hi < (unsigned)H && (unsigned)wi < (unsigned)W;
      f16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (ok)
        v = *reinterpret_cast<const f16x8*>(
            &xb[((long)hi * W + wi) * IC + ic0 + icq * 8]);
      *reinterpret_cast<f16x8*>(&patch[cell * PPITCH_C + icq * 8]) = v;
    }
    __syncthreads();

    // 18 (tap, K-chunk) steps over this ic chunk; B-fragments are
    // software-pipelined one step ahead so the L2 load latency hides under
    // the current step's MFMAs
    const long ocrow0 = (long)min(n0 + wn * 32 + (lane & 15), OC - 1) * K;
    const long ocrow1 = (long)min(n0 + wn * 32 + 16 + (lane & 15), OC - 1) * K;
    const int klane = (lane >> 4) * 8;
    auto kglob_at = [&](int step) {
      // step = (r*3+sc)*2 + kk
      return (step >> 1) * IC + ic0 + (step & 1) * 32 + klane;
    };
    f16x8 bcur[2], bnext[2];
    {
      const int kg = kglob_at(0);
      bcur[0] = *reinterpret_cast<const f16x8*>(&w[ocrow0 + kg]);
      bcur[1] = *reinterpret_cast<const f16x8*>(&w[ocrow1 + kg]);
    }
#pragma unroll
    for (int step = 0; step < 18; ++step) {
      if (step + 1 < 18) {
        const int kg = kglob_at(step + 1);
        bnext[0] = *reinterpret_cast<const f16x8*>(&w[ocrow0 + kg]);
        bnext[1] = *reinterpret_cast<const f16x8*>(&w[ocrow1 + kg]);
      }
      const int rs = step >> 1, kk = step & 1;
      const int r = rs / 3, sc = rs - (rs / 3) * 3;
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        const int m = wm * 32 + mi * 16 + (lane & 15);
        const int pr = (m >> 3) + r, pc = (m & 7) + sc;
        f16x8 afrag = *reinterpret_cast<const f16x8*>(
            &patch[(pr * 10 + pc) * PPITCH_C + kk * 32 + klane]);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = mfma16x16x32(afrag, bcur[ni], acc[mi][ni]);
      }
      bcur[0] = bnext[0];
      bcur[1] = bnext[1];
    }
  }

  // fused epilogue; output rows = this tile's pixels
  f16* ob = out + b * (long)H * W * OC;
  const long cb_off = b * OC;
#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int col = n0 + wn * 32 + ni * 16 + (lane & 15);
    if (col >= OC) continue;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int m = wm * 32 + mi * 16 + (lane >> 4) * 4 + j;
        const int ho = tr0 + (m >> 3), wo = tc0 + (m & 7);
        const long idx = (b * (long)H * W + (long)ho * W + wo) * OC + col;
        ob[((long)ho * W + wo) * OC + col] =
            epilogue(acc[mi][ni][j], bias, cbias, cb_off, residual, idx, col, act);
      }
  }
}

__global__ void conv_splitk_finalize(const float* __restrict__ ws,
                                     const float* __restrict__ bias,
                                     const f16* __restrict__ cbias,
                                     const f16* __restrict__ residual,
                                     f16* __restrict__ out, int M, int OC,
                                     int splitk, int act, long total) {
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * (long)blockDim.x) {
    const int oc = i % OC;
    const long bm = i / OC;
    const long b = bm / M;
    const float* p = ws + (long)b * splitk * M * OC + (bm - b * M) * OC + oc;
    float a = 0.f;
    for (int s = 0; s < splitk; ++s) a += p[(long)s * M * OC];
    out[i] = epilogue(a, bias, cbias, b * OC, residual, i, oc, act);
  }
}

// ---------------------------------------------------------------------------
// Small-IC conv (conv_in with IC=3/4): one thread per output PIXEL computing
// a 64-wide OC tile in registers, weights staged once in LDS (uniform k ->
// broadcast reads). The one-thread-per-output direct kernel re-reads each
// input pixel OC times (measured 169us for TAESD conv_in @512²); this reads
// x once per pixel (~14 MB total) and streams the output.
// ---------------------------------------------------------------------------
#define OCT 64
#define SMALLIC_MAX_K 288  // 3x3 x IC<32

__global__ __launch_bounds__(256) void conv2d_smallic_kernel(
    const f16* __restrict__ x, const f16* __restrict__ w,
    const float* __restrict__ bias, const f16* __restrict__ cbias,
    const f16* __restrict__ residual, f16* __restrict__ out, int H, int W,
    int IC, int HO, int WO, int OC, int R, int S, int stride, int pad,
    int act, int K, long total_pix, const float* __restrict__ in_aff,
    int in_act) {
  __shared__ f16 wlds[SMALLIC_MAX_K * OCT];
  const int oc0 = blockIdx.y * OCT;
  for (int i = threadIdx.x; i < K * OCT; i += 256) {
    const int k = i / OCT, oc = i - (i / OCT) * OCT;
    wlds[i] = (oc0 + oc < OC) ? w[(long)(oc0 + oc) * K + k] : (f16)0;
  }
  __syncthreads();

  const int M = HO * WO;
  const int noc = min(OCT, OC - oc0);
  for (long pix = blockIdx.x * (long)blockDim.x + threadIdx.x; pix < total_pix;
       pix += gridDim.x * (long)blockDim.x) {
    const int m = pix % M;
    const long b = pix / M;
    const int ho = m / WO, wo = m % WO;
    const f16* xb = x + b * (long)H * W * IC;
    const float* affb = in_aff ? in_aff + b * (long)IC * 2 : nullptr;
    float acc[OCT];
#pragma unroll
    for (int o = 0; o < OCT; ++o) acc[o] = 0.f;
    for (int r = 0; r < R; ++r) {
      const int hi = ho * stride + r - pad;
      if ((unsigned)hi >= (unsigned)H) continue;
      for (int s = 0; s < S; ++s) {
        const int wi = wo * stride + s - pad;
        if ((unsigned)wi >= (unsigned)W) continue;
        const f16* xr = &xb[((long)hi * W + wi) * IC];
        const f16* wr = &wlds[(r * S + s) * IC * OCT];
        for (int c = 0; c < IC; ++c) {
          float xv = (float)xr[c];
          if (affb)
            xv = apply_act(xv * affb[c * 2] + affb[c * 2 + 1], in_act);
          const f16* wv = &wr[c * OCT];
#pragma unroll
          for (int o = 0; o < OCT; ++o) acc[o] += xv * (float)wv[o];
        }
      }
    }
    const long obase = (b * M + m) * (long)OC + oc0;
    const long cb_off = b * OC;
    for (int o = 0; o < noc; ++o)
      out[obase + o] = epilogue(acc[o], bias, cbias, cb_off, residual,
                                obase + o, oc0 + o, act);
  }
}

// ---------------------------------------------------------------------------
// Direct conv fallback (ragged shapes the other kernels exclude)
// ---------------------------------------------------------------------------
__global__ void conv2d_direct_kernel(
    const f16* __restrict__ x, const f16* __restrict__ w,
    const float* __restrict__ bias, const f16* __restrict__ cbias,
    const f16* __restrict__ residual, f16* __restrict__ out, int H, int W,
    int IC, int HO, int WO, int OC, int R, int S, int stride, int pad, int act,
    int K, long total, const float* __restrict__ in_aff, int in_act) {
  const int M = HO * WO;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += gridDim.x * (long)blockDim.x) {
    const int oc = i % OC;
    long rest = i / OC;
    const int m = rest % M;
    const long b = rest / M;
    const int ho = m / WO, wo = m % WO;
    const f16* xb = x + b * (long)H * W * IC;
    const float* affb = in_aff ? in_aff + b * (long)IC * 2 : nullptr;
    const f16* wk = w + (long)oc * K;
    float a = 0.f;
    for (int r = 0; r < R; ++r) {
      const int hi = ho * stride + r - pad;
      if ((unsigned)hi >= (unsigned)H) continue;
      for (int s = 0; s < S; ++s) {
        const int wi = wo * stride + s - pad;
        if ((unsigned)wi >= (unsigned)W) continue;
        const f16* xr = &xb[((long)hi * W + wi) * IC];
        const f16* wr = &wk[(r * S + s) * IC];
        for (int c = 0; c < IC; ++c) {
          float xv = (float)xr[c];
          if (affb)
            xv = apply_act(xv * affb[c * 2] + affb[c * 2 + 1], in_act);
          a += xv * (float)wr[c];
        }
      }
    }
    out[i] = epilogue(a, bias, cbias, b * OC, residual, i, oc, act);
  }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
static int conv_bn() {
  static int bn = -1;
  if (bn < 0) {
    const char* e = getenv("AIRTC_CONV_BN128");
    bn = (e && atoi(e)) ? 128 : 64;
  }
  return bn;
}

extern "C" int airtc_conv2d_splitk_for(int B, int HO, int WO, int OC, int IC) {
  // path/geometry decision, exported so the host can size the workspace:
  // returns 0 = direct, +k = BM128 split-K k, -k = BM64 split-K k.
  // Target ~2 blocks/CU (>=480 workgroups): at 1 block/CU a wave can only
  // hide latency with its own ILP (measured: 64x64x320 conv at 160 blocks
  // ran 87us = 86 TF; split-K over the K loop fills the chip).
  if (IC % 32 != 0) return 0;
  const int BNSEL = conv_bn();
  const int M = HO * WO;
  // NOTE: a BM256 (MFRAG8) geometry measured SLOWER on the TAESD hi-res
  // layers (24.5 -> 37.1us @256²x64ch: fewer blocks + bigger staging
  // footprint cost more than the barrier amortisation bought) — path 100
  // exists but is never selected.
  if (M >= 2048) {
    const long blocks = (long)ceil_div(M, 128) * ceil_div(OC, BNSEL) * B;
    long k = (480 + blocks - 1) / blocks;
    if (k < 1) k = 1;
    if (k > 8) k = 8;
    return (int)k;
  }
  const long blocks64 = (long)ceil_div(M, 64) * ceil_div(OC, BNSEL) * B;
  long k = (512 + blocks64 - 1) / blocks64;
  if (k < 1) k = 1;
  if (k > 32) k = 32;
  return (int)(-k);
}

extern "C" void airtc_conv2d_mfma(const uint16_t* x, const uint16_t* w,
                                  const float* bias, const uint16_t* cbias,
                                  const uint16_t* residual, uint16_t* out,
                                  float* ws, int B, int H, int W, int IC,
                                  int HO, int WO, int OC, int R, int S,
                                  int stride, int pad, int act, int path,
                                  const float* in_aff, int in_act,
                                  int* counters, hipStream_t s) {
  const int K = R * S * IC;
  const f16* xp = reinterpret_cast<const f16*>(x);
  const f16* wp = reinterpret_cast<const f16*>(w);
  const f16* cb = reinterpret_cast<const f16*>(cbias);
  const f16* res = reinterpret_cast<const f16*>(residual);
  f16* op = reinterpret_cast<f16*>(out);
  const int M = HO * WO;

  static int tiled = -1;
  if (tiled < 0) {
    // Measured SLOWER than im2col (109.5-110.4 vs 119.2-119.6 fps, two
    // same-box A/Bs, with and without B-prefetch): the 3x3 activation
    // re-gather is already L2-absorbed, while per-wave B-fragment loads
    // cost 4x the issue of cooperative LDS staging. Kept for experiments.
    const char* t = getenv("AIRTC_CONV_TILED");
    tiled = t ? atoi(t) : 0;
  }
  // fused input-affine runs on the register-staged schedule only (the
  // glds DMA path writes HBM straight to LDS; the tiled path has its own
  // gather) — both are env-gated experiments anyway
  if (!in_aff && tiled && R == 3 && S == 3 && stride == 1 && pad == 1 &&
      IC % 64 == 0 && (H & 7) == 0 && (W & 7) == 0 && M >= 4096) {
    dim3 tgrid((H >> 3) * (W >> 3), ceil_div(OC, BN), B);
    hipLaunchKernelGGL(conv3x3_tiled_kernel, tgrid, dim3(256), 0, s, xp, wp,
                       bias, cb, res, op, H, W, IC, OC, act, K);
    return;
  }
  const int splitk = path > 0 ? path : -path;
  const int bm = path > 0 ? 128 : 64;
  const int bn = (IC % 64 == 0) ? conv_bn() : 64;  // BN128 needs BK64 staging
  dim3 grid(ceil_div(M, bm) * ceil_div(OC, bn), 1, B * splitk);
  // In-kernel fused finalize measured 2.2x SLOWER end-to-end (128.5 ->
  // 57.4 fps): the device-scope __threadfence each split block must issue
  // before its counter increment is an L2 writeback on CDNA, destroying
  // the per-XCD L2 reuse that feeds the K-loop — far costlier than the
  // ~105 finalize launches it removes. Env-gated for experiments
  // (AIRTC_CONV_FUSED_FIN=1); default stays the two-pass scheme.
  static int ffin_env = -1;
  if (ffin_env < 0) {
    const char* e = getenv("AIRTC_CONV_FUSED_FIN");
    ffin_env = e ? atoi(e) : 0;
  }
  const bool fuse_fin = ffin_env && counters != nullptr && splitk > 1;
  const float* b1 = (splitk == 1 || fuse_fin) ? bias : nullptr;
  const f16* cb1 = (splitk == 1 || fuse_fin) ? cb : nullptr;
  const f16* res1 = (splitk == 1 || fuse_fin) ? res : nullptr;
  const bool bk64 = (IC % 64 == 0);
  static int dbuf = -1, glds = -1, xcdmap = -1;
  if (dbuf < 0) {
    // Measured A/B on MI355X: the double-buffer single-barrier REGISTER
    // schedule is SLOWER here (114.8 vs 118.6 fps end-to-end) — the 2x LDS
    // footprint costs more occupancy than the barrier removal buys at these
    // tiles (guide §5.5: regime-gated). Default stays 2-barrier.
    const char* e = getenv("AIRTC_CONV_DBUF");
    dbuf = e ? atoi(e) : 0;
    const char* g = getenv("AIRTC_CONV_GLDS");
    glds = g ? atoi(g) : 0;
    // XCD-chunked mapping measured NEUTRAL (119.9 vs 120.3 fps plain):
    // the 256 MB LLC already absorbs the A-panel re-reads at SD shapes.
    const char* xm = getenv("AIRTC_CONV_XCD");
    xcdmap = xm ? atoi(xm) : 0;
  }
  // splitk sign selects the tile mapping (positive = XCD-chunked, negative
  // = plain) — kernels take |splitk| as the split factor
  const int spk_arg = xcdmap ? splitk : -splitk;
#define CONV_LAUNCH(MF, NF, BKV, DB)                                          \
  hipLaunchKernelGGL((conv2d_mfma_kernel<MF, NF, BKV, DB>), grid, dim3(256),  \
                     0, s, xp, wp, b1, cb1, res1, op, ws, H, W, IC, HO, WO,   \
                     OC, R, S, stride, pad, act, K, spk_arg, in_aff, in_act,  \
                     fuse_fin ? counters : nullptr)
#define CONV_LAUNCH_GLDS(MF)                                                  \
  hipLaunchKernelGGL((conv2d_mfma_glds_kernel<MF>), grid, dim3(256), 0, s,    \
                     xp, wp, b1, cb1, res1, op, ws, H, W, IC, HO, WO, OC, R,  \
                     S, stride, pad, act, K, spk_arg)
  const bool bn128 = bn == 128;
  const int use_glds = glds && !in_aff && !fuse_fin;
  if (path > 0) {
    if (bk64 && use_glds) CONV_LAUNCH_GLDS(4);
    else if (bk64 && bn128) CONV_LAUNCH(4, 4, 64, false);
    else if (bk64) { if (dbuf) CONV_LAUNCH(4, 2, 64, true); else CONV_LAUNCH(4, 2, 64, false); }
    else CONV_LAUNCH(4, 2, 32, false);
  } else {
    if (bk64 && use_glds) CONV_LAUNCH_GLDS(2);
    else if (bk64 && bn128) CONV_LAUNCH(2, 4, 64, false);
    else if (bk64) { if (dbuf) CONV_LAUNCH(2, 2, 64, true); else CONV_LAUNCH(2, 2, 64, false); }
    else CONV_LAUNCH(2, 2, 32, false);
  }
#undef CONV_LAUNCH
#undef CONV_LAUNCH_GLDS
  if (splitk > 1 && !fuse_fin) {
    long total = (long)B * M * OC;
    int blocks = (int)min((long)2048, (total + 255) / 256);
    hipLaunchKernelGGL(conv_splitk_finalize, dim3(blocks), dim3(256), 0, s, ws,
                       bias, cb, res, op, M, OC, splitk, act, total);
  }
}

extern "C" void airtc_conv2d_direct(const uint16_t* x, const uint16_t* w,
                                    const float* bias, const uint16_t* cbias,
                                    const uint16_t* residual, uint16_t* out,
                                    int B, int H, int W, int IC, int HO,
                                    int WO, int OC, int R, int S, int stride,
                                    int pad, int act, const float* in_aff,
                                    int in_act, hipStream_t s) {
  const int K = R * S * IC;
  if (K <= SMALLIC_MAX_K) {
    long pix = (long)B * HO * WO;
    dim3 grid((uint32_t)min((long)4096, (pix + 255) / 256),
              ceil_div(OC, OCT));
    hipLaunchKernelGGL(conv2d_smallic_kernel, grid, dim3(256), 0, s,
                       reinterpret_cast<const f16*>(x),
                       reinterpret_cast<const f16*>(w), bias,
                       reinterpret_cast<const f16*>(cbias),
                       reinterpret_cast<const f16*>(residual),
                       reinterpret_cast<f16*>(out), H, W, IC, HO, WO, OC, R,
                       S, stride, pad, act, K, pix, in_aff, in_act);
    return;
  }
  long total = (long)B * HO * WO * OC;
  int blocks = (int)min((long)4096, (total + 255) / 256);
  hipLaunchKernelGGL(conv2d_direct_kernel, dim3(blocks), dim3(256), 0, s,
                     reinterpret_cast<const f16*>(x),
                     reinterpret_cast<const f16*>(w), bias,
                     reinterpret_cast<const f16*>(cbias),
                     reinterpret_cast<const f16*>(residual),
                     reinterpret_cast<f16*>(out), H, W, IC, HO, WO, OC, R, S,
                     stride, pad, act, K, total, in_aff, in_act);
}
