// Conv2d NHWC — fp8 (OCP e4m3) implicit-GEMM on the MX-scaled MFMA path.
//
// The CDNA4 2x-rate tier: v_mfma_scale_f32_16x16x128_f8f6f4 (block-scaled
// MX, the only large-K low-precision MFMA on gfx950 — non-scaled fp8 runs
// at the bf16 rate). Facts verified on hardware by tools/fp8_probe.py:
//   - A: lane l holds A[row=l&15][k=(l>>4)*32 + j], j=0..31 linear bytes
//     (B same law over [N][K]); D: lane l reg j -> D[(l>>4)*4+j][l&15]
//   - e8m0 scale byte b = 2^(b-127); we run HW scales at 1.0 (127) and do
//     software per-tensor/per-channel scaling instead
//   - v_cvt_scalef32_pk_fp8_* DIVIDES by its scale on encode, MULTIPLIES
//     on decode; encode at scale 1 is bit-identical to torch float8_e4m3fn
//
// Quantization scheme (serving tier, ROADMAP item 1):
//   activations: per-tensor scale sa, quantized IN THE STAGING LOADS
//     (f16 -> fp8 with the fused GN affine + activation applied first,
//     exactly like the f16 kernel's load-time transform) — intermediate
//     tensors stay f16, nothing else in the pipeline changes
//   weights: pre-quantized per-out-channel on the host ([OC][K] bytes)
//   dequant: one multiply per output element by dq[oc] = sa * sw[oc]
//     (before bias/residual/activation, which run in f32 as always)
//
// Geometry mirrors conv2d.hip: BM=128/64 x BN=64, 4 waves (2x2), BK=128
// fp8 BYTES per K-tile — byte-for-byte the same LDS image as the proven
// f16 BK=64 tile (144B row pitch, same bank law), with the 2-barrier
// register-staged schedule that measured fastest there. Split-K stores
// PRE-DEQUANTIZED f32 slabs so the shared conv_splitk_finalize pass is
// reused unchanged.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(4))) int i32x4;
typedef __attribute__((ext_vector_type(2))) short s16x2;
typedef __attribute__((ext_vector_type(2))) _Float16 hf16x2;

#define BN8 64
#define BK8 128  // fp8 bytes per K-tile (= K elements)

struct KPos8 {
  int r, s, ic0;
};

__device__ __forceinline__ KPos8 kpos8_at(int k0, int IC, int S) {
  const int rs = k0 / IC;
  return {rs / S, rs - (rs / S) * S, k0 - rs * IC};
}

// D = A(16x128) @ B(128x16) + C, both operands fp8 e4m3, HW block scales 1.0
__device__ __forceinline__ f32x4 mfma_mx_fp8(i32x8 a, i32x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(a, b, c, 0, 0, 0,
                                                          0x7F, 0, 0x7F);
}

// stage 16 fp8 bytes from 16 f16 inputs: optional per-channel affine +
// activation (f32 math), then quantize: q = clamp(v * (1/sa), +-448) via
// the NON-scaled v_cvt_pk_fp8_f32 — bit-identical to torch's e4m3 RNE.
// MEASURED HAZARD (tools/fp8_debug.py): v_cvt_scalef32_pk_fp8_* at
// arbitrary scales rounds every code UP one ULP, turning max-magnitude
// codes (126) into NaN (127); the scalef32 forms are NOT used here. The
// clamp also covers post-calibration drift (HW encode overflows to NaN,
// it does not saturate). OOB -> 0.
__device__ __forceinline__ i32x4 load_a_fp8(const f16* xb, int ho_s, int wo_s,
                                            int r, int s, int pad, int H,
                                            int W, int IC, int ic,
                                            const float* aff, int in_act,
                                            float inv_sa, bool kok) {
  const int hi = ho_s + r - pad;
  const int wi = wo_s + s - pad;
  const bool ok = kok && (unsigned)hi < (unsigned)H && (unsigned)wi < (unsigned)W;
  if (!ok) return i32x4{0, 0, 0, 0};
  const f16* src = &xb[((long)hi * W + wi) * IC + ic];
  const f16x8 lo = *reinterpret_cast<const f16x8*>(src);      // two b128 loads
  const f16x8 hi8 = *reinterpret_cast<const f16x8*>(src + 8);
  i32x4 outv;
  if (aff) {
    // affine path (env-gated AIRTC_FUSE_GN): f32 math, non-scaled encode
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int j0 = (q & 1) * 4;
      const f16x8& vsrc = (q < 2) ? lo : hi8;
      float a[4];
#pragma unroll
      for (int h = 0; h < 4; ++h) {
        const int c = ic + (q >> 1) * 8 + j0 + h;
        float v = apply_act((float)vsrc[j0 + h] * aff[c * 2] + aff[c * 2 + 1],
                            in_act);
        // min(max()) lowers to v_med3_f32; 448 = e4m3 max
        a[h] = fminf(fmaxf(v * inv_sa, -448.0f), 448.0f);
      }
      int p2 = 0;
      p2 = __builtin_amdgcn_cvt_pk_fp8_f32(a[0], a[1], p2, false);
      p2 = __builtin_amdgcn_cvt_pk_fp8_f32(a[2], a[3], p2, true);
      outv[q] = p2;
    }
    return outv;
  }
  // default path: stay in f16 — packed clamp + packed mul, then the
  // scale-convert AT SCALE 1.0 (measured bit-exact RNE there; the +1-ULP
  // rounding hazard only appears at arbitrary scales). The f16 interme-
  // diate adds <=2^-10 relative double-rounding noise — far below e4m3's
  // 2^-4 step.
  const f16 lim = (f16)(448.0f / inv_sa);
  const f16 invh = (f16)inv_sa;
  const hf16x2 limv = {lim, lim}, nlimv = {(f16)(-lim), (f16)(-lim)};
  const hf16x2 inv2 = {invh, invh};
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    const int j0 = (q & 1) * 4;
    const f16x8& vsrc = (q < 2) ? lo : hi8;
    hf16x2 v0 = {vsrc[j0], vsrc[j0 + 1]};
    hf16x2 v1 = {vsrc[j0 + 2], vsrc[j0 + 3]};
    v0 = __builtin_elementwise_min(__builtin_elementwise_max(v0, nlimv), limv) * inv2;
    v1 = __builtin_elementwise_min(__builtin_elementwise_max(v1, nlimv), limv) * inv2;
    s16x2 packed = {0, 0};
    packed = __builtin_amdgcn_cvt_scalef32_pk_fp8_f16(packed, v0, 1.0f, false);
    packed = __builtin_amdgcn_cvt_scalef32_pk_fp8_f16(packed, v1, 1.0f, true);
    outv[q] = (int)(((unsigned short)packed[0]) |
                    (((unsigned)(unsigned short)packed[1]) << 16));
  }
  return outv;
}

// pre-quantized input (u8 e4m3 codes from the producing kernel): staging
// is a straight 16B byte copy — zero encode VALU, half the global traffic
__device__ __forceinline__ i32x4 load_a_q8(const uint8_t* xb, int ho_s,
                                           int wo_s, int r, int s, int pad,
                                           int H, int W, int IC, int ic,
                                           bool kok) {
  const int hi = ho_s + r - pad;
  const int wi = wo_s + s - pad;
  const bool ok = kok && (unsigned)hi < (unsigned)H && (unsigned)wi < (unsigned)W;
  if (!ok) return i32x4{0, 0, 0, 0};
  return *reinterpret_cast<const i32x4*>(&xb[((long)hi * W + wi) * IC + ic]);
}

// fused epilogue (dequant -> bias -> cbias -> residual -> act), f32 math
__device__ __forceinline__ float epilogue_fp8_val(float acc, float dq,
                                                  const float* bias,
                                                  const f16* cbias, long cb_off,
                                                  const f16* residual, long idx,
                                                  int oc, int act) {
  float v = acc * dq;
  if (bias) v += bias[oc];
  if (cbias) v += (float)cbias[cb_off + oc];
  if (residual) v += (float)residual[idx];
  return apply_act(v, act);
}

__device__ __forceinline__ f16 epilogue_fp8(float acc, float dq,
                                            const float* bias, const f16* cbias,
                                            long cb_off, const f16* residual,
                                            long idx, int oc, int act) {
  return (f16)epilogue_fp8_val(acc, dq, bias, cbias, cb_off, residual, idx, oc,
                               act);
}

// quantized-output store: code = e4m3(clamp(v * out_inv, +-448)) — the
// CONSUMING fp8 conv then stages these bytes directly (chained fp8 layers,
// e.g. the TAESD conv stacks)
__device__ __forceinline__ uint8_t q8_of(float v, float out_inv) {
  const float a = fminf(fmaxf(v * out_inv, -448.0f), 448.0f);
  const int p = __builtin_amdgcn_cvt_pk_fp8_f32(a, a, 0, false);
  return (uint8_t)(p & 0xFF);
}

// XCD-aware mapping shared with conv2d.hip (same dispatcher law)
__device__ __forceinline__ void xcd_tile_map8(int wg, int nwg, int n_tiles,
                                              int* mt, int* nt) {
  const int x = wg & 7, pos = wg >> 3;
  const int q = nwg >> 3, r = nwg & 7;
  const int wgid = (x < r ? x * (q + 1) : r * (q + 1) + (x - r) * q) + pos;
  *nt = wgid % n_tiles;
  *mt = wgid / n_tiles;
}

// ---------------------------------------------------------------------------
// MFRAG = 4 (BM=128) or 2 (BM=64); BN fixed 64 (NFRAG=2), 4 waves as 2x2.
// Requires IC % 64 == 0 (every 64-aligned K run sits inside one (r,s) tap)
// — the same constraint as the f16 BK=64 path, covering all SD/TAESD
// MFMA layers. K may be any multiple of 64 (tail tile zero-padded).
// ---------------------------------------------------------------------------
template <int MFRAG, typename XT>  // XT = f16 (inline encode) | uint8_t (codes)
__global__ __launch_bounds__(256) void conv2d_mfma_fp8_kernel(
    const XT* __restrict__ x, const uint8_t* __restrict__ w,
    const float* __restrict__ dq, const float* __restrict__ bias,
    const f16* __restrict__ cbias, const f16* __restrict__ residual,
    f16* __restrict__ out, float* __restrict__ ws, int H, int W, int IC,
    int HO, int WO, int OC, int R, int S, int stride, int pad, int act, int K,
    int splitk, const float* __restrict__ in_aff, int in_act, float sa,
    uint8_t* __restrict__ out_q8, float out_inv) {
  constexpr int BM = MFRAG * 32;
  constexpr int KPITCH = BK8 + 16;            // bytes; same 144B law as f16
  constexpr int ALOADS = MFRAG;               // 16B units: BM*8/256
  constexpr int BLOADS = 2;                   // BN8*8/256
  __shared__ uint8_t ldsA[BM * KPITCH];
  __shared__ uint8_t ldsB[BN8 * KPITCH];

  const int M = HO * WO;
  const int n_tiles = ceil_div_dev(OC, BN8);
  int mt, nt;
  if (splitk > 0)
    xcd_tile_map8(blockIdx.x, gridDim.x, n_tiles, &mt, &nt);
  else {
    mt = blockIdx.x / n_tiles;
    nt = blockIdx.x - mt * n_tiles;
  }
  const int spk = splitk > 0 ? splitk : -splitk;
  const int m0 = mt * BM;
  const int n0 = nt * BN8;
  const int b = blockIdx.z / spk;
  const int split = blockIdx.z - b * spk;
  const XT* xb = x + (long)b * H * W * IC;
  const float* affb = in_aff ? in_aff + (long)b * IC * 2 : nullptr;
  const float inv_sa = 1.0f / sa;  // encode multiplies; clamp at +-448

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid & 1;
  const int wn = wid >> 1;

  // staging map: flat unit index -> (row, 16B unit u); u's 64-aligned
  // sub-block (u>>2) picks which of the tile's two taps applies
  int a_row[ALOADS], a_ho[ALOADS], a_wo[ALOADS], a_u[ALOADS];
#pragma unroll
  for (int i = 0; i < ALOADS; ++i) {
    const int flat = tid + i * 256;  // [0, BM*8)
    a_row[i] = flat >> 3;
    a_u[i] = flat & 7;
    const int m = min(m0 + a_row[i], M - 1);
    a_ho[i] = (m / WO) * stride;
    a_wo[i] = (m % WO) * stride;
  }
  int b_row[BLOADS], b_u[BLOADS];
#pragma unroll
  for (int i = 0; i < BLOADS; ++i) {
    const int flat = tid + i * 256;  // [0, BN8*8)
    b_row[i] = flat >> 3;
    b_u[i] = flat & 7;
  }

  f32x4 acc[MFRAG][2];
#pragma unroll
  for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int nk = ceil_div_dev(K, BK8);
  const int per = (nk + spk - 1) / spk;
  const int k_lo = split * per;
  const int k_hi = min(nk, k_lo + per);

  i32x4 regA[ALOADS], regB[BLOADS];
  auto load_tile = [&](int kt) {
    const int kbase = kt * BK8;
    const KPos8 p0 = kpos8_at(kbase, IC, S);
    const KPos8 p1 =
        (kbase + 64 < K) ? kpos8_at(kbase + 64, IC, S) : KPos8{0, 0, 0};
#pragma unroll
    for (int i = 0; i < ALOADS; ++i) {
      const KPos8& p = (a_u[i] >= 4) ? p1 : p0;
      const int kg = kbase + a_u[i] * 16;
      if constexpr (__is_same(XT, uint8_t))
        regA[i] = load_a_q8(xb, a_ho[i], a_wo[i], p.r, p.s, pad, H, W, IC,
                            p.ic0 + (a_u[i] & 3) * 16, kg < K);
      else
        regA[i] = load_a_fp8(xb, a_ho[i], a_wo[i], p.r, p.s, pad, H, W, IC,
                             p.ic0 + (a_u[i] & 3) * 16, affb, in_act, inv_sa,
                             kg < K);
    }
#pragma unroll
    for (int i = 0; i < BLOADS; ++i) {
      const int kg = kbase + b_u[i] * 16;
      regB[i] = (kg < K)
                    ? *reinterpret_cast<const i32x4*>(
                          &w[(long)min(n0 + b_row[i], OC - 1) * K + kg])
                    : i32x4{0, 0, 0, 0};
    }
  };
  auto write_tile = [&]() {
#pragma unroll
    for (int i = 0; i < ALOADS; ++i)
      *reinterpret_cast<i32x4*>(&ldsA[a_row[i] * KPITCH + a_u[i] * 16]) =
          regA[i];
#pragma unroll
    for (int i = 0; i < BLOADS; ++i)
      *reinterpret_cast<i32x4*>(&ldsB[b_row[i] * KPITCH + b_u[i] * 16]) =
          regB[i];
  };
  const int arow_base = wm * (MFRAG * 16) + (lane & 15);
  const int fcol = (lane >> 4) * 32;  // 32 consecutive fp8 bytes per lane
  auto compute_tile = [&]() {
    i32x8 bfrag[2];
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const uint8_t* src =
          &ldsB[(wn * 32 + ni * 16 + (lane & 15)) * KPITCH + fcol];
      *reinterpret_cast<i32x4*>(&bfrag[ni]) =
          *reinterpret_cast<const i32x4*>(src);
      *(reinterpret_cast<i32x4*>(&bfrag[ni]) + 1) =
          *reinterpret_cast<const i32x4*>(src + 16);
    }
#pragma unroll
    for (int mi = 0; mi < MFRAG; ++mi) {
      const uint8_t* src = &ldsA[(arow_base + mi * 16) * KPITCH + fcol];
      i32x8 afrag;
      *reinterpret_cast<i32x4*>(&afrag) =
          *reinterpret_cast<const i32x4*>(src);
      *(reinterpret_cast<i32x4*>(&afrag) + 1) =
          *reinterpret_cast<const i32x4*>(src + 16);
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        acc[mi][ni] = mfma_mx_fp8(afrag, bfrag[ni], acc[mi][ni]);
    }
  };

  // 2-barrier single-buffer loop with T14 prefetch (the schedule that
  // measured fastest for the f16 kernel at SD shapes)
  if (k_lo < k_hi) load_tile(k_lo);
  for (int kt = k_lo; kt < k_hi; ++kt) {
    __syncthreads();
    write_tile();
    __syncthreads();
    if (kt + 1 < k_hi) load_tile(kt + 1);
    compute_tile();
  }

  if (spk == 1) {
    f16* ob = out + (long)b * M * OC;
    uint8_t* oq = out_q8 ? out_q8 + (long)b * M * OC : nullptr;
    const long cb_off = (long)b * OC;
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int col = n0 + wn * 32 + ni * 16 + (lane & 15);
      if (col >= OC) continue;
      const float d = dq[col];
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int m = m0 + wm * (MFRAG * 16) + mi * 16 + (lane >> 4) * 4 + j;
          if (m < M) {
            const long idx = (long)b * M * OC + (long)m * OC + col;
            const float v = epilogue_fp8_val(acc[mi][ni][j], d, bias, cbias,
                                             cb_off, residual, idx, col, act);
            if (oq)
              oq[(long)m * OC + col] = q8_of(v, out_inv);
            else
              ob[(long)m * OC + col] = (f16)v;
          }
        }
    }
  } else {
    // pre-dequantized f32 slab: the shared finalize pass then applies the
    // bias/residual/act epilogue with no fp8 knowledge
    float* wsb = ws + ((long)b * spk + split) * M * OC;
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int col = n0 + wn * 32 + ni * 16 + (lane & 15);
      if (col >= OC) continue;
      const float d = dq[col];
#pragma unroll
      for (int mi = 0; mi < MFRAG; ++mi)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int m = m0 + wm * (MFRAG * 16) + mi * 16 + (lane >> 4) * 4 + j;
          if (m < M) wsb[(long)m * OC + col] = acc[mi][ni][j] * d;
        }
    }
  }
}

// finalize pass shared with the f16 kernel (conv2d.hip)
__global__ void conv_splitk_finalize_fp8(const float* __restrict__ ws,
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
    a = epilogue_fp8(a, 1.0f, bias, cbias, b * OC, residual, i, oc, act);
    out[i] = (f16)a;
  }
}

extern "C" void airtc_conv2d_fp8_mfma(
    const uint16_t* x, const uint8_t* w_fp8, const float* dq,
    const float* bias, const uint16_t* cbias, const uint16_t* residual,
    uint16_t* out, float* ws, int B, int H, int W, int IC, int HO, int WO,
    int OC, int R, int S, int stride, int pad, int act, int path,
    const float* in_aff, int in_act, float a_scale, int x_is_q8,
    uint8_t* out_q8, float out_inv, hipStream_t s) {
  const int K = R * S * IC;
  const f16* cb = reinterpret_cast<const f16*>(cbias);
  const f16* res = reinterpret_cast<const f16*>(residual);
  f16* op = reinterpret_cast<f16*>(out);
  const int M = HO * WO;

  const int splitk = path > 0 ? path : -path;
  const int bm = path > 0 ? 128 : 64;
  dim3 grid(ceil_div(M, bm) * ceil_div(OC, BN8), 1, B * splitk);
  const float* b1 = splitk == 1 ? bias : nullptr;
  const f16* cb1 = splitk == 1 ? cb : nullptr;
  const f16* res1 = splitk == 1 ? res : nullptr;
  // tile mapping selection mirrors conv2d.hip (XCD map env-gated there;
  // fp8 uses the measured default = plain mapping, sign encodes it)
#define FP8_LAUNCH(MF, XT, XP)                                                \
  hipLaunchKernelGGL((conv2d_mfma_fp8_kernel<MF, XT>), grid, dim3(256), 0, s, \
                     XP, w_fp8, dq, b1, cb1, res1, op, ws, H, W, IC, HO, WO,  \
                     OC, R, S, stride, pad, act, K, -splitk, in_aff, in_act,  \
                     a_scale, splitk == 1 ? out_q8 : nullptr, out_inv)
  if (x_is_q8) {
    const uint8_t* xq = reinterpret_cast<const uint8_t*>(x);
    if (path > 0) FP8_LAUNCH(4, uint8_t, xq);
    else FP8_LAUNCH(2, uint8_t, xq);
  } else {
    const f16* xp = reinterpret_cast<const f16*>(x);
    if (path > 0) FP8_LAUNCH(4, f16, xp);
    else FP8_LAUNCH(2, f16, xp);
  }
#undef FP8_LAUNCH
  if (splitk > 1) {
    long total = (long)B * M * OC;
    int blocks = (int)min((long)2048, (total + 255) / 256);
    hipLaunchKernelGGL(conv_splitk_finalize_fp8, dim3(blocks), dim3(256), 0, s,
                       ws, bias, cb, res, op, M, OC, splitk, act, total);
  }
}
