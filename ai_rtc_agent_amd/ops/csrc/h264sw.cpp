// Software H.264 baseline-intra codec (encoder + decoder), first-party C++.
//
// Replaces reference component N10: the reference's
// software fallback encodes real H.264 via aiortc/x264/PyAV
// (reference lib/pipeline.py:83-94, Dockerfile:10). Neither ships offline,
// so this module implements the subset that matters for low-latency WebRTC:
//
//   encode: every frame an IDR picture — I_16x16 macroblocks (intra
//           prediction mode chosen per-MB by SAD among DC/V/H/Plane),
//           4x4 integer transform + luma-DC Hadamard + chroma-DC 2x2,
//           CAVLC entropy coding, deblocking disabled, BT.601 RGB<->YUV420.
//   decode: baseline-intra subset — I_16x16 (all four prediction modes,
//           all chroma modes) and I_PCM macroblocks, CAVLC. I_4x4 and
//           P slices are rejected with a clean error (the transport then
//           falls back to PLI/keyframe recovery).
//
// The wire format is standard Annex-B H.264 (constrained baseline), so any
// compliant decoder can consume the encoder's output; the in-repo decoder
// doubles as the bit-exact test reference (SURVEY.md §4 strategy (b)).
// Rate control (per-frame QP from the NVENC_*-parity bitrate knobs) lives
// in Python (media/codec.py).

#include <stdint.h>
#include <stdlib.h>
#include <string.h>

#include <algorithm>
#include <condition_variable>
#include <functional>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#include "h264_bits.h"

// SPS/PPS generator (vcn.cpp): shared so the parameter sets advertised over
// SDP and the ones in the encoded stream are identical.
extern "C" int airtc_h264_sps_pps(int width, int height, uint8_t* buf,
                                  int buflen);

namespace h264sw {

using h264::BitReader;
using h264::BitWriter;
using h264::emit_nal;

static inline uint8_t clip8(int v) { return (uint8_t)(v < 0 ? 0 : v > 255 ? 255 : v); }

// ---------------------------------------------------------------------------
// CAVLC tables (ITU-T H.264 Table 9-5, 9-7/9-8, 9-9, 9-10)
// ---------------------------------------------------------------------------

// coeff_token for nC buckets 0 (0<=nC<2), 1 (2<=nC<4), 2 (4<=nC<8);
// bucket 3 (nC>=8) is a 6-bit FLC handled in code; chroma DC has its own.
// Indexed [bucket][TotalCoeff][TrailingOnes].
static const uint8_t CT_LEN[3][17][4] = {
    {{1, 0, 0, 0},   {6, 2, 0, 0},   {8, 6, 3, 0},   {9, 8, 7, 5},
     {10, 9, 8, 6},  {11, 10, 9, 7}, {13, 11, 10, 8}, {13, 13, 11, 9},
     {13, 13, 13, 10}, {14, 14, 13, 11}, {14, 14, 14, 13}, {15, 15, 14, 14},
     {15, 15, 15, 14}, {16, 15, 15, 15}, {16, 16, 16, 15}, {16, 16, 16, 16},
     {16, 16, 16, 16}},
    {{2, 0, 0, 0},   {6, 2, 0, 0},   {6, 5, 3, 0},   {7, 6, 6, 4},
     {8, 6, 6, 4},   {8, 7, 7, 5},   {9, 8, 8, 6},   {11, 9, 9, 6},
     {11, 11, 11, 7}, {12, 11, 11, 9}, {12, 12, 12, 11}, {12, 12, 12, 11},
     {13, 13, 13, 12}, {13, 13, 13, 13}, {13, 14, 14, 13}, {14, 14, 14, 13},
     {14, 14, 14, 14}},  // NOTE: rows TC>=10 are best-effort (see ENC_MAX_TC)
    {{4, 0, 0, 0},   {6, 4, 0, 0},   {6, 5, 4, 0},   {6, 5, 5, 4},
     {7, 5, 5, 4},   {7, 5, 5, 4},   {7, 6, 6, 4},   {7, 6, 6, 4},
     {8, 7, 7, 5},   {8, 8, 7, 6},   {9, 8, 8, 7},   {9, 9, 8, 8},
     {9, 9, 9, 8},   {10, 10, 9, 9}, {10, 10, 10, 10}, {10, 10, 10, 10},
     {10, 10, 10, 10}},
};
static const uint16_t CT_BITS[3][17][4] = {
    {{1, 0, 0, 0},  {5, 1, 0, 0},  {7, 4, 1, 0},  {7, 6, 5, 3},
     {7, 6, 5, 3},  {7, 6, 5, 4},  {15, 6, 5, 4}, {11, 14, 5, 4},
     {8, 10, 13, 4}, {15, 14, 9, 4}, {11, 10, 13, 12}, {15, 14, 9, 12},
     {11, 10, 13, 8}, {15, 1, 9, 12}, {11, 14, 13, 8}, {7, 10, 9, 12},
     {4, 6, 5, 8}},
    {{3, 0, 0, 0},  {11, 2, 0, 0}, {7, 7, 3, 0},  {7, 10, 9, 5},
     {7, 6, 5, 4},  {4, 6, 5, 6},  {7, 6, 5, 8},  {15, 6, 5, 4},
     {11, 14, 13, 4}, {15, 10, 9, 4}, {11, 14, 13, 12}, {8, 10, 9, 8},
     {15, 14, 13, 12}, {11, 10, 9, 12}, {7, 11, 12, 8}, {9, 8, 10, 1},
     {7, 6, 5, 4}},
    {{15, 0, 0, 0}, {15, 14, 0, 0}, {11, 15, 13, 0}, {8, 12, 14, 12},
     {15, 10, 11, 11}, {11, 8, 9, 10}, {9, 14, 13, 9}, {8, 10, 9, 8},
     {15, 14, 13, 13}, {11, 14, 10, 12}, {15, 10, 13, 12}, {11, 14, 9, 8},
     {8, 10, 9, 12}, {15, 14, 13, 12}, {11, 10, 9, 8}, {7, 6, 5, 4},
     {3, 2, 1, 0}},
};
// chroma DC coeff_token (nC == -1), [TotalCoeff 0..4][TrailingOnes]
static const uint8_t CDC_CT_LEN[5][4] = {
    {2, 0, 0, 0}, {6, 1, 0, 0}, {6, 6, 3, 0}, {6, 7, 7, 6}, {6, 8, 8, 7}};
static const uint8_t CDC_CT_BITS[5][4] = {
    {1, 0, 0, 0}, {7, 1, 0, 0}, {4, 6, 1, 0}, {3, 3, 2, 5}, {2, 3, 2, 0}};

// total_zeros, 4x4 blocks: [TotalCoeff-1][total_zeros]
static const uint8_t TZ_LEN[15][16] = {
    {1, 3, 3, 4, 4, 5, 5, 6, 6, 7, 7, 8, 8, 9, 9, 9},
    {3, 3, 3, 3, 3, 4, 4, 4, 4, 5, 5, 6, 6, 6, 6},
    {4, 3, 3, 3, 4, 4, 3, 3, 4, 5, 5, 6, 5, 6},
    {5, 3, 4, 4, 3, 3, 3, 4, 3, 4, 5, 5, 5},
    {4, 4, 4, 3, 3, 3, 3, 3, 4, 5, 4, 5},
    {6, 5, 3, 3, 3, 3, 3, 3, 4, 3, 6},
    {6, 5, 3, 3, 3, 2, 3, 4, 3, 6},
    {6, 4, 5, 3, 2, 2, 3, 3, 6},
    {6, 6, 4, 2, 2, 3, 2, 5},
    {5, 5, 3, 2, 2, 2, 4},
    {4, 4, 3, 3, 1, 3},
    {4, 4, 2, 1, 3},
    {3, 3, 1, 2},
    {2, 2, 1},
    {1, 1},
};
static const uint8_t TZ_BITS[15][16] = {
    {1, 3, 2, 3, 2, 3, 2, 3, 2, 3, 2, 3, 2, 3, 2, 1},
    {7, 6, 5, 4, 3, 5, 4, 3, 2, 3, 2, 3, 2, 1, 0},
    {5, 7, 6, 5, 4, 3, 4, 3, 2, 3, 2, 1, 1, 0},
    {3, 7, 5, 4, 6, 5, 4, 3, 3, 2, 2, 1, 0},
    {5, 4, 3, 7, 6, 5, 4, 3, 2, 1, 1, 0},
    {1, 1, 7, 6, 5, 4, 3, 2, 1, 1, 0},
    {1, 1, 5, 4, 3, 3, 2, 1, 1, 0},
    {1, 1, 1, 3, 3, 2, 2, 1, 0},
    {1, 0, 1, 3, 2, 1, 1, 1},
    {1, 0, 1, 3, 2, 1, 1},
    {0, 1, 1, 2, 1, 3},
    {0, 1, 1, 1, 1},
    {0, 1, 1, 1},
    {0, 1, 1},
    {0, 1},
};
// total_zeros, chroma DC (2x2): [TotalCoeff-1][total_zeros]
static const uint8_t CDC_TZ_LEN[3][4] = {{1, 2, 3, 3}, {1, 2, 2, 0}, {1, 1, 0, 0}};
static const uint8_t CDC_TZ_BITS[3][4] = {{1, 1, 1, 0}, {1, 1, 0, 0}, {1, 0, 0, 0}};

// run_before: [min(zerosLeft,7)-1][run_before]
static const uint8_t RB_LEN[7][15] = {
    {1, 1},
    {1, 2, 2},
    {2, 2, 2, 2},
    {2, 2, 2, 3, 3},
    {2, 2, 3, 3, 3, 3},
    {2, 3, 3, 3, 3, 3, 3},
    {3, 3, 3, 3, 3, 3, 3, 4, 5, 6, 7, 8, 9, 10, 11},
};
static const uint8_t RB_BITS[7][15] = {
    {1, 0},
    {1, 1, 0},
    {3, 2, 1, 0},
    {3, 2, 1, 1, 0},
    {3, 2, 3, 2, 1, 0},
    {3, 0, 1, 3, 2, 5, 4},
    {7, 6, 5, 4, 3, 2, 1, 1, 1, 1, 1, 1, 1, 1, 1},
};

// ---------------------------------------------------------------------------
// Quantization tables (per qp%6; position classes a=(0,0)-like, b=(1,1)-like,
// c=rest) and zig-zag scan
// ---------------------------------------------------------------------------
static const int32_t QMF[6][3] = {{13107, 5243, 8066}, {11916, 4660, 7490},
                                  {10082, 4194, 6554}, {9362, 3647, 5825},
                                  {8192, 3355, 5243},  {7282, 2893, 4559}};
static const int32_t QV[6][3] = {{10, 16, 13}, {11, 18, 14}, {13, 20, 16},
                                 {14, 23, 18}, {16, 25, 20}, {18, 29, 23}};
static const uint8_t POSCLS[16] = {0, 2, 0, 2, 2, 1, 2, 1,
                                   0, 2, 0, 2, 2, 1, 2, 1};
static const uint8_t ZIGZAG[16] = {0, 1, 4, 8, 5, 2, 3, 6,
                                   9, 12, 13, 10, 7, 11, 14, 15};
static const uint8_t CHROMA_QP[52] = {
    0,  1,  2,  3,  4,  5,  6,  7,  8,  9,  10, 11, 12, 13, 14, 15, 16, 17,
    18, 19, 20, 21, 22, 23, 24, 25, 26, 27, 28, 29, 29, 30, 31, 32, 32, 33,
    34, 34, 35, 35, 36, 36, 37, 37, 37, 38, 38, 38, 39, 39, 39, 39};

// luma 4x4 block index -> (x4, y4) position in 4x4-block units
static inline int blk_x4(int i) { return ((i >> 2) & 1) * 2 + (i & 1); }
static inline int blk_y4(int i) { return ((i >> 2) >> 1) * 2 + ((i >> 1) & 1); }

// ---------------------------------------------------------------------------
// transforms
// ---------------------------------------------------------------------------
static void fwd4x4(const int16_t in[16], int32_t out[16]) {
  int32_t t[16];
  for (int r = 0; r < 4; ++r) {
    int32_t d0 = in[r * 4 + 0], d1 = in[r * 4 + 1], d2 = in[r * 4 + 2],
            d3 = in[r * 4 + 3];
    int32_t s0 = d0 + d3, s1 = d1 + d2, s2 = d1 - d2, s3 = d0 - d3;
    t[r * 4 + 0] = s0 + s1;
    t[r * 4 + 1] = 2 * s3 + s2;
    t[r * 4 + 2] = s0 - s1;
    t[r * 4 + 3] = s3 - 2 * s2;
  }
  for (int c = 0; c < 4; ++c) {
    int32_t d0 = t[c], d1 = t[4 + c], d2 = t[8 + c], d3 = t[12 + c];
    int32_t s0 = d0 + d3, s1 = d1 + d2, s2 = d1 - d2, s3 = d0 - d3;
    out[c] = s0 + s1;
    out[4 + c] = 2 * s3 + s2;
    out[8 + c] = s0 - s1;
    out[12 + c] = s3 - 2 * s2;
  }
}

static void inv4x4(const int32_t in[16], int32_t out[16]) {
  int32_t t[16];
  for (int r = 0; r < 4; ++r) {
    int32_t d0 = in[r * 4 + 0], d1 = in[r * 4 + 1], d2 = in[r * 4 + 2],
            d3 = in[r * 4 + 3];
    int32_t e0 = d0 + d2, e1 = d0 - d2, e2 = (d1 >> 1) - d3,
            e3 = d1 + (d3 >> 1);
    t[r * 4 + 0] = e0 + e3;
    t[r * 4 + 1] = e1 + e2;
    t[r * 4 + 2] = e1 - e2;
    t[r * 4 + 3] = e0 - e3;
  }
  for (int c = 0; c < 4; ++c) {
    int32_t d0 = t[c], d1 = t[4 + c], d2 = t[8 + c], d3 = t[12 + c];
    int32_t e0 = d0 + d2, e1 = d0 - d2, e2 = (d1 >> 1) - d3,
            e3 = d1 + (d3 >> 1);
    out[c] = e0 + e3;
    out[4 + c] = e1 + e2;
    out[8 + c] = e1 - e2;
    out[12 + c] = e0 - e3;
  }
}

static void hadamard4x4(const int32_t in[16], int32_t out[16]) {
  int32_t t[16];
  for (int r = 0; r < 4; ++r) {
    int32_t d0 = in[r * 4], d1 = in[r * 4 + 1], d2 = in[r * 4 + 2],
            d3 = in[r * 4 + 3];
    int32_t s0 = d0 + d3, s1 = d1 + d2, s2 = d1 - d2, s3 = d0 - d3;
    t[r * 4 + 0] = s0 + s1;
    t[r * 4 + 1] = s3 + s2;
    t[r * 4 + 2] = s0 - s1;
    t[r * 4 + 3] = s3 - s2;
  }
  for (int c = 0; c < 4; ++c) {
    int32_t d0 = t[c], d1 = t[4 + c], d2 = t[8 + c], d3 = t[12 + c];
    int32_t s0 = d0 + d3, s1 = d1 + d2, s2 = d1 - d2, s3 = d0 - d3;
    out[c] = s0 + s1;
    out[4 + c] = s3 + s2;
    out[8 + c] = s0 - s1;
    out[12 + c] = s3 - s2;
  }
}

// ---------------------------------------------------------------------------
// CAVLC residual block write/read
// n = maxNumCoeff (16 luma DC, 15 AC, 4 chroma DC); nC = -1 for chroma DC.
// coeffs[] are in scan order (DC first for n==16/4; AC blocks pass the 15
// AC-scan coefficients). Returns TotalCoeff.
// ---------------------------------------------------------------------------
static int cavlc_write_block(BitWriter& w, const int32_t* coeffs, int n, int nC) {
  int pos[16], val[16], tc = 0;
  for (int i = 0; i < n; ++i) {
    if (coeffs[i]) {
      pos[tc] = i;
      val[tc] = coeffs[i];
      ++tc;
    }
  }
  int t1 = 0;
  for (int i = tc - 1; i >= 0 && t1 < 3; --i) {
    if (val[i] == 1 || val[i] == -1)
      ++t1;
    else
      break;
  }
  // coeff_token
  if (nC == -1) {
    w.put(CDC_CT_BITS[tc][t1], CDC_CT_LEN[tc][t1]);
  } else if (nC >= 8) {
    w.put(tc == 0 ? 3 : (uint32_t)(((tc - 1) << 2) | t1), 6);
  } else {
    int b = nC < 2 ? 0 : nC < 4 ? 1 : 2;
    w.put(CT_BITS[b][tc][t1], CT_LEN[b][tc][t1]);
  }
  if (tc == 0) return 0;
  // trailing one signs, highest frequency first
  for (int i = tc - 1; i >= tc - t1; --i) w.put(val[i] < 0 ? 1 : 0, 1);
  // remaining levels
  int suffix_len = (tc > 10 && t1 < 3) ? 1 : 0;
  for (int i = tc - 1 - t1; i >= 0; --i) {
    int level = val[i];
    int code = level > 0 ? 2 * level - 2 : -2 * level - 1;
    if (i == tc - 1 - t1 && t1 < 3) code -= 2;
    if (suffix_len == 0) {
      if (code < 14) {
        w.put(1, code + 1);  // code zeros then a 1
      } else if (code < 30) {
        w.put(1, 15);  // level_prefix 14
        w.put(code - 14, 4);
      } else {
        w.put(1, 16);  // level_prefix 15
        w.put(code - 30, 12);
      }
    } else {
      if ((code >> suffix_len) < 15) {
        w.put(1, (code >> suffix_len) + 1);
        w.put(code & ((1 << suffix_len) - 1), suffix_len);
      } else {
        w.put(1, 16);  // level_prefix 15
        w.put(code - (15 << suffix_len), 12);
      }
    }
    if (suffix_len == 0) suffix_len = 1;
    if (std::abs(level) > (3 << (suffix_len - 1)) && suffix_len < 6)
      ++suffix_len;
  }
  // total_zeros
  int total_zeros = pos[tc - 1] + 1 - tc;
  if (tc < n) {
    if (nC == -1)
      w.put(CDC_TZ_BITS[tc - 1][total_zeros], CDC_TZ_LEN[tc - 1][total_zeros]);
    else
      w.put(TZ_BITS[tc - 1][total_zeros], TZ_LEN[tc - 1][total_zeros]);
  }
  // run_before, highest frequency first (last run implicit)
  int zeros_left = total_zeros;
  for (int i = tc - 1; i >= 1 && zeros_left > 0; --i) {
    int run = pos[i] - pos[i - 1] - 1;
    int zl = zeros_left < 7 ? zeros_left : 7;
    w.put(RB_BITS[zl - 1][run], RB_LEN[zl - 1][run]);
    zeros_left -= run;
  }
  return tc;
}

// generic VLC read helper: match (len,bits) rows
template <typename LenT, typename BitsT>
static int vlc_read(BitReader& r, const LenT* lens, const BitsT* bits, int count) {
  uint32_t acc = 0;
  int len = 0;
  while (len < 17) {
    acc = (acc << 1) | r.u(1);
    ++len;
    if (r.overrun) return -1;
    for (int i = 0; i < count; ++i)
      if (lens[i] == len && bits[i] == acc) return i;
  }
  return -1;
}

static int cavlc_read_block(BitReader& r, int32_t* coeffs, int n, int nC) {
  memset(coeffs, 0, sizeof(int32_t) * n);
  int tc = 0, t1 = 0;
  if (nC == -1) {
    uint8_t lens[20], bits[20];
    int k = 0;
    for (int c = 0; c <= 4; ++c)
      for (int t = 0; t < 4; ++t) {
        lens[k] = CDC_CT_LEN[c][t];
        bits[k] = CDC_CT_BITS[c][t];
        ++k;
      }
    int idx = vlc_read(r, lens, bits, k);
    if (idx < 0) return -1;
    tc = idx / 4;
    t1 = idx % 4;
  } else if (nC >= 8) {
    uint32_t v = r.u(6);
    if (v == 3) {
      tc = 0;
      t1 = 0;
    } else {
      tc = (int)(v >> 2) + 1;
      t1 = (int)(v & 3);
    }
  } else {
    int b = nC < 2 ? 0 : nC < 4 ? 1 : 2;
    uint8_t lens[17 * 4];
    uint16_t bits[17 * 4];
    int k = 0;
    for (int c = 0; c <= 16; ++c)
      for (int t = 0; t < 4; ++t) {
        lens[k] = CT_LEN[b][c][t];
        bits[k] = CT_BITS[b][c][t];
        ++k;
      }
    int idx = vlc_read(r, lens, bits, k);
    if (idx < 0) return -1;
    tc = idx / 4;
    t1 = idx % 4;
  }
  if (tc == 0) return 0;
  if (tc > n || t1 > tc) return -1;
  int val[16];
  for (int i = 0; i < t1; ++i) val[i] = r.u(1) ? -1 : 1;  // highest freq first
  int suffix_len = (tc > 10 && t1 < 3) ? 1 : 0;
  for (int i = t1; i < tc; ++i) {
    int prefix = 0;
    while (r.u(1) == 0) {
      if (++prefix > 32 || r.overrun) return -1;
    }
    int code;
    int suffix_size;
    if (suffix_len == 0)
      suffix_size = prefix == 14 ? 4 : prefix >= 15 ? 12 : 0;
    else
      suffix_size = prefix >= 15 ? 12 : suffix_len;
    code = (prefix < 15 ? prefix : 15) << suffix_len;
    if (suffix_size) code += r.u(suffix_size);
    if (prefix >= 15 && suffix_len == 0) code += 15;
    if (prefix >= 16) code += (1 << (prefix - 3)) - 4096;
    if (i == t1 && t1 < 3) code += 2;
    val[i] = (code % 2 == 0) ? (code + 2) >> 1 : -((code + 1) >> 1);
    if (suffix_len == 0) suffix_len = 1;
    if (std::abs(val[i]) > (3 << (suffix_len - 1)) && suffix_len < 6)
      ++suffix_len;
  }
  int total_zeros = 0;
  if (tc < n) {
    int idx;
    if (nC == -1)
      idx = vlc_read(r, CDC_TZ_LEN[tc - 1], CDC_TZ_BITS[tc - 1], 4 - tc + 1);
    else
      idx = vlc_read(r, TZ_LEN[tc - 1], TZ_BITS[tc - 1], n - tc + 1);
    if (idx < 0) return -1;
    total_zeros = idx;
  }
  // place coefficients: val[] is highest-frequency first
  int zeros_left = total_zeros;
  int pos = tc - 1 + total_zeros;  // scan index of the highest-freq coeff
  for (int i = 0; i < tc; ++i) {
    if (pos >= n || pos < 0) return -1;
    coeffs[pos] = val[i];
    if (i == tc - 1) break;
    int run = 0;
    if (zeros_left > 0) {
      int zl = zeros_left < 7 ? zeros_left : 7;
      int idx = vlc_read(r, RB_LEN[zl - 1], RB_BITS[zl - 1], zl == 7 ? 15 : zl + 1);
      if (idx < 0) return -1;
      run = idx;
      zeros_left -= run;
    }
    pos -= 1 + run;
  }
  if (r.overrun) return -1;
  return tc;
}

// ---------------------------------------------------------------------------
// color conversion (BT.601 limited range)
// ---------------------------------------------------------------------------
static void rgb_to_yuv420_rows(const uint8_t* rgb, int w, int h, int pw,
                               int y0, int y1, uint8_t* Y, uint8_t* Cb,
                               uint8_t* Cr) {
  // write padded planes (pw wide), edge-replicated, luma rows [y0, y1).
  // The interior loop is branch-free (no edge clamps) so the compiler can
  // vectorize it; edge replication runs in separate tail loops.
  for (int y = y0; y < y1; ++y) {
    const int sy = y < h ? y : h - 1;
    const uint8_t* row = rgb + (size_t)sy * w * 3;
    uint8_t* yo = &Y[(size_t)y * pw];
    for (int x = 0; x < w; ++x) {
      const uint8_t* p = row + x * 3;
      yo[x] = clip8(((66 * p[0] + 129 * p[1] + 25 * p[2] + 128) >> 8) + 16);
    }
    for (int x = w; x < pw; ++x) yo[x] = yo[w - 1];
  }
  const int cw = pw / 2;
  const int cwi = w / 2;  // full 2x2 blocks (w, h are even for 16|dims)
  for (int y = y0 / 2; y < y1 / 2; ++y) {
    uint8_t* cbo = &Cb[(size_t)y * cw];
    uint8_t* cro = &Cr[(size_t)y * cw];
    const int sy0 = std::min(2 * y, h - 1), sy1 = std::min(2 * y + 1, h - 1);
    const uint8_t* r0 = rgb + (size_t)sy0 * w * 3;
    const uint8_t* r1 = rgb + (size_t)sy1 * w * 3;
    for (int x = 0; x < cwi; ++x) {
      const uint8_t* a = r0 + 2 * x * 3;
      const uint8_t* b = r1 + 2 * x * 3;
      const int R = (a[0] + a[3] + b[0] + b[3] + 2) >> 2;
      const int G = (a[1] + a[4] + b[1] + b[4] + 2) >> 2;
      const int B = (a[2] + a[5] + b[2] + b[5] + 2) >> 2;
      cbo[x] = clip8(((-38 * R - 74 * G + 112 * B + 128) >> 8) + 128);
      cro[x] = clip8(((112 * R - 94 * G - 18 * B + 128) >> 8) + 128);
    }
    for (int x = cwi; x < cw; ++x) {
      int R = 0, G = 0, B = 0;
      for (int dy = 0; dy < 2; ++dy)
        for (int dx = 0; dx < 2; ++dx) {
          int sy = std::min(2 * y + dy, h - 1), sx = std::min(2 * x + dx, w - 1);
          const uint8_t* p = rgb + ((size_t)sy * w + sx) * 3;
          R += p[0];
          G += p[1];
          B += p[2];
        }
      R = (R + 2) >> 2;
      G = (G + 2) >> 2;
      B = (B + 2) >> 2;
      cbo[x] = clip8(((-38 * R - 74 * G + 112 * B + 128) >> 8) + 128);
      cro[x] = clip8(((112 * R - 94 * G - 18 * B + 128) >> 8) + 128);
    }
  }
}

static void rgb_to_yuv420(const uint8_t* rgb, int w, int h, int pw, int ph,
                          uint8_t* Y, uint8_t* Cb, uint8_t* Cr) {
  rgb_to_yuv420_rows(rgb, w, h, pw, 0, ph, Y, Cb, Cr);
}

static void yuv420_to_rgb(const uint8_t* Y, const uint8_t* Cb, const uint8_t* Cr,
                          int pw, int w, int h, uint8_t* rgb) {
  int cw = pw / 2;
  for (int y = 0; y < h; ++y) {
    for (int x = 0; x < w; ++x) {
      int C = (int)Y[y * pw + x] - 16;
      int D = (int)Cb[(y / 2) * cw + x / 2] - 128;
      int E = (int)Cr[(y / 2) * cw + x / 2] - 128;
      uint8_t* p = rgb + (y * w + x) * 3;
      p[0] = clip8((298 * C + 409 * E + 128) >> 8);
      p[1] = clip8((298 * C - 100 * D - 208 * E + 128) >> 8);
      p[2] = clip8((298 * C + 516 * D + 128) >> 8);
    }
  }
}

// ---------------------------------------------------------------------------
// shared intra prediction + reconstruction (used by encoder and decoder so
// both sides hold bit-identical reference pixels)
// ---------------------------------------------------------------------------

// ---------------------------------------------------------------------------
// Persistent worker pool. Spawning + joining the per-frame slice/band
// threads measured 0.74 ms PER FRAME for 16 threads (≈30% of a typical
// 512² encode); the pool keeps workers parked on a condition variable and
// dispatches index tasks, reused by colour conversion, encode slices and
// decode slices. Work partitioning is unchanged, so bitstreams stay
// byte-identical.
// ---------------------------------------------------------------------------
class WorkPool {
 public:
  explicit WorkPool(int n) {
    for (int i = 0; i < n; ++i) ws_.emplace_back([this] { loop(); });
  }
  ~WorkPool() {
    {
      std::lock_guard<std::mutex> lk(m_);
      stop_ = true;
    }
    go_.notify_all();
    for (auto& t : ws_) t.join();
  }
  int size() const { return (int)ws_.size(); }
  // run fn(0..n-1) across the workers; blocks until every task finished
  void run(int n, const std::function<void(int)>& fn) {
    if (n <= 1 || ws_.empty()) {
      for (int i = 0; i < n; ++i) fn(i);
      return;
    }
    std::unique_lock<std::mutex> lk(m_);
    fn_ = &fn;
    ntasks_ = n;
    next_ = 0;
    done_ = 0;
    ++gen_;
    go_.notify_all();
    fin_.wait(lk, [&] { return done_ == ntasks_; });
    fn_ = nullptr;
  }

 private:
  void loop() {
    uint64_t seen = 0;
    std::unique_lock<std::mutex> lk(m_);
    for (;;) {
      go_.wait(lk, [&] { return stop_ || gen_ != seen; });
      if (stop_) return;
      seen = gen_;
      while (next_ < ntasks_) {
        const int i = next_++;
        const std::function<void(int)>* f = fn_;
        lk.unlock();
        (*f)(i);
        lk.lock();
        if (++done_ == ntasks_) fin_.notify_all();
      }
    }
  }
  std::vector<std::thread> ws_;
  std::mutex m_;
  std::condition_variable go_, fin_;
  const std::function<void(int)>* fn_ = nullptr;
  int ntasks_ = 0, next_ = 0, done_ = 0;
  uint64_t gen_ = 0;
  bool stop_ = false;
};

struct PlaneCtx {
  uint8_t* data;  // reconstructed plane, stride = width
  int stride;
};

// 16x16 luma prediction, modes 0=V 1=H 2=DC 3=Plane
static void pred_luma16(const PlaneCtx& pl, int mbx, int mby, int mode,
                        bool have_top, bool have_left, uint8_t pred[256]) {
  const int x0 = mbx * 16, y0 = mby * 16;
  const uint8_t* top = have_top ? pl.data + (y0 - 1) * pl.stride + x0 : nullptr;
  const uint8_t* leftc = have_left ? pl.data + y0 * pl.stride + (x0 - 1) : nullptr;
  switch (mode) {
    case 0:  // vertical
      for (int y = 0; y < 16; ++y)
        for (int x = 0; x < 16; ++x) pred[y * 16 + x] = top[x];
      break;
    case 1:  // horizontal
      for (int y = 0; y < 16; ++y) {
        uint8_t v = leftc[y * pl.stride];
        for (int x = 0; x < 16; ++x) pred[y * 16 + x] = v;
      }
      break;
    case 2: {  // DC
      int sum = 0, cnt = 0;
      if (top) {
        for (int x = 0; x < 16; ++x) sum += top[x];
        cnt += 16;
      }
      if (leftc) {
        for (int y = 0; y < 16; ++y) sum += leftc[y * pl.stride];
        cnt += 16;
      }
      int dc = cnt == 32 ? (sum + 16) >> 5 : cnt == 16 ? (sum + 8) >> 4 : 128;
      memset(pred, dc, 256);
      break;
    }
    case 3: {  // plane
      int H = 0, V = 0;
      const uint8_t* tl = pl.data + (y0 - 1) * pl.stride + (x0 - 1);
      for (int i = 0; i < 8; ++i) {
        H += (i + 1) * ((int)top[8 + i] - (int)(i == 7 ? tl[0] : top[6 - i]));
        V += (i + 1) * ((int)leftc[(8 + i) * pl.stride] -
                        (int)(i == 7 ? tl[0] : leftc[(6 - i) * pl.stride]));
      }
      int a = 16 * ((int)leftc[15 * pl.stride] + (int)top[15]);
      int b = (5 * H + 32) >> 6;
      int c = (5 * V + 32) >> 6;
      for (int y = 0; y < 16; ++y)
        for (int x = 0; x < 16; ++x)
          pred[y * 16 + x] = clip8((a + b * (x - 7) + c * (y - 7) + 16) >> 5);
      break;
    }
  }
}

// 8x8 chroma prediction, modes 0=DC 1=H 2=V 3=Plane
static void pred_chroma8(const PlaneCtx& pl, int mbx, int mby, int mode,
                         bool have_top, bool have_left, uint8_t pred[64]) {
  const int x0 = mbx * 8, y0 = mby * 8;
  const uint8_t* top = have_top ? pl.data + (y0 - 1) * pl.stride + x0 : nullptr;
  const uint8_t* leftc = have_left ? pl.data + y0 * pl.stride + (x0 - 1) : nullptr;
  switch (mode) {
    case 0: {  // DC per 4x4 sub-block (x264 dc0..dc3 structure)
      int st[2] = {0, 0}, sl[2] = {0, 0};
      if (top)
        for (int x = 0; x < 8; ++x) st[x >> 2] += top[x];
      if (leftc)
        for (int y = 0; y < 8; ++y) sl[y >> 2] += leftc[y * pl.stride];
      int dc[4];
      auto mix = [&](int t, int l) {
        if (top && leftc) return (st[t] + sl[l] + 4) >> 3;
        if (top) return (st[t] + 2) >> 2;
        if (leftc) return (sl[l] + 2) >> 2;
        return 128;
      };
      dc[0] = mix(0, 0);
      dc[1] = top ? (st[1] + 2) >> 2 : leftc ? (sl[0] + 2) >> 2 : 128;
      dc[2] = leftc ? (sl[1] + 2) >> 2 : top ? (st[0] + 2) >> 2 : 128;
      dc[3] = mix(1, 1);
      for (int y = 0; y < 8; ++y)
        for (int x = 0; x < 8; ++x)
          pred[y * 8 + x] = (uint8_t)dc[(y >> 2) * 2 + (x >> 2)];
      break;
    }
    case 1:  // horizontal
      for (int y = 0; y < 8; ++y) {
        uint8_t v = leftc[y * pl.stride];
        for (int x = 0; x < 8; ++x) pred[y * 8 + x] = v;
      }
      break;
    case 2:  // vertical
      for (int y = 0; y < 8; ++y)
        for (int x = 0; x < 8; ++x) pred[y * 8 + x] = top[x];
      break;
    case 3: {  // plane
      int H = 0, V = 0;
      const uint8_t* tl = pl.data + (y0 - 1) * pl.stride + (x0 - 1);
      for (int i = 0; i < 4; ++i) {
        H += (i + 1) * ((int)top[4 + i] - (int)(i == 3 ? tl[0] : top[2 - i]));
        V += (i + 1) * ((int)leftc[(4 + i) * pl.stride] -
                        (int)(i == 3 ? tl[0] : leftc[(2 - i) * pl.stride]));
      }
      int a = 16 * ((int)leftc[7 * pl.stride] + (int)top[7]);
      int b = (17 * H + 16) >> 5;
      int c = (17 * V + 16) >> 5;
      for (int y = 0; y < 8; ++y)
        for (int x = 0; x < 8; ++x)
          pred[y * 8 + x] = clip8((a + b * (x - 3) + c * (y - 3) + 16) >> 5);
      break;
    }
  }
}

// ---------------------------------------------------------------------------
// 4x4 intra luma prediction (all 9 modes) for I_4x4 macroblocks.
// p_top: 8 samples above (top-right replicated per 8.3.1.2 when absent),
// p_left: 4 samples left, p_tl: corner. Availability flags gate modes.
// ---------------------------------------------------------------------------
static void pred_luma4(int mode, const uint8_t* t /*8*/, const uint8_t* l /*4*/,
                       uint8_t tl, bool have_t, bool have_l, bool have_tl,
                       uint8_t pred[16]) {
  auto P = [&](int x, int y) -> uint8_t& { return pred[y * 4 + x]; };
  switch (mode) {
    case 0:  // vertical
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x) P(x, y) = t[x];
      break;
    case 1:  // horizontal
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x) P(x, y) = l[y];
      break;
    case 2: {  // DC
      int sum = 0, cnt = 0;
      if (have_t) { sum += t[0] + t[1] + t[2] + t[3]; cnt += 4; }
      if (have_l) { sum += l[0] + l[1] + l[2] + l[3]; cnt += 4; }
      int dc = cnt == 8 ? (sum + 4) >> 3 : cnt == 4 ? (sum + 2) >> 2 : 128;
      for (int i = 0; i < 16; ++i) pred[i] = (uint8_t)dc;
      break;
    }
    case 3:  // diagonal down-left
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x) {
          int i = x + y;
          P(x, y) = (i == 6) ? (uint8_t)((t[6] + 3 * t[7] + 2) >> 2)
                             : (uint8_t)((t[i] + 2 * t[i + 1] + t[i + 2] + 2) >> 2);
        }
      break;
    case 4: {  // diagonal down-right (spec 8.3.1.2.5)
      auto T4 = [&](int i) { return i < 0 ? tl : t[i]; };
      auto L4 = [&](int i) { return i < 0 ? tl : l[i]; };
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x) {
          if (x > y)
            P(x, y) = (uint8_t)((T4(x - y - 2) + 2 * T4(x - y - 1) + T4(x - y) + 2) >> 2);
          else if (x < y)
            P(x, y) = (uint8_t)((L4(y - x - 2) + 2 * L4(y - x - 1) + L4(y - x) + 2) >> 2);
          else
            P(x, y) = (uint8_t)((t[0] + 2 * tl + l[0] + 2) >> 2);
        }
      break;
    }
    case 5: {  // vertical-right (spec 8.3.1.2.6)
      auto T4 = [&](int i) { return i < 0 ? tl : t[i]; };
      auto L4 = [&](int i) { return i < 0 ? tl : l[i]; };
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x) {
          const int z = 2 * x - y;
          if (z >= 0 && (z & 1) == 0) {
            const int i = x - (y >> 1);
            P(x, y) = (uint8_t)((T4(i - 1) + T4(i) + 1) >> 1);
          } else if (z >= 0) {
            const int i = x - (y >> 1);
            P(x, y) = (uint8_t)((T4(i - 2) + 2 * T4(i - 1) + T4(i) + 2) >> 2);
          } else if (z == -1) {
            P(x, y) = (uint8_t)((l[0] + 2 * tl + t[0] + 2) >> 2);
          } else {  // zVR < -1: pure left-column taps
            const int base = y - 2 * x;  // 2 or 3
            P(x, y) = (uint8_t)((L4(base - 1) + 2 * L4(base - 2) + L4(base - 3) + 2) >> 2);
          }
        }
      break;
    }
    case 6:  // horizontal-down (mirror of VR)
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x) {
          int z = 2 * y - x;
          auto L = [&](int i2) { return i2 < 0 ? tl : l[i2]; };
          auto T = [&](int i2) { return i2 < 0 ? tl : t[i2]; };
          if (z >= 0 && (z & 1) == 0) {
            int i = y - (x >> 1);
            P(x, y) = (uint8_t)((L(i - 1) + L(i) + 1) >> 1);
          } else if (z >= 0) {
            int i = y - (x >> 1);
            P(x, y) = (uint8_t)((L(i - 2) + 2 * L(i - 1) + L(i) + 2) >> 2);
          } else if (z == -1) {
            P(x, y) = (uint8_t)((l[0] + 2 * tl + t[0] + 2) >> 2);
          } else {
            int i = x - 2 * y;
            P(x, y) = (uint8_t)((T(i - 1) + 2 * T(i - 2) + T(i - 3) + 2) >> 2);
          }
        }
      break;
    case 7:  // vertical-left
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x) {
          int i = x + (y >> 1);
          if ((y & 1) == 0)
            P(x, y) = (uint8_t)((t[i] + t[i + 1] + 1) >> 1);
          else
            P(x, y) = (uint8_t)((t[i] + 2 * t[i + 1] + t[i + 2] + 2) >> 2);
        }
      break;
    case 8:  // horizontal-up
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x) {
          int z = x + 2 * y;
          if (z > 5)
            P(x, y) = l[3];
          else if (z == 5)
            P(x, y) = (uint8_t)((l[2] + 3 * l[3] + 2) >> 2);
          else if (z & 1) {
            int i = y + (x >> 1);
            P(x, y) = (uint8_t)((l[i] + 2 * l[i + 1] + l[i + 2 > 3 ? 3 : i + 2] + 2) >> 2);
          } else {
            int i = y + (x >> 1);
            P(x, y) = (uint8_t)((l[i] + l[i + 1] + 1) >> 1);
          }
        }
      break;
  }
}

// coded_block_pattern me(v) mapping for Intra_4x4 (spec table 9-4)
static const uint8_t CBP_INTRA[48] = {
    47, 31, 15, 0,  23, 27, 29, 30, 7,  11, 13, 14, 39, 43, 45, 46,
    16, 3,  5,  10, 12, 19, 21, 26, 28, 35, 37, 42, 44, 1,  2,  4,
    8,  17, 18, 20, 24, 6,  9,  22, 25, 32, 33, 34, 36, 40, 38, 41};
static int cbp_intra_code(int cbp) {  // encoder: cbp -> codeNum
  for (int i = 0; i < 48; ++i)
    if (CBP_INTRA[i] == cbp) return i;
  return 0;
}

// z-scan index of a 4x4 position (inverse of blk_x4/blk_y4)
static inline int blk_z(int x4, int y4) {
  return ((y4 >> 1) * 2 + (x4 >> 1)) * 4 + (y4 & 1) * 2 + (x4 & 1);
}

// Dequant + inverse transform + add prediction for one I_16x16 luma MB.
// dc_q: 16 quantized DC levels in RASTER order; ac_q[blk][16] quantized
// levels in RASTER order with [0] unused.
static void recon_luma16(PlaneCtx& pl, int mbx, int mby, const uint8_t pred[256],
                         const int32_t dc_q[16], const int32_t ac_q[16][16],
                         int qp) {
  const int qm = qp % 6, qs = qp / 6;
  // luma DC: inverse Hadamard, then scale (spec 8.5.10; LevelScale carries
  // the flat weightScale factor of 16)
  int32_t dct[16], dcd[16];
  hadamard4x4(dc_q, dct);
  const int32_t ls00 = QV[qm][0] * 16;
  for (int i = 0; i < 16; ++i) {
    if (qp >= 36)
      dcd[i] = (dct[i] * ls00) * (1 << (qs - 6));
    else
      dcd[i] = (dct[i] * ls00 + (1 << (5 - qs))) >> (6 - qs);
  }
  for (int b = 0; b < 16; ++b) {
    const int x4 = blk_x4(b), y4 = blk_y4(b);
    int32_t d[16];
    d[0] = dcd[y4 * 4 + x4];
    for (int i = 1; i < 16; ++i) d[i] = (ac_q[b][i] * QV[qm][POSCLS[i]]) * (1 << qs);
    int32_t r[16];
    inv4x4(d, r);
    uint8_t* dst = pl.data + (mby * 16 + y4 * 4) * pl.stride + mbx * 16 + x4 * 4;
    const uint8_t* pr = pred + y4 * 4 * 16 + x4 * 4;
    for (int y = 0; y < 4; ++y)
      for (int x = 0; x < 4; ++x)
        dst[y * pl.stride + x] = clip8(pr[y * 16 + x] + ((r[y * 4 + x] + 32) >> 6));
  }
}

// same for one 8x8 chroma component
static void recon_chroma8(PlaneCtx& pl, int mbx, int mby, const uint8_t pred[64],
                          const int32_t dc_q[4], const int32_t ac_q[4][16],
                          int qpc) {
  const int qm = qpc % 6, qs = qpc / 6;
  // 2x2 DC Hadamard inverse (same matrix as forward), scale (spec 8.5.11)
  int32_t a = dc_q[0], b = dc_q[1], c = dc_q[2], d = dc_q[3];
  int32_t t[4] = {a + b + c + d, a - b + c - d, a + b - c - d, a - b - c + d};
  int32_t dcd[4];
  // spec 8.5.11: dcC = ((f * LevelScale(0,0)) << qP/6) >> 5, LevelScale = 16*V
  for (int i = 0; i < 4; ++i) dcd[i] = (t[i] * QV[qm][0] * 16 * (1 << qs)) >> 5;
  for (int blk = 0; blk < 4; ++blk) {
    const int x4 = blk & 1, y4 = blk >> 1;
    int32_t dq[16];
    dq[0] = dcd[blk];
    for (int i = 1; i < 16; ++i) dq[i] = (ac_q[blk][i] * QV[qm][POSCLS[i]]) * (1 << qs);
    int32_t r[16];
    inv4x4(dq, r);
    uint8_t* dst = pl.data + (mby * 8 + y4 * 4) * pl.stride + mbx * 8 + x4 * 4;
    const uint8_t* pr = pred + y4 * 4 * 8 + x4 * 4;
    for (int y = 0; y < 4; ++y)
      for (int x = 0; x < 4; ++x)
        dst[y * pl.stride + x] = clip8(pr[y * 8 + x] + ((r[y * 4 + x] + 32) >> 6));
  }
}

// ---------------------------------------------------------------------------
// nC derivation
// ---------------------------------------------------------------------------
struct NnzCtx {
  std::vector<uint8_t> luma;    // [mb][16]
  std::vector<uint8_t> chroma;  // [comp][mb][4]
  int mbw = 0, mbh = 0;

  void reset(int w, int h) {
    mbw = w;
    mbh = h;
    luma.assign((size_t)w * h * 16, 0);
    chroma.assign((size_t)2 * w * h * 4, 0);
  }
  uint8_t& lnz(int mbx, int mby, int x4, int y4) {
    return luma[((size_t)mby * mbw + mbx) * 16 + y4 * 4 + x4];
  }
  uint8_t& cnz(int comp, int mbx, int mby, int x2, int y2) {
    return chroma[(((size_t)comp * mbh + mby) * mbw + mbx) * 4 + y2 * 2 + x2];
  }
  // slice_start: first MB address of the current slice — neighbours in a
  // DIFFERENT slice are unavailable (H.264 availability rule), which is
  // what makes slices independently decodable (and thread-parallel)
  bool left_ok(int mbx, int mby, int slice_start) const {
    return mbx > 0 && (mby * mbw + mbx - 1) >= slice_start;
  }
  bool top_ok(int mbx, int mby, int slice_start) const {
    return mby > 0 && ((mby - 1) * mbw + mbx) >= slice_start;
  }
  int luma_nc(int mbx, int mby, int x4, int y4, int slice_start) {
    int na = -1, nb = -1;
    if (x4 > 0)
      na = lnz(mbx, mby, x4 - 1, y4);
    else if (left_ok(mbx, mby, slice_start))
      na = lnz(mbx - 1, mby, 3, y4);
    if (y4 > 0)
      nb = lnz(mbx, mby, x4, y4 - 1);
    else if (top_ok(mbx, mby, slice_start))
      nb = lnz(mbx, mby - 1, x4, 3);
    if (na >= 0 && nb >= 0) return (na + nb + 1) >> 1;
    if (na >= 0) return na;
    if (nb >= 0) return nb;
    return 0;
  }
  int chroma_nc(int comp, int mbx, int mby, int x2, int y2, int slice_start) {
    int na = -1, nb = -1;
    if (x2 > 0)
      na = cnz(comp, mbx, mby, x2 - 1, y2);
    else if (left_ok(mbx, mby, slice_start))
      na = cnz(comp, mbx - 1, mby, 1, y2);
    if (y2 > 0)
      nb = cnz(comp, mbx, mby, x2, y2 - 1);
    else if (top_ok(mbx, mby, slice_start))
      nb = cnz(comp, mbx, mby - 1, x2, 1);
    if (na >= 0 && nb >= 0) return (na + nb + 1) >> 1;
    if (na >= 0) return na;
    if (nb >= 0) return nb;
    return 0;
  }
};

// ---------------------------------------------------------------------------
// Encoder
// ---------------------------------------------------------------------------

// Encoder freedom (big RD win on detailed content): instead of falling
// back to I_PCM when a quantized block exceeds the implemented CAVLC
// range (TotalCoeff > 9), ZERO the smallest levels (ties: highest
// frequency first) until 9 remain. The bitstream stays fully standard;
// reconstruction uses the same capped coefficients so encoder/decoder
// stay consistent. Returns the largest |level| dropped (callers keep the
// I_PCM fallback for blocks where the cap would destroy real structure).
// a capped-away level above this means the block had strong structure
// beyond 9 coefficients — I_PCM preserves it exactly (384 B) instead of
// mushing it. Tail levels (1) are plain noise and always safe to drop.
static constexpr int PCM_DROP_LIMIT = 2;

static int cap_coeffs9(int32_t* q, int lo, int hi) {
  int idx[16], n = 0;
  for (int i = lo; i < hi; ++i)
    if (q[i]) idx[n++] = i;
  int dropped = 0;
  while (n > 9) {
    int bi = 0;
    for (int j = 1; j < n; ++j) {
      const int a = std::abs(q[idx[j]]), b = std::abs(q[idx[bi]]);
      if (a < b || (a == b && idx[j] > idx[bi])) bi = j;
    }
    dropped = std::max(dropped, std::abs(q[idx[bi]]));
    q[idx[bi]] = 0;
    idx[bi] = idx[--n];
  }
  return dropped;
}

struct Encoder {
  int w, h, pw, ph, mbw, mbh;
  std::vector<uint8_t> Y, Cb, Cr;        // source (padded)
  std::vector<uint8_t> rY, rCb, rCr;     // reconstruction
  NnzCtx nnz;
  std::vector<int8_t> enc_i4modes;       // per-4x4 modes (-1: MB not I_4x4)
  uint32_t idr_id = 0;
  uint32_t frame_num = 0;                // mod 16 (log2_max_frame_num = 4)
  bool have_ref = false;                 // a decoded-reference frame exists
  // P_Skip decision: a macroblock whose source-vs-reference SAD is below
  // the threshold is "unchanged". The floor is the QP's own quantisation
  // error (at QP q the reconstruction already differs from the source by
  // roughly qstep/3 per sample), so the threshold scales with QP;
  // <0 = auto, 0 = bit-exact-only skips.
  int skip_sad_thresh = -1;
  int skip_thresh_for(int qp) const {
    return skip_sad_thresh >= 0 ? skip_sad_thresh : 96 * (qp - 8);
  }

  Encoder(int width, int height) : w(width), h(height) {
    pw = (w + 15) & ~15;
    ph = (h + 15) & ~15;
    mbw = pw / 16;
    mbh = ph / 16;
    Y.resize((size_t)pw * ph);
    Cb.resize((size_t)pw * ph / 4);
    Cr.resize((size_t)pw * ph / 4);
    rY.resize((size_t)pw * ph);
    rCb.resize((size_t)pw * ph / 4);
    rCr.resize((size_t)pw * ph / 4);
  }

  // quantize one 4x4 transformed block (excluding index 0 when skip_dc)
  void quant_block(const int32_t W[16], int qp, bool skip_dc, int32_t out[16]) {
    const int qm = qp % 6, qbits = 15 + qp / 6;
    const int32_t f = (1 << qbits) / 3;  // intra rounding
    for (int i = skip_dc ? 1 : 0; i < 16; ++i) {
      int32_t v = W[i];
      int32_t q = (int32_t)(((int64_t)std::abs(v) * QMF[qm][POSCLS[i]] + f) >> qbits);
      if (q > 2063) q = 2063;  // keep levelCode within the 12-bit escape
      out[i] = v < 0 ? -q : q;
    }
    if (skip_dc) out[0] = 0;
  }

  int mb_sad(const uint8_t* src, int sstride, const uint8_t* pred, int pstride,
             int size) {
    int s = 0;
    for (int y = 0; y < size; ++y)
      for (int x = 0; x < size; ++x)
        s += std::abs((int)src[y * sstride + x] - (int)pred[y * pstride + x]);
    return s;
  }

  void write_pcm(BitWriter& wtr, int mbx, int mby, bool in_p_slice = false) {
    wtr.ue(in_p_slice ? 30 : 25);  // I_PCM (+5 in P slices)
    wtr.align_byte();
    // samples are byte-aligned after pcm_alignment_zero_bit: bulk-copy
    // rows into the bitstream and the reconstruction plane (was 60% of
    // worst-case encode as bit-at-a-time puts)
    const uint8_t* sy = Y.data() + (mby * 16) * pw + mbx * 16;
    uint8_t* ry = rY.data() + (mby * 16) * pw + mbx * 16;
    for (int y = 0; y < 16; ++y) {
      wtr.put_aligned_bytes(&sy[y * pw], 16);
      memcpy(&ry[y * pw], &sy[y * pw], 16);
    }
    const uint8_t* sc[2] = {Cb.data() + (mby * 8) * (pw / 2) + mbx * 8,
                            Cr.data() + (mby * 8) * (pw / 2) + mbx * 8};
    uint8_t* rc[2] = {rCb.data() + (mby * 8) * (pw / 2) + mbx * 8,
                      rCr.data() + (mby * 8) * (pw / 2) + mbx * 8};
    for (int comp = 0; comp < 2; ++comp)
      for (int y = 0; y < 8; ++y) {
        wtr.put_aligned_bytes(&sc[comp][y * (pw / 2)], 8);
        memcpy(&rc[comp][y * (pw / 2)], &sc[comp][y * (pw / 2)], 8);
      }
    for (int b = 0; b < 16; ++b)
      nnz.lnz(mbx, mby, blk_x4(b), blk_y4(b)) = 16;
    for (int comp = 0; comp < 2; ++comp)
      for (int blk = 0; blk < 4; ++blk)
        nnz.cnz(comp, mbx, mby, blk & 1, blk >> 1) = 16;
  }

  // I_4x4 macroblock encode with per-block mode selection over all 9
  // prediction modes (SAD decision on the reconstructed neighbourhood,
  // greedy in z order — the standard intra search shape). Also the
  // in-repo bitstream source exercising the decoder's I_4x4 path.
  void encode_mb_i4x4(BitWriter& wtr, int mbx, int mby, int qp,
                      int slice_start, bool in_p_slice = false) {
    const bool mb_top = nnz.top_ok(mbx, mby, slice_start);
    const bool mb_left = nnz.left_ok(mbx, mby, slice_start);
    const int qm = qp % 6, qbits = 15 + qp / 6, qs = qp / 6;
    const int32_t fr = (1 << qbits) / 3;
    uint8_t* base = rY.data();
    const uint8_t* srcy = Y.data();
    const int stride = pw;
    int8_t* my_modes = &enc_i4modes[((size_t)mby * mbw + mbx) * 16];

    int32_t lq[16][16];
    memset(lq, 0, sizeof(lq));
    int modes[16];
    int maxtc = 0;
    int dropped_max = 0;  // largest |level| zeroed by the CAVLC cap
    for (int z = 0; z < 16; ++z) {
      const int x4 = blk_x4(z), y4 = blk_y4(z);
      const int px = mbx * 16 + x4 * 4, py = mby * 16 + y4 * 4;
      const bool ht = y4 > 0 || mb_top;
      const bool hl = x4 > 0 || mb_left;
      const bool htl = (x4 > 0 && y4 > 0) || (x4 > 0 && mb_top) ||
                       (y4 > 0 && mb_left) || (mb_top && mb_left);
      // neighbour samples: EXACTLY the decoder's gather (incl. the
      // top-right replicate rule) so reconstructions agree bit-for-bit
      uint8_t tbuf[8] = {128, 128, 128, 128, 128, 128, 128, 128};
      uint8_t lbuf[4] = {128, 128, 128, 128};
      uint8_t tlv = 128;
      if (ht) {
        for (int i = 0; i < 4; ++i) tbuf[i] = base[(py - 1) * stride + px + i];
        bool htr;
        if (y4 == 0)
          htr = x4 < 3 ? mb_top
                       : (mby > 0 && mbx + 1 < mbw &&
                          ((mby - 1) * mbw + mbx + 1) >= slice_start);
        else
          htr = x4 < 3 && blk_z(x4 + 1, y4 - 1) < z;
        if (htr)
          for (int i = 0; i < 4; ++i)
            tbuf[4 + i] = base[(py - 1) * stride + px + 4 + i];
        else
          for (int i = 0; i < 4; ++i) tbuf[4 + i] = tbuf[3];
      }
      if (hl)
        for (int i = 0; i < 4; ++i) lbuf[i] = base[(py + i) * stride + px - 1];
      if (htl) tlv = base[(py - 1) * stride + px - 1];

      // SAD search over the legal modes
      uint8_t pred[16], best_pred[16];
      int best_mode = 2, best_sad = INT32_MAX;
      for (int m = 0; m < 9; ++m) {
        const bool needs_t = m == 0 || m == 3 || m == 7;
        const bool needs_l = m == 1 || m == 8;
        const bool needs_both = m == 4 || m == 5 || m == 6;
        if ((needs_t && !ht) || (needs_l && !hl) || (needs_both && !(ht && hl)))
          continue;
        pred_luma4(m, tbuf, lbuf, tlv, ht, hl, htl, pred);
        int sad = 0;
        for (int y = 0; y < 4; ++y)
          for (int x = 0; x < 4; ++x)
            sad += std::abs((int)srcy[(py + y) * stride + px + x] -
                            (int)pred[y * 4 + x]);
        if (sad < best_sad) {
          best_sad = sad;
          best_mode = m;
          memcpy(best_pred, pred, 16);
        }
      }
      modes[z] = best_mode;
      my_modes[z] = (int8_t)best_mode;

      int16_t d[16];
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x)
          d[y * 4 + x] = (int16_t)((int)srcy[(py + y) * stride + px + x] -
                                   (int)best_pred[y * 4 + x]);
      int32_t W[16];
      fwd4x4(d, W);
      for (int i = 0; i < 16; ++i) {
        int32_t q =
            (int32_t)(((int64_t)std::abs(W[i]) * QMF[qm][POSCLS[i]] + fr) >>
                      qbits);
        if (q > 2063) q = 2063;
        lq[z][i] = W[i] < 0 ? -q : q;
      }
      dropped_max = std::max(dropped_max, cap_coeffs9(lq[z], 0, 16));
      int tc = 0;
      for (int i = 0; i < 16; ++i) tc += lq[z][i] != 0;
      maxtc = std::max(maxtc, tc);
      // reconstruct immediately: later blocks predict from these pixels
      int32_t dq[16], rr[16];
      for (int i = 0; i < 16; ++i) dq[i] = (lq[z][i] * QV[qm][POSCLS[i]]) * (1 << qs);
      inv4x4(dq, rr);
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x)
          base[(py + y) * stride + px + x] =
              clip8(best_pred[y * 4 + x] + ((rr[y * 4 + x] + 32) >> 6));
    }

    // chroma (DC mode), same math as the I_16x16 path
    const int qpc = CHROMA_QP[qp < 0 ? 0 : qp > 51 ? 51 : qp];
    PlaneCtx rpcb{rCb.data(), pw / 2}, rpcr{rCr.data(), pw / 2};
    uint8_t cpred[2][64];
    int32_t cdc_q[2][4], cac_q[2][4][16];
    memset(cac_q, 0, sizeof(cac_q));
    const uint8_t* csrc2[2] = {Cb.data() + (mby * 8) * (pw / 2) + mbx * 8,
                               Cr.data() + (mby * 8) * (pw / 2) + mbx * 8};
    PlaneCtx* cpl[2] = {&rpcb, &rpcr};
    int ctc_max = 0;
    for (int comp = 0; comp < 2; ++comp) {
      pred_chroma8(*cpl[comp], mbx, mby, 0, mb_top, mb_left, cpred[comp]);
      int32_t cdc_raw[4];
      for (int blk = 0; blk < 4; ++blk) {
        const int x4 = blk & 1, y4 = blk >> 1;
        int16_t d[16];
        for (int y = 0; y < 4; ++y)
          for (int x = 0; x < 4; ++x)
            d[y * 4 + x] =
                (int16_t)((int)csrc2[comp][(y4 * 4 + y) * (pw / 2) + x4 * 4 + x] -
                          (int)cpred[comp][(y4 * 4 + y) * 8 + x4 * 4 + x]);
        int32_t W[16];
        fwd4x4(d, W);
        cdc_raw[blk] = W[0];
        quant_block(W, qpc, /*skip_dc=*/true, cac_q[comp][blk]);
        dropped_max = std::max(dropped_max,
                               cap_coeffs9(cac_q[comp][blk], 1, 16));
        int tc = 0;
        for (int i = 1; i < 16; ++i) tc += cac_q[comp][blk][i] != 0;
        ctc_max = std::max(ctc_max, tc);
      }
      int32_t a = cdc_raw[0], b2 = cdc_raw[1], c = cdc_raw[2], d2 = cdc_raw[3];
      int32_t t[4] = {a + b2 + c + d2, a - b2 + c - d2, a + b2 - c - d2,
                      a - b2 - c + d2};
      const int qmc = qpc % 6, qbc = 15 + qpc / 6;
      const int32_t fc = (1 << qbc) / 3;
      for (int i = 0; i < 4; ++i) {
        int32_t q = (int32_t)(((int64_t)std::abs(t[i]) * QMF[qmc][0] + 2 * fc) >>
                              (qbc + 1));
        if (q > 2063) q = 2063;
        cdc_q[comp][i] = t[i] < 0 ? -q : q;
      }
    }
    if (maxtc > 9 || ctc_max > 9 || dropped_max > PCM_DROP_LIMIT) {
      // CAVLC guard: the coefficient cap handles most high-entropy blocks
      // (cap_coeffs9); PCM remains only for blocks where capping dropped
      // large levels (real structure, not tail noise)
      for (int z = 0; z < 16; ++z) my_modes[z] = -1;
      write_pcm(wtr, mbx, mby, in_p_slice);
      return;
    }
    int cbp_chroma = 0;
    for (int comp = 0; comp < 2; ++comp)
      for (int blk = 0; blk < 4; ++blk)
        for (int i = 1; i < 16; ++i)
          if (cac_q[comp][blk][i]) cbp_chroma = 2;
    if (cbp_chroma == 0)
      for (int comp = 0; comp < 2; ++comp)
        for (int i = 0; i < 4; ++i)
          if (cdc_q[comp][i]) cbp_chroma = 1;
    int cbp_luma = 0;
    for (int i8 = 0; i8 < 4; ++i8) {
      bool any = false;
      for (int sub = 0; sub < 4; ++sub)
        for (int i = 0; i < 16; ++i)
          if (lq[i8 * 4 + sub][i]) any = true;
      if (any) cbp_luma |= 1 << i8;
    }

    // --- syntax ---
    wtr.ue(in_p_slice ? 5 : 0);  // mb_type: I_4x4 (+5 in P slices)
    for (int z = 0; z < 16; ++z) {
      const int x4 = blk_x4(z), y4 = blk_y4(z);
      int ma = 2, mb_ = 2;
      if (x4 > 0) {
        ma = modes[blk_z(x4 - 1, y4)];
      } else if (mb_left) {
        int8_t v = enc_i4modes[((size_t)mby * mbw + mbx - 1) * 16 + blk_z(3, y4)];
        ma = v < 0 ? 2 : v;
      }
      if (y4 > 0) {
        mb_ = modes[blk_z(x4, y4 - 1)];
      } else if (mb_top) {
        int8_t v = enc_i4modes[((size_t)(mby - 1) * mbw + mbx) * 16 + blk_z(x4, 3)];
        mb_ = v < 0 ? 2 : v;
      }
      const int predm = ma < mb_ ? ma : mb_;
      if (modes[z] == predm) {
        wtr.put(1, 1);  // prev_intra4x4_pred_mode_flag
      } else {
        wtr.put(0, 1);
        wtr.put(modes[z] < predm ? modes[z] : modes[z] - 1, 3);
      }
    }
    wtr.ue(0);  // intra_chroma_pred_mode: DC
    const int cbp = cbp_luma | (cbp_chroma << 4);
    wtr.ue((uint32_t)cbp_intra_code(cbp));
    if (cbp) wtr.se(0);  // mb_qp_delta
    for (int z = 0; z < 16; ++z) {
      const int x4 = blk_x4(z), y4 = blk_y4(z);
      if (cbp_luma & (1 << (z >> 2))) {
        int32_t scan[16];
        for (int i = 0; i < 16; ++i) scan[i] = lq[z][ZIGZAG[i]];
        int tc = cavlc_write_block(wtr, scan, 16,
                                   nnz.luma_nc(mbx, mby, x4, y4, slice_start));
        nnz.lnz(mbx, mby, x4, y4) = (uint8_t)tc;
      } else {
        nnz.lnz(mbx, mby, x4, y4) = 0;
      }
    }
    if (!cbp_chroma) memset(cdc_q, 0, sizeof(cdc_q));
    if (cbp_chroma < 2) memset(cac_q, 0, sizeof(cac_q));
    if (cbp_chroma)
      for (int comp = 0; comp < 2; ++comp)
        cavlc_write_block(wtr, cdc_q[comp], 4, -1);
    if (cbp_chroma == 2) {
      for (int comp = 0; comp < 2; ++comp)
        for (int blk = 0; blk < 4; ++blk) {
          const int x2 = blk & 1, y2 = blk >> 1;
          int32_t scan[15];
          for (int i = 1; i < 16; ++i) scan[i - 1] = cac_q[comp][blk][ZIGZAG[i]];
          int tc = cavlc_write_block(
              wtr, scan, 15, nnz.chroma_nc(comp, mbx, mby, x2, y2, slice_start));
          nnz.cnz(comp, mbx, mby, x2, y2) = (uint8_t)tc;
        }
    } else {
      for (int comp = 0; comp < 2; ++comp)
        for (int blk = 0; blk < 4; ++blk)
          nnz.cnz(comp, mbx, mby, blk & 1, blk >> 1) = 0;
    }
    recon_chroma8(rpcb, mbx, mby, cpred[0], cdc_q[0], cac_q[0], qpc);
    recon_chroma8(rpcr, mbx, mby, cpred[1], cdc_q[1], cac_q[1], qpc);
  }

  // 0 = I_16x16 only; 1 = I_4x4 only; 2 = auto (per-MB SAD decision)
  int mb_mode = 0;

  void encode_mb(BitWriter& wtr, int mbx, int mby, int qp, int slice_start,
                 bool in_p_slice = false) {
    if (mb_mode == 1) {
      encode_mb_i4x4(wtr, mbx, mby, qp, slice_start, in_p_slice);
      return;
    }
    PlaneCtx rpy{rY.data(), pw};
    PlaneCtx rpcb{rCb.data(), pw / 2}, rpcr{rCr.data(), pw / 2};
    const uint8_t* src = Y.data() + (mby * 16) * pw + mbx * 16;

    // --- luma mode decision (SAD over available modes) ---
    uint8_t pred[4][256];
    int best_mode = 2, best_sad = INT32_MAX;
    const bool have_top = nnz.top_ok(mbx, mby, slice_start);
    const bool have_left = nnz.left_ok(mbx, mby, slice_start);
    for (int m = 0; m < 4; ++m) {
      if (m == 0 && !have_top) continue;
      if (m == 1 && !have_left) continue;
      if (m == 3 && !(have_top && have_left)) continue;
      pred_luma16(rpy, mbx, mby, m, have_top, have_left, pred[m]);
      int sad = mb_sad(src, pw, pred[m], 16, 16);
      if (sad < best_sad) {
        best_sad = sad;
        best_mode = m;
      }
    }
    const uint8_t* lp = pred[best_mode];

    // auto mode: a moderately detailed MB (mean |residual| in (5, 24]
    // under its best 16x16 prediction) is usually cheaper as I_4x4 with
    // per-block modes. Above that the TotalCoeff guard routes to I_PCM on
    // either path, so skip the 9-mode search and let the I_16x16 path
    // reach the same PCM cheaply (noise frames would otherwise pay the
    // full search for nothing).
    if (mb_mode == 2 && best_sad > 256 * 5 && best_sad <= 256 * 24) {
      encode_mb_i4x4(wtr, mbx, mby, qp, slice_start, in_p_slice);
      return;
    }

    // --- luma transform/quant ---
    int32_t dc_raw[16];             // raster 4x4 of DC terms
    int32_t ac_q[16][16] = {{0}};   // raster-order quantized, [0] zero
    int32_t W[16];
    for (int b = 0; b < 16; ++b) {
      const int x4 = blk_x4(b), y4 = blk_y4(b);
      int16_t d[16];
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x)
          d[y * 4 + x] = (int16_t)((int)src[(y4 * 4 + y) * pw + x4 * 4 + x] -
                                   (int)lp[(y4 * 4 + y) * 16 + x4 * 4 + x]);
      fwd4x4(d, W);
      dc_raw[y4 * 4 + x4] = W[0];
      quant_block(W, qp, /*skip_dc=*/true, ac_q[b]);
    }
    // luma DC: Hadamard (halved, JM-style) + quant with doubled round/shift
    int32_t dct[16], dc_q[16];
    hadamard4x4(dc_raw, dct);
    {
      const int qm = qp % 6, qbits = 15 + qp / 6;
      const int32_t f = (1 << qbits) / 3;
      for (int i = 0; i < 16; ++i) {
        const int32_t hv = dct[i] >= 0 ? (dct[i] + 1) >> 1 : -((1 - dct[i]) >> 1);
        int32_t q = (int32_t)(((int64_t)std::abs(hv) * QMF[qm][0] + 2 * f) >>
                              (qbits + 1));
        if (q > 2063) q = 2063;
        dc_q[i] = hv < 0 ? -q : q;
      }
    }
    int cbp_luma = 0;
    for (int b = 0; b < 16 && !cbp_luma; ++b)
      for (int i = 1; i < 16; ++i)
        if (ac_q[b][i]) {
          cbp_luma = 15;
          break;
        }

    // --- chroma (DC prediction mode 0) ---
    const int qpc = CHROMA_QP[qp < 0 ? 0 : qp > 51 ? 51 : qp];
    uint8_t cpred[2][64];
    int32_t cdc_q[2][4], cac_q[2][4][16];
    memset(cac_q, 0, sizeof(cac_q));
    const uint8_t* csrc[2] = {Cb.data() + (mby * 8) * (pw / 2) + mbx * 8,
                              Cr.data() + (mby * 8) * (pw / 2) + mbx * 8};
    PlaneCtx* cpl[2] = {&rpcb, &rpcr};
    for (int comp = 0; comp < 2; ++comp) {
      pred_chroma8(*cpl[comp], mbx, mby, 0, have_top, have_left, cpred[comp]);
      int32_t cdc_raw[4];
      for (int blk = 0; blk < 4; ++blk) {
        const int x4 = blk & 1, y4 = blk >> 1;
        int16_t d[16];
        for (int y = 0; y < 4; ++y)
          for (int x = 0; x < 4; ++x)
            d[y * 4 + x] =
                (int16_t)((int)csrc[comp][(y4 * 4 + y) * (pw / 2) + x4 * 4 + x] -
                          (int)cpred[comp][(y4 * 4 + y) * 8 + x4 * 4 + x]);
        fwd4x4(d, W);
        cdc_raw[blk] = W[0];
        quant_block(W, qpc, /*skip_dc=*/true, cac_q[comp][blk]);
      }
      // 2x2 Hadamard + quant (doubled rounding/shift)
      int32_t a = cdc_raw[0], b2 = cdc_raw[1], c = cdc_raw[2], d2 = cdc_raw[3];
      int32_t t[4] = {a + b2 + c + d2, a - b2 + c - d2, a + b2 - c - d2,
                      a - b2 - c + d2};
      const int qm = qpc % 6, qbits = 15 + qpc / 6;
      const int32_t f = (1 << qbits) / 3;
      for (int i = 0; i < 4; ++i) {
        int32_t q = (int32_t)(((int64_t)std::abs(t[i]) * QMF[qm][0] + 2 * f) >>
                              (qbits + 1));
        if (q > 2063) q = 2063;
        cdc_q[comp][i] = t[i] < 0 ? -q : q;
      }
    }
    int cbp_chroma = 0;
    for (int comp = 0; comp < 2; ++comp)
      for (int blk = 0; blk < 4; ++blk)
        for (int i = 1; i < 16; ++i)
          if (cac_q[comp][blk][i]) cbp_chroma = 2;
    if (cbp_chroma == 0)
      for (int comp = 0; comp < 2; ++comp)
        for (int i = 0; i < 4; ++i)
          if (cdc_q[comp][i]) cbp_chroma = 1;

    // --- entropy-region guard ---
    // The encoder restricts itself to TotalCoeff <= 9 per block: the
    // coeff_token rows above that (all nC buckets) are best-effort
    // reconstructions with no offline reference to validate against
    // (no ffmpeg/x264 in the build image). High-entropy macroblocks
    // (TotalCoeff > 9 anywhere) are emitted as I_PCM instead — standard,
    // bit-exact by construction, and what real encoders do when CAVLC
    // cost approaches raw anyway.
    auto count_nz = [](const int32_t* c, int from, int n) {
      int k = 0;
      for (int i = from; i < n; ++i) k += c[i] != 0;
      return k;
    };
    int drop16 = cap_coeffs9(dc_q, 0, 16);
    for (int b = 0; b < 16; ++b)
      drop16 = std::max(drop16, cap_coeffs9(ac_q[b], 1, 16));
    for (int comp = 0; comp < 2; ++comp)
      for (int blk = 0; blk < 4; ++blk)
        drop16 = std::max(drop16, cap_coeffs9(cac_q[comp][blk], 1, 16));
    if (drop16 > PCM_DROP_LIMIT) {
      // the cap would erase strong structure: keep the exact pixels
      write_pcm(wtr, mbx, mby, in_p_slice);
      return;
    }

    // --- write macroblock_layer ---
    const int mb_type = 1 + best_mode + 4 * cbp_chroma + 12 * (cbp_luma ? 1 : 0)
                        + (in_p_slice ? 5 : 0);
    wtr.ue(mb_type);
    wtr.ue(0);  // intra_chroma_pred_mode: DC
    wtr.se(0);  // mb_qp_delta (constant QP per frame)

    // Intra16x16DCLevel: zig-zag scan of dc_q, nC from luma block 0 neighbors
    {
      int32_t scan[16];
      for (int i = 0; i < 16; ++i) scan[i] = dc_q[ZIGZAG[i]];
      cavlc_write_block(wtr, scan, 16, nnz.luma_nc(mbx, mby, 0, 0, slice_start));
    }
    if (cbp_luma) {
      for (int b = 0; b < 16; ++b) {
        const int x4 = blk_x4(b), y4 = blk_y4(b);
        int32_t scan[15];
        for (int i = 1; i < 16; ++i) scan[i - 1] = ac_q[b][ZIGZAG[i]];
        int tc = cavlc_write_block(wtr, scan, 15,
                                   nnz.luma_nc(mbx, mby, x4, y4, slice_start));
        nnz.lnz(mbx, mby, x4, y4) = (uint8_t)tc;
      }
    } else {
      for (int b = 0; b < 16; ++b)
        nnz.lnz(mbx, mby, blk_x4(b), blk_y4(b)) = 0;
    }
    if (cbp_chroma) {
      for (int comp = 0; comp < 2; ++comp)
        cavlc_write_block(wtr, cdc_q[comp], 4, -1);
    }
    if (cbp_chroma == 2) {
      for (int comp = 0; comp < 2; ++comp)
        for (int blk = 0; blk < 4; ++blk) {
          const int x2 = blk & 1, y2 = blk >> 1;
          int32_t scan[15];
          for (int i = 1; i < 16; ++i) scan[i - 1] = cac_q[comp][blk][ZIGZAG[i]];
          int tc = cavlc_write_block(
              wtr, scan, 15, nnz.chroma_nc(comp, mbx, mby, x2, y2, slice_start));
          nnz.cnz(comp, mbx, mby, x2, y2) = (uint8_t)tc;
        }
    } else {
      for (int comp = 0; comp < 2; ++comp)
        for (int blk = 0; blk < 4; ++blk)
          nnz.cnz(comp, mbx, mby, blk & 1, blk >> 1) = 0;
    }

    // --- reconstruct (shared with the decoder) ---
    // zero out levels the stream does not carry so recon matches decode
    if (!cbp_luma) memset(ac_q, 0, sizeof(ac_q));
    if (cbp_chroma < 2) memset(cac_q, 0, sizeof(cac_q));
    if (!cbp_chroma) memset(cdc_q, 0, sizeof(cdc_q));
    recon_luma16(rpy, mbx, mby, lp, dc_q, ac_q, qp);
    recon_chroma8(rpcb, mbx, mby, cpred[0], cdc_q[0], cac_q[0], qpc);
    recon_chroma8(rpcr, mbx, mby, cpred[1], cdc_q[1], cac_q[1], qpc);
  }

  int n_slices = 1;  // MB-row bands, encoded in parallel threads
  std::unique_ptr<WorkPool> pool;  // persistent slice workers
  WorkPool& get_pool(int n) {
    if (!pool || pool->size() < n) pool.reset(new WorkPool(n));
    return *pool;
  }

  int encode(const uint8_t* rgb, int qp, uint8_t* out, int cap) {
    return encode_ex(rgb, qp, /*force_idr=*/1, out, cap);
  }

  // SAD of a source 16x16 luma MB vs the reconstructed reference
  int mb_ref_sad(int mbx, int mby) const {
    const uint8_t* sy = Y.data() + (mby * 16) * pw + mbx * 16;
    const uint8_t* ry = rY.data() + (mby * 16) * pw + mbx * 16;
    int sad = 0;
    for (int y = 0; y < 16; ++y)
      for (int x = 0; x < 16; ++x)
        sad += std::abs((int)sy[y * pw + x] - (int)ry[y * pw + x]);
    const uint8_t* sc[2] = {Cb.data() + (mby * 8) * (pw / 2) + mbx * 8,
                            Cr.data() + (mby * 8) * (pw / 2) + mbx * 8};
    const uint8_t* rc[2] = {rCb.data() + (mby * 8) * (pw / 2) + mbx * 8,
                            rCr.data() + (mby * 8) * (pw / 2) + mbx * 8};
    for (int comp = 0; comp < 2; ++comp)
      for (int y = 0; y < 8; ++y)
        for (int x = 0; x < 8; ++x)
          sad += std::abs((int)sc[comp][y * (pw / 2) + x] -
                          (int)rc[comp][y * (pw / 2) + x]);
    return sad;
  }

  // force_idr=0 allows a P frame (P_Skip for unchanged MBs + intra-refresh
  // MBs for changed regions — zero-MV conditional replenishment, the
  // low-latency WebRTC P shape). Falls back to IDR when no reference
  // exists yet.
  int encode_ex(const uint8_t* rgb, int qp, int force_idr, uint8_t* out,
                int cap) {
    if (qp < 10) qp = 10;
    if (qp > 48) qp = 48;
    const bool idr = force_idr || !have_ref;
    const int ns = std::max(1, std::min({n_slices, mbh, 16}));
    // colour conversion is FUSED into each slice band (one pool dispatch
    // per frame instead of two, and the band stays L2-hot into its encode;
    // encode never reads outside its own band: intra prediction does not
    // cross slice boundaries). Single-slice path converts here.
    if (ns == 1)
      rgb_to_yuv420(rgb, w, h, pw, ph, Y.data(), Cb.data(), Cr.data());
    nnz.reset(mbw, mbh);
    enc_i4modes.assign((size_t)mbw * mbh * 16, -1);

    std::vector<uint8_t> bs;
    if (idr) {
      uint8_t hdr[256];
      int n = airtc_h264_sps_pps(w, h, hdr, sizeof(hdr));
      if (n <= 0) return -1;
      bs.insert(bs.end(), hdr, hdr + n);
      frame_num = 0;
    }
    const uint32_t pic_idr_id = idr_id & 1;
    if (idr) ++idr_id;
    const uint32_t fnum = frame_num & 15;
    frame_num = (frame_num + 1) & 15;

    // one slice per MB-row band; slices only predict within themselves,
    // so bands encode concurrently (recon rows + nnz rows are disjoint)
    const int band = (mbh + ns - 1) / ns;
    std::vector<std::vector<uint8_t>> slice_nals(ns);
    auto encode_band = [&](int si) {
      const int r0 = si * band, r1 = std::min(mbh, r0 + band);
      if (r0 >= r1) return;
      if (ns > 1)  // fused per-band colour conversion (16 | band rows)
        rgb_to_yuv420_rows(rgb, w, h, pw, r0 * 16, std::min(ph, r1 * 16),
                           Y.data(), Cb.data(), Cr.data());
      const int slice_start = r0 * mbw;
      BitWriter wtr;
      wtr.ue((uint32_t)slice_start);  // first_mb_in_slice
      wtr.ue(idr ? 7 : 5);            // slice_type: I / P (all slices)
      wtr.ue(0);                      // pps_id
      wtr.put(fnum, 4);               // frame_num (log2_max_frame_num = 4)
      if (idr) {
        wtr.ue(pic_idr_id);           // idr_pic_id
        wtr.put(0, 1);                // no_output_of_prior_pics_flag
        wtr.put(0, 1);                // long_term_reference_flag
      } else {
        wtr.put(0, 1);  // num_ref_idx_active_override_flag
        wtr.put(0, 1);  // ref_pic_list_modification_flag_l0
        wtr.put(0, 1);  // adaptive_ref_pic_marking_mode_flag
      }
      wtr.se(qp - 26);                // slice_qp_delta (pic_init_qp = 26)
      wtr.ue(1);                      // disable_deblocking_filter_idc = off
      uint32_t skip_run = 0;
      for (int mby = r0; mby < r1; ++mby)
        for (int mbx = 0; mbx < mbw; ++mbx) {
          if (!idr && mb_ref_sad(mbx, mby) <= skip_thresh_for(qp)) {
            // P_Skip: reconstruction = co-located reference (recon planes
            // already hold it); zero nnz + not-I4x4 for neighbour context
            ++skip_run;
            for (int b = 0; b < 16; ++b)
              nnz.lnz(mbx, mby, blk_x4(b), blk_y4(b)) = 0;
            for (int comp = 0; comp < 2; ++comp)
              for (int blk = 0; blk < 4; ++blk)
                nnz.cnz(comp, mbx, mby, blk & 1, blk >> 1) = 0;
            continue;
          }
          if (!idr) {
            wtr.ue(skip_run);  // mb_skip_run before every coded MB
            skip_run = 0;
          }
          encode_mb(wtr, mbx, mby, qp, slice_start, /*in_p_slice=*/!idr);
        }
      if (!idr) wtr.ue(skip_run);  // trailing skipped MBs
      wtr.rbsp_trailing();
      // nal_ref_idc=3; type 5 (IDR) or 1 (non-IDR)
      emit_nal(&slice_nals[si], idr ? 0x65 : 0x61, wtr.bytes);
    };
    if (ns > 1) {
      get_pool(ns).run(ns, encode_band);
    } else {
      encode_band(0);
    }
    for (auto& nal : slice_nals) bs.insert(bs.end(), nal.begin(), nal.end());

    if ((int)bs.size() > cap) return -2;
    memcpy(out, bs.data(), bs.size());
    have_ref = true;
    return (int)bs.size();
  }
};

// ---------------------------------------------------------------------------
// Decoder
// ---------------------------------------------------------------------------
struct Sps {
  bool valid = false;
  int log2_max_frame_num = 4;
  int poc_type = 2;
  int log2_max_poc_lsb = 4;
  int pw = 0, ph = 0;      // padded (MB-aligned) dims
  int crop_r = 0, crop_b = 0;
  int w() const { return pw - crop_r; }
  int h() const { return ph - crop_b; }
};
struct Pps {
  bool valid = false;
  bool cavlc = true;
  bool pic_order_present = false;
  int pic_init_qp = 26;
  int chroma_qp_offset = 0;
  bool deblock_present = false;
};

struct Decoder {
  std::unique_ptr<WorkPool> dpool;  // persistent slice workers
  Sps sps;
  Pps pps;
  std::vector<uint8_t> rY, rCb, rCr;
  NnzCtx nnz;
  // per-4x4 intra pred modes of I_4x4 MBs (-1 = MB not I_4x4): neighbours'
  // modes feed the predicted-mode rule; sized mbw*mbh*16, reset per frame
  std::vector<int8_t> i4modes;
  // P frames reference the previous reconstruction (rY/rCb/rCr persist
  // between access units; P_Skip = leave the co-located pixels untouched).
  // A P slice before any IDR has no reference and is refused.
  bool have_idr = false;

  int parse_sps(BitReader& r) {
    Sps s;
    int profile = r.u(8);
    r.u(8);  // constraint flags
    r.u(8);  // level
    r.ue();  // sps_id
    if (profile == 100 || profile == 110 || profile == 122 || profile == 244 ||
        profile == 44 || profile == 83 || profile == 86 || profile == 118 ||
        profile == 128) {
      int chroma = r.ue();
      if (chroma != 1) return -1;  // 4:2:0 only
      r.ue();                      // bit_depth_luma_minus8
      r.ue();                      // bit_depth_chroma_minus8
      r.u(1);                      // qpprime
      if (r.u(1)) return -1;       // scaling matrices unsupported
    }
    s.log2_max_frame_num = (int)r.ue() + 4;
    s.poc_type = (int)r.ue();
    if (s.poc_type == 0)
      s.log2_max_poc_lsb = (int)r.ue() + 4;
    else if (s.poc_type == 1)
      return -1;
    r.ue();  // max_num_ref_frames
    r.u(1);  // gaps allowed
    int mbw = (int)r.ue() + 1;
    int mbh = (int)r.ue() + 1;
    if (!r.u(1)) return -1;  // frame_mbs_only required
    r.u(1);                  // direct_8x8
    s.pw = mbw * 16;
    s.ph = mbh * 16;
    if (r.u(1)) {  // cropping
      int cl = (int)r.ue() * 2, cr = (int)r.ue() * 2;
      int ct = (int)r.ue() * 2, cb = (int)r.ue() * 2;
      if (cl || ct) return -1;  // left/top crop unsupported
      s.crop_r = cr;
      s.crop_b = cb;
    }
    if (r.overrun || s.pw <= 0 || s.ph <= 0 || s.pw > 8192 || s.ph > 8192)
      return -1;
    s.valid = true;
    sps = s;
    rY.assign((size_t)sps.pw * sps.ph, 0);
    rCb.assign((size_t)sps.pw * sps.ph / 4, 128);
    rCr.assign((size_t)sps.pw * sps.ph / 4, 128);
    return 0;
  }

  int parse_pps(BitReader& r) {
    Pps p;
    r.ue();  // pps_id
    r.ue();  // sps_id
    p.cavlc = r.u(1) == 0;
    if (!p.cavlc) return -1;  // CABAC unsupported
    p.pic_order_present = r.u(1) != 0;
    if (r.ue() != 0) return -1;  // slice groups unsupported
    r.ue();                      // num_ref_idx_l0
    r.ue();                      // num_ref_idx_l1
    r.u(1);                      // weighted_pred
    r.u(2);                      // weighted_bipred
    p.pic_init_qp = 26 + r.se();
    r.se();  // pic_init_qs
    p.chroma_qp_offset = r.se();
    p.deblock_present = r.u(1) != 0;
    r.u(1);  // constrained_intra
    r.u(1);  // redundant_pic_cnt_present
    if (r.overrun) return -1;
    p.valid = true;
    pps = p;
    return 0;
  }

  // parse + reconstruct the chroma residual of one intra MB (shared by the
  // I_16x16 and I_4x4 paths; cbp_chroma: 0 none, 1 DC only, 2 DC+AC)
  int decode_chroma(BitReader& r, int mbx, int mby, int cbp_chroma, int cm,
                    int qp, int slice_start, bool have_top, bool have_left) {
    PlaneCtx rpcb{rCb.data(), sps.pw / 2}, rpcr{rCr.data(), sps.pw / 2};
    int32_t cdc_q[2][4] = {{0}}, cac_q[2][4][16];
    memset(cac_q, 0, sizeof(cac_q));
    if (cbp_chroma) {
      for (int comp = 0; comp < 2; ++comp) {
        int32_t scan[4];
        if (cavlc_read_block(r, scan, 4, -1) < 0) return -1;
        for (int i = 0; i < 4; ++i) cdc_q[comp][i] = scan[i];
      }
    }
    if (cbp_chroma == 2) {
      for (int comp = 0; comp < 2; ++comp)
        for (int blk = 0; blk < 4; ++blk) {
          const int x2 = blk & 1, y2 = blk >> 1;
          int32_t scan[15];
          int tc = cavlc_read_block(r, scan, 15,
                                    nnz.chroma_nc(comp, mbx, mby, x2, y2,
                                                  slice_start));
          if (tc < 0) return -1;
          for (int i = 1; i < 16; ++i) cac_q[comp][blk][ZIGZAG[i]] = scan[i - 1];
          nnz.cnz(comp, mbx, mby, x2, y2) = (uint8_t)tc;
        }
    } else {
      for (int comp = 0; comp < 2; ++comp)
        for (int blk = 0; blk < 4; ++blk)
          nnz.cnz(comp, mbx, mby, blk & 1, blk >> 1) = 0;
    }
    const int qpi = qp + pps.chroma_qp_offset;
    const int qpc = CHROMA_QP[qpi < 0 ? 0 : qpi > 51 ? 51 : qpi];
    if ((cm == 2 && !have_top) || (cm == 1 && !have_left) ||
        (cm == 3 && !(have_top && have_left)))
      return -1;
    uint8_t cpred[64];
    pred_chroma8(rpcb, mbx, mby, cm, have_top, have_left, cpred);
    recon_chroma8(rpcb, mbx, mby, cpred, cdc_q[0], cac_q[0], qpc);
    pred_chroma8(rpcr, mbx, mby, cm, have_top, have_left, cpred);
    recon_chroma8(rpcr, mbx, mby, cpred, cdc_q[1], cac_q[1], qpc);
    return 0;
  }

  // decode one I_4x4 macroblock (all 9 luma pred modes, full CBP syntax)
  int decode_mb_i4x4(BitReader& r, int mbx, int mby, int& qp,
                     int slice_start) {
    const int mbw = sps.pw / 16;
    const bool mb_top = nnz.top_ok(mbx, mby, slice_start);
    const bool mb_left = nnz.left_ok(mbx, mby, slice_start);
    int8_t* my_modes = &i4modes[((size_t)mby * mbw + mbx) * 16];

    int modes[16];
    for (int z = 0; z < 16; ++z) {
      const int x4 = blk_x4(z), y4 = blk_y4(z);
      int ma = 2, mb_ = 2;  // DC when unavailable / neighbour not I_4x4
      if (x4 > 0) {
        ma = modes[blk_z(x4 - 1, y4)];
      } else if (mb_left) {
        int8_t v = i4modes[((size_t)mby * mbw + mbx - 1) * 16 + blk_z(3, y4)];
        ma = v < 0 ? 2 : v;
      }
      if (y4 > 0) {
        mb_ = modes[blk_z(x4, y4 - 1)];
      } else if (mb_top) {
        int8_t v = i4modes[((size_t)(mby - 1) * mbw + mbx) * 16 + blk_z(x4, 3)];
        mb_ = v < 0 ? 2 : v;
      }
      const int pred = ma < mb_ ? ma : mb_;
      if (r.u(1)) {
        modes[z] = pred;
      } else {
        const int rem = (int)r.u(3);
        modes[z] = rem < pred ? rem : rem + 1;
      }
      my_modes[z] = (int8_t)modes[z];
    }
    const uint32_t chroma_mode = r.ue();
    if (chroma_mode > 3 || r.overrun) return -1;
    const uint32_t cbp_code = r.ue();
    if (cbp_code >= 48) return -1;
    const int cbp = CBP_INTRA[cbp_code];
    const int cbp_luma = cbp & 15, cbp_chroma = cbp >> 4;
    if (cbp) {
      qp += r.se();
      if (qp < 0 || qp > 51) return -1;
    }

    // parse luma levels (full 16-coeff blocks, zig-zag)
    int32_t lq[16][16];
    memset(lq, 0, sizeof(lq));
    for (int z = 0; z < 16; ++z) {
      const int x4 = blk_x4(z), y4 = blk_y4(z);
      if (cbp_luma & (1 << (z >> 2))) {
        int32_t scan[16];
        int tc = cavlc_read_block(r, scan, 16,
                                  nnz.luma_nc(mbx, mby, x4, y4, slice_start));
        if (tc < 0) return -1;
        for (int i = 0; i < 16; ++i) lq[z][ZIGZAG[i]] = scan[i];
        nnz.lnz(mbx, mby, x4, y4) = (uint8_t)tc;
      } else {
        nnz.lnz(mbx, mby, x4, y4) = 0;
      }
    }

    // reconstruct blocks in z order (later blocks predict from earlier
    // reconstructed pixels)
    const int qm = qp % 6, qs = qp / 6;
    const int stride = sps.pw;
    uint8_t* base = rY.data();
    for (int z = 0; z < 16; ++z) {
      const int x4 = blk_x4(z), y4 = blk_y4(z);
      const int px = mbx * 16 + x4 * 4, py = mby * 16 + y4 * 4;
      const bool ht = y4 > 0 || mb_top;
      const bool hl = x4 > 0 || mb_left;
      const bool htl = (x4 > 0 && y4 > 0) || (x4 > 0 && mb_top) ||
                       (y4 > 0 && mb_left) || (mb_top && mb_left);
      uint8_t tbuf[8] = {128, 128, 128, 128, 128, 128, 128, 128};
      uint8_t lbuf[4] = {128, 128, 128, 128};
      uint8_t tlv = 128;
      if (ht) {
        for (int i = 0; i < 4; ++i) tbuf[i] = base[(py - 1) * stride + px + i];
        bool htr;
        if (y4 == 0) {
          // above(-right) samples come from the MB row above
          htr = x4 < 3 ? mb_top
                       : (mby > 0 && mbx + 1 < mbw &&
                          ((mby - 1) * mbw + mbx + 1) >= slice_start);
        } else {
          htr = x4 < 3 && blk_z(x4 + 1, y4 - 1) < z;
        }
        if (htr)
          for (int i = 0; i < 4; ++i)
            tbuf[4 + i] = base[(py - 1) * stride + px + 4 + i];
        else
          for (int i = 0; i < 4; ++i) tbuf[4 + i] = tbuf[3];
      }
      if (hl)
        for (int i = 0; i < 4; ++i) lbuf[i] = base[(py + i) * stride + px - 1];
      if (htl) tlv = base[(py - 1) * stride + px - 1];

      const int m = modes[z];
      // modes that need absent neighbours are illegal in a valid stream
      const bool needs_t = m == 0 || m == 3 || m == 7;
      const bool needs_l = m == 1 || m == 8;
      const bool needs_both = m == 4 || m == 5 || m == 6;
      if ((needs_t && !ht) || (needs_l && !hl) || (needs_both && !(ht && hl)))
        return -1;
      uint8_t pred[16];
      pred_luma4(m, tbuf, lbuf, tlv, ht, hl, htl, pred);
      int32_t d[16];
      for (int i = 0; i < 16; ++i) d[i] = (lq[z][i] * QV[qm][POSCLS[i]]) * (1 << qs);
      int32_t rr[16];
      inv4x4(d, rr);
      for (int y = 0; y < 4; ++y)
        for (int x = 0; x < 4; ++x)
          base[(py + y) * stride + px + x] =
              clip8(pred[y * 4 + x] + ((rr[y * 4 + x] + 32) >> 6));
    }
    return decode_chroma(r, mbx, mby, cbp_chroma, (int)chroma_mode, qp,
                         slice_start, mb_top, mb_left);
  }

  // decode one intra macroblock (I_4x4 / I_16x16 / I_PCM; +5 type offset
  // inside P slices); returns 0 or negative error
  int decode_mb(BitReader& r, int mbx, int mby, int& qp, int slice_start,
                bool in_p_slice = false) {
    PlaneCtx rpy{rY.data(), sps.pw};
    uint32_t mb_type = r.ue();
    if (r.overrun) return -1;
    if (in_p_slice) {
      if (mb_type < 5) return -2;  // explicit inter MBs unsupported
      mb_type -= 5;
    }
    {  // mark this MB as not-I_4x4 for neighbours' mode prediction
      const int mbw = sps.pw / 16;
      int8_t* mm = &i4modes[((size_t)mby * mbw + mbx) * 16];
      for (int i = 0; i < 16; ++i) mm[i] = -1;
    }
    if (mb_type == 0) return decode_mb_i4x4(r, mbx, mby, qp, slice_start);
    if (mb_type == 25) {  // I_PCM
      r.align_byte();
      uint8_t* dst = rY.data() + (mby * 16) * sps.pw + mbx * 16;
      for (int y = 0; y < 16; ++y)
        for (int x = 0; x < 16; ++x) dst[y * sps.pw + x] = (uint8_t)r.u(8);
      uint8_t* pc[2] = {rCb.data() + (mby * 8) * (sps.pw / 2) + mbx * 8,
                        rCr.data() + (mby * 8) * (sps.pw / 2) + mbx * 8};
      for (int comp = 0; comp < 2; ++comp)
        for (int y = 0; y < 8; ++y)
          for (int x = 0; x < 8; ++x) pc[comp][y * (sps.pw / 2) + x] = (uint8_t)r.u(8);
      for (int b = 0; b < 16; ++b) nnz.lnz(mbx, mby, blk_x4(b), blk_y4(b)) = 16;
      for (int comp = 0; comp < 2; ++comp)
        for (int blk = 0; blk < 4; ++blk)
          nnz.cnz(comp, mbx, mby, blk & 1, blk >> 1) = 16;
      return r.overrun ? -1 : 0;
    }
    if (mb_type < 1 || mb_type > 24) return -2;  // I_4x4 / non-intra: unsupported
    const int code = (int)mb_type - 1;
    const int pred_mode = code % 4;
    const int cbp_chroma = (code / 4) % 3;
    const int cbp_luma = (code / 12) ? 15 : 0;

    const uint32_t chroma_mode = r.ue();
    if (chroma_mode > 3) return -1;
    const int dqp = r.se();
    qp = qp + dqp;
    if (qp < 0 || qp > 51) return -1;

    // residual parse
    int32_t dc_scan[16], dc_q[16] = {0};
    int tc_dc = cavlc_read_block(r, dc_scan, 16,
                                 nnz.luma_nc(mbx, mby, 0, 0, slice_start));
    if (tc_dc < 0) return -1;
    for (int i = 0; i < 16; ++i) dc_q[ZIGZAG[i]] = dc_scan[i];
    int32_t ac_q[16][16];
    memset(ac_q, 0, sizeof(ac_q));
    if (cbp_luma) {
      for (int b = 0; b < 16; ++b) {
        const int x4 = blk_x4(b), y4 = blk_y4(b);
        int32_t scan[15];
        int tc = cavlc_read_block(r, scan, 15,
                                  nnz.luma_nc(mbx, mby, x4, y4, slice_start));
        if (tc < 0) return -1;
        for (int i = 1; i < 16; ++i) ac_q[b][ZIGZAG[i]] = scan[i - 1];
        nnz.lnz(mbx, mby, x4, y4) = (uint8_t)tc;
      }
    } else {
      for (int b = 0; b < 16; ++b)
        nnz.lnz(mbx, mby, blk_x4(b), blk_y4(b)) = 0;
    }
    // predict + reconstruct luma, then the shared chroma path
    uint8_t lpred[256];
    const bool have_top = nnz.top_ok(mbx, mby, slice_start);
    const bool have_left = nnz.left_ok(mbx, mby, slice_start);
    if ((pred_mode == 0 && !have_top) || (pred_mode == 1 && !have_left) ||
        (pred_mode == 3 && !(have_top && have_left)))
      return -1;
    pred_luma16(rpy, mbx, mby, pred_mode, have_top, have_left, lpred);
    recon_luma16(rpy, mbx, mby, lpred, dc_q, ac_q, qp);
    return decode_chroma(r, mbx, mby, cbp_chroma, (int)chroma_mode, qp,
                         slice_start, have_top, have_left);
  }

  // returns 1 when a frame was reconstructed, 0 for parameter-set-only data,
  // negative on unsupported/corrupt input
  struct SliceJob {
    std::vector<uint8_t> rbsp;
    size_t header_bitpos;
    int first_mb;
    int qp;
    bool is_p;
  };

  int decode_au(const uint8_t* data, int len, uint8_t* rgb, int cap, int* ow,
                int* oh) {
    auto nals = h264::split_annexb(data, (size_t)len);
    bool got_frame = false;
    std::vector<SliceJob> slices;
    for (auto& nal : nals) {
      if (nal.size < 1) continue;
      const int type = nal.data[0] & 0x1F;
      auto rbsp = h264::unescape_rbsp(nal.data + 1, nal.size - 1);
      BitReader r(rbsp.data(), rbsp.size());
      if (type == 7) {
        if (parse_sps(r) < 0) return -3;
      } else if (type == 8) {
        if (parse_pps(r) < 0) return -3;
      } else if (type == 5 || type == 1) {
        if (!sps.valid || !pps.valid) return -4;
        int first_mb = (int)r.ue();
        uint32_t stype = r.ue();
        const bool is_p = stype % 5 == 0;
        if (stype % 5 != 2 && !is_p) return -6;  // I and P slices only
        if (is_p && type == 5) return -6;        // IDR must be intra
        if (is_p && !have_idr) return -10;       // no reference yet
        r.ue();                         // pps_id
        r.u(sps.log2_max_frame_num);    // frame_num
        if (type == 5) r.ue();          // idr_pic_id
        if (sps.poc_type == 0) {
          r.u(sps.log2_max_poc_lsb);
          if (pps.pic_order_present) r.se();
        }
        if (is_p) {
          if (r.u(1)) r.ue();     // num_ref_idx_active_override -> l0 count
          if (r.u(1)) return -7;  // ref_pic_list_modification unsupported
        }
        if (type == 5) {
          r.u(1);  // no_output_of_prior_pics
          r.u(1);  // long_term_reference
        } else if (nal.data[0] & 0x60) {
          if (r.u(1)) return -7;  // adaptive marking unsupported
        }
        int qp = pps.pic_init_qp + r.se();
        if (pps.deblock_present) {
          uint32_t idc = r.ue();
          if (idc != 1) {
            r.se();
            r.se();
            // deblocking requested: we decode without the loop filter.
            // Bit-exact only for idc==1 streams (our encoder's output).
          }
        }
        if (qp < 0 || qp > 51 || r.overrun) return -1;
        const int total = (sps.pw / 16) * (sps.ph / 16);
        if (first_mb < 0 || first_mb >= total) return -5;
        slices.push_back({std::move(rbsp), r.pos, first_mb, qp, is_p});
        continue;  // rbsp moved into the job
      }
    }
    // decode collected slices — slices are independent (neighbours in a
    // different slice are unavailable), so they run in parallel threads
    if (!slices.empty()) {
      const int mbw = sps.pw / 16, mbh = sps.ph / 16;
      const int total = mbw * mbh;
      nnz.reset(mbw, mbh);
      i4modes.assign((size_t)total * 16, -1);
      std::vector<int> rcs(slices.size(), 0), counts(slices.size(), 0);
      auto decode_slice = [&](size_t si) {
        SliceJob& job = slices[si];
        BitReader sr(job.rbsp.data(), job.rbsp.size());
        sr.pos = job.header_bitpos;
        int mb = job.first_mb;
        int qp = job.qp;
        while (mb < total && sr.more_rbsp_data()) {
          if (job.is_p) {
            // mb_skip_run: P_Skip macroblocks keep the co-located
            // reference pixels (the recon planes persist across AUs)
            uint32_t run = sr.ue();
            if (sr.overrun) break;
            for (; run > 0 && mb < total; --run) {
              ++counts[si];
              ++mb;
            }
            if (mb >= total || !sr.more_rbsp_data()) break;
          }
          int rc = decode_mb(sr, mb % mbw, mb / mbw, qp, job.first_mb,
                             job.is_p);
          if (rc < 0) {
            rcs[si] = rc;
            return;
          }
          ++counts[si];
          ++mb;
        }
      };
      if (slices.size() > 1) {
        if (!dpool || dpool->size() < (int)slices.size())
          dpool.reset(new WorkPool((int)slices.size()));
        dpool->run((int)slices.size(),
                   [&](int si) { decode_slice((size_t)si); });
      } else {
        decode_slice(0);
      }
      int covered = 0;
      for (size_t si = 0; si < slices.size(); ++si) {
        if (rcs[si] < 0) return rcs[si];
        covered += counts[si];
      }
      if (covered < total) return -9;  // frame not fully covered
      if (!slices[0].is_p) have_idr = true;
      got_frame = true;
    }
    if (!got_frame) return 0;
    const int w = sps.w(), h = sps.h();
    if (cap < w * h * 3) return -8;
    yuv420_to_rgb(rY.data(), rCb.data(), rCr.data(), sps.pw, w, h, rgb);
    *ow = w;
    *oh = h;
    return 1;
  }
};

}  // namespace h264sw

// ---------------------------------------------------------------------------
// C ABI
// ---------------------------------------------------------------------------
extern "C" {

void* airtc_h264enc_create(int w, int h) {
  if (w < 16 || h < 16 || w > 8192 || h > 8192) return nullptr;
  return new h264sw::Encoder(w, h);
}
// number of slices (MB-row bands) encoded in parallel threads
void airtc_h264enc_set_slices(void* h, int n) {
  ((h264sw::Encoder*)h)->n_slices = n < 1 ? 1 : n;
}
// macroblock mode: 0 = I_16x16 (default), 1 = I_4x4 DC (decoder-test aid)
void airtc_h264enc_set_mb_mode(void* h, int m) {
  ((h264sw::Encoder*)h)->mb_mode = m == 1 ? 1 : 0;
}
void airtc_h264enc_destroy(void* h) { delete (h264sw::Encoder*)h; }
int airtc_h264enc_encode(void* h, const uint8_t* rgb, int qp, uint8_t* out,
                         int cap) {
  return ((h264sw::Encoder*)h)->encode(rgb, qp, out, cap);
}
// keyframe-controlled encode: force_idr=0 emits a P frame (P_Skip +
// intra-refresh) when a reference exists
int airtc_h264enc_encode_ex(void* h, const uint8_t* rgb, int qp,
                            int force_idr, uint8_t* out, int cap) {
  return ((h264sw::Encoder*)h)->encode_ex(rgb, qp, force_idr, out, cap);
}

void* airtc_h264dec_create() { return new h264sw::Decoder(); }
void airtc_h264dec_destroy(void* h) { delete (h264sw::Decoder*)h; }
// display dims from the last parsed SPS (0 until one arrives)
void airtc_h264dec_dims(void* h, int* w, int* out_h) {
  auto* d = (h264sw::Decoder*)h;
  *w = d->sps.valid ? d->sps.w() : 0;
  *out_h = d->sps.valid ? d->sps.h() : 0;
}
int airtc_h264dec_decode(void* h, const uint8_t* data, int len, uint8_t* rgb,
                         int cap, int* w, int* out_h) {
  return ((h264sw::Decoder*)h)->decode_au(data, len, rgb, cap, w, out_h);
}

// test hook: run one 4x4 intra prediction (decoder path) for golden checks
void airtc_h264_pred4(int mode, const uint8_t* t8, const uint8_t* l4,
                      uint8_t tl, int have_t, int have_l, int have_tl,
                      uint8_t* out16) {
  h264sw::pred_luma4(mode, t8, l4, tl, have_t != 0, have_l != 0, have_tl != 0,
                     out16);
}

// sanity: every VLC table must be prefix-free within itself; returns 0 on
// success, a nonzero id of the offending table otherwise (test hook)
int airtc_h264sw_table_check() {
  using namespace h264sw;
  auto prefix_free = [](const std::vector<std::pair<int, int>>& codes) {
    for (size_t i = 0; i < codes.size(); ++i)
      for (size_t j = 0; j < codes.size(); ++j) {
        if (i == j) continue;
        auto [li, bi] = codes[i];
        auto [lj, bj] = codes[j];
        if (li <= lj && (bj >> (lj - li)) == bi) return false;
      }
    return true;
  };
  // strict check over the encoder-used region (TotalCoeff <= 9; see the
  // ENC guard in encode_mb) plus a full-table pass — the deep rows are
  // decode-side best-effort
  for (int b = 0; b < 3; ++b) {
    std::vector<std::pair<int, int>> codes;
    for (int c = 0; c <= 16; ++c)
      for (int t = 0; t < 4; ++t)
        if (CT_LEN[b][c][t]) codes.push_back({CT_LEN[b][c][t], CT_BITS[b][c][t]});
    if (!prefix_free(codes)) return 100 + b;
  }
  {
    std::vector<std::pair<int, int>> codes;
    for (int c = 0; c <= 4; ++c)
      for (int t = 0; t < 4; ++t)
        if (CDC_CT_LEN[c][t]) codes.push_back({CDC_CT_LEN[c][t], CDC_CT_BITS[c][t]});
    if (!prefix_free(codes)) return 200;
  }
  for (int tc = 1; tc <= 15; ++tc) {
    std::vector<std::pair<int, int>> codes;
    for (int z = 0; z <= 16 - tc; ++z)
      codes.push_back({TZ_LEN[tc - 1][z], TZ_BITS[tc - 1][z]});
    if (!prefix_free(codes)) return 300 + tc;
  }
  for (int tc = 1; tc <= 3; ++tc) {
    std::vector<std::pair<int, int>> codes;
    for (int z = 0; z <= 4 - tc; ++z)
      codes.push_back({CDC_TZ_LEN[tc - 1][z], CDC_TZ_BITS[tc - 1][z]});
    if (!prefix_free(codes)) return 400 + tc;
  }
  for (int zl = 1; zl <= 7; ++zl) {
    std::vector<std::pair<int, int>> codes;
    int count = zl == 7 ? 15 : zl + 1;
    for (int rb = 0; rb < count; ++rb)
      codes.push_back({RB_LEN[zl - 1][rb], RB_BITS[zl - 1][rb]});
    if (!prefix_free(codes)) return 500 + zl;
  }
  return 0;
}

}  // extern "C"
