// Shared H.264 bitstream primitives: exp-Golomb bit I/O and NAL
// (de-)escaping. Used by the SPS/PPS generator (vcn.cpp) and the software
// baseline-intra codec (h264sw.cpp).
//
// Replaces reference dependency plumbing: the reference gets its bitstream
// layer from PyNvVideoCodec/x264 inside the aiortc fork
// (reference requirements.txt:12-13); here it is first-party.
#pragma once

#include <stdint.h>

#include <vector>

namespace h264 {

struct BitWriter {
  std::vector<uint8_t> bytes;
  uint32_t cur = 0;
  int nbits = 0;

  void put(uint32_t value, int width) {
    // bulk bit packing (bit-exact with the bit-at-a-time form): fill the
    // current byte, then whole bytes, then the remainder
    while (width > 0) {
      const int take = (8 - nbits) < width ? (8 - nbits) : width;
      const uint32_t mask = (take == 32) ? 0xFFFFFFFFu : ((1u << take) - 1u);
      cur = (cur << take) | ((value >> (width - take)) & mask);
      nbits += take;
      width -= take;
      if (nbits == 8) {
        bytes.push_back((uint8_t)cur);
        cur = 0;
        nbits = 0;
      }
    }
  }
  // raw byte run; caller must be byte-aligned (nbits == 0), e.g. right
  // after align_byte() for I_PCM samples
  void put_aligned_bytes(const uint8_t* p, size_t n) {
    bytes.insert(bytes.end(), p, p + n);
  }
  void ue(uint32_t v) {  // unsigned exp-Golomb
    uint32_t vp1 = v + 1;
    int lead = 0;
    for (uint32_t t = vp1; t > 1; t >>= 1) ++lead;
    put(0, lead);
    put(vp1, lead + 1);
  }
  void se(int32_t v) {  // signed exp-Golomb
    ue(v <= 0 ? (uint32_t)(-2 * v) : (uint32_t)(2 * v - 1));
  }
  void align_byte() {  // pcm_alignment_zero_bit*
    while (nbits) put(0, 1);
  }
  void rbsp_trailing() {
    put(1, 1);
    if (nbits) put(0, 8 - nbits);
  }
  size_t bitpos() const { return bytes.size() * 8 + nbits; }
};

// RBSP -> NAL with emulation prevention (00 00 {00,01,02,03} -> 00 00 03 xx)
inline void emit_nal(std::vector<uint8_t>* out, uint8_t nal_header,
                     const std::vector<uint8_t>& rbsp) {
  out->insert(out->end(), {0, 0, 0, 1, nal_header});
  int zeros = 0;
  for (uint8_t b : rbsp) {
    if (zeros >= 2 && b <= 3) {
      out->push_back(3);
      zeros = 0;
    }
    out->push_back(b);
    zeros = (b == 0) ? zeros + 1 : 0;
  }
}

// NAL payload -> RBSP (strip emulation-prevention 03 bytes)
inline std::vector<uint8_t> unescape_rbsp(const uint8_t* p, size_t n) {
  std::vector<uint8_t> out;
  out.reserve(n);
  int zeros = 0;
  for (size_t i = 0; i < n; ++i) {
    if (zeros >= 2 && p[i] == 3 && i + 1 < n && p[i + 1] <= 3) {
      zeros = 0;
      continue;  // drop the escape byte
    }
    out.push_back(p[i]);
    zeros = (p[i] == 0) ? zeros + 1 : 0;
  }
  return out;
}

struct BitReader {
  const uint8_t* p;
  size_t n;     // bytes
  size_t pos = 0;  // bit position
  bool overrun = false;

  BitReader(const uint8_t* data, size_t nbytes) : p(data), n(nbytes) {}

  uint32_t u(int width) {
    uint32_t v = 0;
    for (int i = 0; i < width; ++i) {
      if (pos >= n * 8) {
        overrun = true;
        return v << (width - i);
      }
      v = (v << 1) | ((p[pos >> 3] >> (7 - (pos & 7))) & 1);
      ++pos;
    }
    return v;
  }
  uint32_t ue() {
    int lead = 0;
    while (pos < n * 8 && u(1) == 0) {
      if (++lead > 31) {
        overrun = true;
        return 0;
      }
    }
    if (lead == 0) return 0;
    return ((1u << lead) | u(lead)) - 1;
  }
  int32_t se() {
    uint32_t k = ue();
    return (k & 1) ? (int32_t)((k + 1) >> 1) : -(int32_t)(k >> 1);
  }
  void align_byte() { pos = (pos + 7) & ~(size_t)7; }
  bool more_rbsp_data() const {
    if (pos >= n * 8) return false;
    // find last set bit (rbsp_stop_one_bit) in the payload
    size_t last = n * 8;
    while (last > 0) {
      --last;
      if ((p[last >> 3] >> (7 - (last & 7))) & 1) break;
    }
    return pos < last;
  }
};

// Split an Annex-B byte stream into NAL units (payload excludes start code).
struct NalView {
  const uint8_t* data;
  size_t size;
};
inline std::vector<NalView> split_annexb(const uint8_t* p, size_t n) {
  std::vector<NalView> nals;
  size_t i = 0;
  size_t start = (size_t)-1;
  int zeros = 0;
  for (; i < n; ++i) {
    if (p[i] == 1 && zeros >= 2) {
      if (start != (size_t)-1) {
        size_t end = i - (zeros > 2 ? 3 : 2);
        nals.push_back({p + start, end - start});
      }
      start = i + 1;
      zeros = 0;
      continue;
    }
    zeros = (p[i] == 0) ? zeros + 1 : 0;
  }
  if (start != (size_t)-1 && start < n) nals.push_back({p + start, n - start});
  return nals;
}

}  // namespace h264
