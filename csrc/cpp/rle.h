// Parquet RLE/bit-packed hybrid codec (host side).
// Decode is used for definition levels and as the CPU fallback for
// dictionary indices; the GPU path expands runs with the HIP kernel in
// csrc/hip/decode.hip fed by parse_rle_runs() descriptors.
#pragma once

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <vector>

namespace lakesoul {

struct RleRun {
  int64_t out_off;     // first output index
  int64_t n;           // number of values
  int32_t is_literal;  // 1 = bit-packed group, 0 = repeat run
  uint32_t value;      // repeat value (if !is_literal)
  int64_t bit_off;     // absolute bit offset of packed data (if literal)
};

// Walk the hybrid stream, producing run descriptors. `base_bit` is the bit
// offset of `p` within the buffer the GPU kernel will see.
inline void parse_rle_runs(const uint8_t* p, size_t len, int bit_width,
                           int64_t num_values, int64_t base_bit,
                           std::vector<RleRun>& out) {
  const uint8_t* start = p;
  const uint8_t* end = p + len;
  int64_t produced = 0;
  int byte_width = (bit_width + 7) / 8;
  while (produced < num_values && p < end) {
    // varint header
    uint64_t h = 0;
    int shift = 0;
    while (p < end) {
      uint8_t b = *p++;
      h |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    if (h & 1) {
      int64_t groups = (int64_t)(h >> 1);
      int64_t n = groups * 8;
      if (n > num_values - produced) n = num_values - produced;
      out.push_back(RleRun{produced, n, 1, 0,
                           base_bit + (int64_t)(p - start) * 8});
      p += groups * bit_width;  // bit_width bytes per 8-value group
      produced += n;
    } else {
      int64_t n = (int64_t)(h >> 1);
      uint32_t v = 0;
      for (int i = 0; i < byte_width && p < end; i++) v |= (uint32_t)(*p++) << (8 * i);
      if (n > num_values - produced) n = num_values - produced;
      out.push_back(RleRun{produced, n, 0, v, 0});
      produced += n;
    }
  }
  if (produced < num_values)
    throw std::runtime_error("rle: stream exhausted early");
}

// Scalar decode (host fallback / def levels).
template <typename T>
inline void rle_decode(const uint8_t* p, size_t len, int bit_width,
                       int64_t num_values, T* out) {
  const uint8_t* end = p + len;
  int64_t produced = 0;
  int byte_width = (bit_width + 7) / 8;
  while (produced < num_values && p < end) {
    uint64_t h = 0;
    int shift = 0;
    while (p < end) {
      uint8_t b = *p++;
      h |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    if (h & 1) {
      int64_t groups = (int64_t)(h >> 1);
      int64_t navail = groups * 8;
      int64_t n = navail < num_values - produced ? navail : num_values - produced;
      // unpack LSB-first bit_width-bit values
      uint64_t bitpos = 0;
      for (int64_t i = 0; i < n; i++) {
        uint64_t byte_idx = bitpos >> 3;
        uint32_t bit_idx = (uint32_t)(bitpos & 7);
        uint64_t window = 0;
        // load up to 8 bytes
        size_t avail = (size_t)(end - p) - byte_idx < 8 ? (size_t)(end - p) - byte_idx : 8;
        std::memcpy(&window, p + byte_idx, avail);
        out[produced + i] = (T)((window >> bit_idx) & ((bit_width == 32) ? 0xFFFFFFFFull : ((1ull << bit_width) - 1)));
        bitpos += bit_width;
      }
      p += groups * bit_width;
      produced += n;
    } else {
      int64_t n = (int64_t)(h >> 1);
      uint32_t v = 0;
      for (int i = 0; i < byte_width && p < end; i++) v |= (uint32_t)(*p++) << (8 * i);
      if (n > num_values - produced) n = num_values - produced;
      for (int64_t i = 0; i < n; i++) out[produced + i] = (T)v;
      produced += n;
    }
  }
  if (produced < num_values) throw std::runtime_error("rle: stream exhausted early");
}

// Encode a validity mask (0/1 per value) as RLE hybrid with bit_width=1.
// All-valid masks collapse to one repeat run.
inline std::vector<uint8_t> encode_def_levels(const uint8_t* validity, int64_t n) {
  std::vector<uint8_t> out;
  auto put_varint = [&](uint64_t v) {
    while (v >= 0x80) {
      out.push_back((uint8_t)(v | 0x80));
      v >>= 7;
    }
    out.push_back((uint8_t)v);
  };
  if (validity == nullptr) {
    put_varint(((uint64_t)n << 1));
    out.push_back(1);
    return out;
  }
  // RLE-encode runs of equal values; fall back to bit-packed groups for
  // short alternating stretches. Simple approach: emit repeat runs when a
  // value repeats >= 16, else accumulate into bit-packed groups of 8.
  int64_t i = 0;
  while (i < n) {
    int64_t j = i;
    while (j < n && validity[j] == validity[i]) j++;
    int64_t run = j - i;
    if (run >= 16 && (i % 8) == 0) {
      put_varint(((uint64_t)run << 1));
      out.push_back(validity[i]);
      i = j;
    } else {
      // bit-packed group(s) covering at least 8 values (pad with zeros)
      int64_t take = run < 8 ? 8 : (run / 8) * 8;
      if (i + take > n) take = ((n - i) + 7) / 8 * 8;  // padded final group
      int64_t groups = take / 8;
      put_varint(((uint64_t)groups << 1) | 1);
      for (int64_t g = 0; g < groups; g++) {
        uint8_t b = 0;
        for (int k = 0; k < 8; k++) {
          int64_t idx = i + g * 8 + k;
          if (idx < n && validity[idx]) b |= (uint8_t)(1 << k);
        }
        out.push_back(b);
      }
      i += take;
    }
  }
  return out;
}

// Encode arbitrary small-int levels (values < 256) as pure RLE repeat
// runs — always a legal RLE/bit-packed hybrid stream regardless of the
// decoder's bit width (repeat-run values are stored byte-wide for
// bit_width <= 8). Used for LIST rep/def levels.
inline std::vector<uint8_t> encode_levels(const uint8_t* levels, int64_t n) {
  std::vector<uint8_t> out;
  auto put_varint = [&](uint64_t v) {
    while (v >= 0x80) {
      out.push_back((uint8_t)(v | 0x80));
      v >>= 7;
    }
    out.push_back((uint8_t)v);
  };
  int64_t i = 0;
  while (i < n) {
    int64_t j = i;
    while (j < n && levels[j] == levels[i]) j++;
    put_varint(((uint64_t)(j - i)) << 1);
    out.push_back(levels[i]);
    i = j;
  }
  return out;
}

}  // namespace lakesoul
