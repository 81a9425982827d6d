// Compression codecs: zstd via libzstd.so.1 (prototypes declared here —
// the image has the runtime lib but no dev header), snappy decode
// implemented from the format spec (the reference's default codec is
// zstd(1), writer/mod.rs:224-245; snappy read support is for foreign files).
#pragma once

#include <cstddef>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

extern "C" {
size_t ZSTD_compress(void* dst, size_t dstCapacity, const void* src,
                     size_t srcSize, int compressionLevel);
size_t ZSTD_decompress(void* dst, size_t dstCapacity, const void* src,
                       size_t compressedSize);
size_t ZSTD_compressBound(size_t srcSize);
unsigned ZSTD_isError(size_t code);
}

namespace lakesoul {

inline std::vector<uint8_t> zstd_compress(const uint8_t* src, size_t n, int level) {
  std::vector<uint8_t> out(ZSTD_compressBound(n));
  size_t r = ZSTD_compress(out.data(), out.size(), src, n, level);
  if (ZSTD_isError(r)) throw std::runtime_error("zstd compress failed");
  out.resize(r);
  return out;
}

inline void zstd_decompress_into(uint8_t* dst, size_t dst_n, const uint8_t* src,
                                 size_t src_n) {
  size_t r = ZSTD_decompress(dst, dst_n, src, src_n);
  if (ZSTD_isError(r) || r != dst_n)
    throw std::runtime_error("zstd decompress failed");
}

// -- snappy raw-format decompressor (decode only) ----------------------- //

inline void snappy_decompress_into(uint8_t* dst, size_t dst_n,
                                   const uint8_t* src, size_t src_n) {
  const uint8_t* p = src;
  const uint8_t* end = src + src_n;
  // preamble: uncompressed length varint
  uint64_t ulen = 0;
  int shift = 0;
  while (p < end) {
    uint8_t b = *p++;
    ulen |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  if (ulen != dst_n) throw std::runtime_error("snappy: length mismatch");
  uint8_t* d = dst;
  uint8_t* dend = dst + dst_n;
  while (p < end && d < dend) {
    uint8_t tag = *p++;
    uint32_t kind = tag & 3;
    if (kind == 0) {  // literal
      uint32_t len = (tag >> 2) + 1;
      if (len > 60) {
        uint32_t nb = len - 60;
        len = 0;
        for (uint32_t i = 0; i < nb; i++) len |= (uint32_t)(*p++) << (8 * i);
        len += 1;
      }
      if (p + len > end || d + len > dend) throw std::runtime_error("snappy: literal overrun");
      std::memcpy(d, p, len);
      p += len;
      d += len;
    } else {
      uint32_t len, off;
      if (kind == 1) {
        len = ((tag >> 2) & 7) + 4;
        off = ((uint32_t)(tag >> 5) << 8) | *p++;
      } else if (kind == 2) {
        len = (tag >> 2) + 1;
        off = (uint32_t)p[0] | ((uint32_t)p[1] << 8);
        p += 2;
      } else {
        len = (tag >> 2) + 1;
        off = (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
              ((uint32_t)p[3] << 24);
        p += 4;
      }
      if (off == 0 || (size_t)(d - dst) < off || d + len > dend)
        throw std::runtime_error("snappy: bad copy");
      const uint8_t* s = d - off;
      for (uint32_t i = 0; i < len; i++) d[i] = s[i];  // may overlap
      d += len;
    }
  }
  if (d != dend) throw std::runtime_error("snappy: short output");
}

inline void decompress_into(int codec, uint8_t* dst, size_t dst_n,
                            const uint8_t* src, size_t src_n) {
  switch (codec) {
    case 0:  // UNCOMPRESSED
      if (src_n != dst_n) throw std::runtime_error("uncompressed size mismatch");
      std::memcpy(dst, src, src_n);
      return;
    case 1:  // SNAPPY
      snappy_decompress_into(dst, dst_n, src, src_n);
      return;
    case 6:  // ZSTD
      zstd_decompress_into(dst, dst_n, src, src_n);
      return;
    default:
      throw std::runtime_error("unsupported parquet codec " + std::to_string(codec));
  }
}

}  // namespace lakesoul
