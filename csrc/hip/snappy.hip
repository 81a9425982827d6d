// Snappy raw-format decompression on GPU — one wave (64 lanes) per page.
//
// Page-level parallelism comes from the scan shape (many pages per scan
// unit decompress concurrently); within a page, lane 0 walks the tag
// stream (inherently serial) and broadcasts each element's geometry so
// all 64 lanes do the byte movement:
//  - literals: cooperative 64-lane copy;
//  - copies with offset >= 1: pattern fill dst[d+i] = dst[d-off + i%off]
//    — every source byte is before the element's start, so lanes can
//    write in any order.
// This is the decompress-side analog of the reference's page decode path
// (its snappy lives in the arrow-rs parquet dependency).

#include <hip/hip_runtime.h>

#include <cstdint>

namespace lakesoul {

// jobs: int64 [n][4] = {src_off, src_len, dst_off, dst_len}
// status[i]: 0 ok, nonzero error
__global__ __launch_bounds__(256) void snappy_decompress_kernel(
    const uint8_t* __restrict__ src_buf, const int64_t* __restrict__ jobs,
    int64_t njobs, uint8_t* __restrict__ dst_buf, int32_t* __restrict__ status) {
  int64_t wave = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int64_t nwaves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  int lane = threadIdx.x & 63;

  for (int64_t j = wave; j < njobs; j += nwaves) {
    const int64_t* job = jobs + j * 4;
    const uint8_t* src = src_buf + job[0];
    int64_t src_len = job[1];
    uint8_t* dst = dst_buf + job[2];
    int64_t dst_len = job[3];

    int64_t s = 0;  // lane-uniform: every lane executes the same control flow
    int64_t d = 0;
    int err = 0;

    // preamble: uncompressed length varint
    {
      uint64_t ulen = 0;
      int shift = 0;
      while (s < src_len) {
        uint8_t b = src[s++];
        ulen |= (uint64_t)(b & 0x7F) << shift;
        if (!(b & 0x80)) break;
        shift += 7;
      }
      if ((int64_t)ulen != dst_len) err = 1;
    }

    while (!err && s < src_len && d < dst_len) {
      uint8_t tag = src[s++];
      uint32_t kind = tag & 3;
      if (kind == 0) {  // literal
        uint32_t len = (tag >> 2) + 1;
        if (len > 60) {
          uint32_t nb = len - 60;
          len = 0;
          for (uint32_t i = 0; i < nb && s < src_len; i++)
            len |= (uint32_t)src[s++] << (8 * i);
          len += 1;
        }
        if (s + len > src_len || d + len > dst_len) {
          err = 2;
          break;
        }
        for (uint32_t i = lane; i < len; i += 64) dst[d + i] = src[s + i];
        s += len;
        d += len;
      } else {
        uint32_t len, off;
        if (kind == 1) {
          len = ((tag >> 2) & 7) + 4;
          off = ((uint32_t)(tag >> 5) << 8) | src[s];
          s += 1;
        } else if (kind == 2) {
          len = (tag >> 2) + 1;
          off = (uint32_t)src[s] | ((uint32_t)src[s + 1] << 8);
          s += 2;
        } else {
          len = (tag >> 2) + 1;
          off = (uint32_t)src[s] | ((uint32_t)src[s + 1] << 8) |
                ((uint32_t)src[s + 2] << 16) | ((uint32_t)src[s + 3] << 24);
          s += 4;
        }
        if (off == 0 || off > d || d + len > dst_len) {
          err = 3;
          break;
        }
        // copy-source bytes were written by OTHER lanes of this wave in
        // earlier elements: go through volatile (L1-bypassing, coherent)
        // accesses so no lane reads a stale L1 line
        volatile const uint8_t* from = dst + d - off;
        for (uint32_t i = lane; i < len; i += 64) dst[d + i] = from[i % off];
        d += len;
      }
      // keep prior element's stores ordered before the next element
      __builtin_amdgcn_s_waitcnt(0);
    }
    if (!err && d != dst_len) err = 4;
    if (lane == 0) status[j] = err;
  }
}

void launch_snappy_decompress(const uint8_t* src, const int64_t* jobs,
                              int64_t njobs, uint8_t* dst, int32_t* status,
                              hipStream_t stream) {
  int64_t waves_needed = njobs;
  int64_t blocks = (waves_needed * 64 + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(snappy_decompress_kernel, dim3((uint32_t)blocks), dim3(256),
                     0, stream, src, jobs, njobs, dst, status);
}

}  // namespace lakesoul
