// MI355X (gfx950/CDNA4) kernels for the lakehouse engine — written
// directly in HIP for 64-wide wavefronts; no CUDA-compat paths.
//
// Kernel inventory (SURVEY.md §2.3 mapping):
//  - spark murmur3-32 hashing (bucket scatter)       <- utils/hash/
//  - RLE/bit-packed + dictionary decode              <- parquet decode path
//  - validity scatter (nullable column materialize)
//  - pairwise merge-path sorted merge + dedup        <- MOR SortedStreamMerger
//  - segmented merge operators (UseLast via dedup; sum ops)
//  - multi-column gather (payload materialization after key merge)
//  - string gather / CDC filter
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  - memory-bound kernels: grid-stride, <=2048 workgroups, vectorized
//    where layout permits (G11/G13);
//  - merge-path kernel stages key tiles in LDS (G3), 2048-element tiles,
//    256 threads x 8 elements/thread;
//  - all block sizes are multiples of 64 (wave64).

#include <hip/hip_runtime.h>

#include <cstdint>

#include "../cpp/murmur3.h"

namespace lakesoul {

#define LS_THREADS 256
#define LS_MAX_BLOCKS 2048

static inline int ls_blocks(int64_t n, int per_thread = 1) {
  int64_t b = (n + (int64_t)LS_THREADS * per_thread - 1) / ((int64_t)LS_THREADS * per_thread);
  if (b > LS_MAX_BLOCKS) b = LS_MAX_BLOCKS;
  if (b < 1) b = 1;
  return (int)b;
}

// ===================================================================== //
// murmur3 hashing
// ===================================================================== //

// dtype codes: 0=u8/bool, 1=i8, 2=i16, 3=i32, 4=i64, 5=f32, 6=f64
template <int DT>
__global__ void hash_fixed_kernel(const void* __restrict__ data,
                                  const uint8_t* __restrict__ validity,
                                  const int64_t* __restrict__ prev,
                                  int64_t* __restrict__ out, int64_t n,
                                  int first) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t seed = first ? kHashSeed : (uint32_t)prev[i];
    if (validity && !validity[i]) {
      out[i] = first ? 0 : (int64_t)seed;
      continue;
    }
    uint32_t h;
    if constexpr (DT == 0) h = spark_hash_u32(((const uint8_t*)data)[i], seed);
    else if constexpr (DT == 1) h = spark_hash_u32((uint32_t)(int32_t)((const int8_t*)data)[i], seed);
    else if constexpr (DT == 2) h = spark_hash_u32((uint32_t)(int32_t)((const int16_t*)data)[i], seed);
    else if constexpr (DT == 3) h = spark_hash_u32((uint32_t)((const int32_t*)data)[i], seed);
    else if constexpr (DT == 4) h = spark_hash_u64((uint64_t)((const int64_t*)data)[i], seed);
    else if constexpr (DT == 5) h = spark_hash_f32(((const float*)data)[i], seed);
    else h = spark_hash_f64(((const double*)data)[i], seed);
    out[i] = (int64_t)h;
  }
}

__global__ void hash_string_kernel(const int32_t* __restrict__ offsets,
                                   const uint8_t* __restrict__ bytes,
                                   const uint8_t* __restrict__ validity,
                                   const int64_t* __restrict__ prev,
                                   int64_t* __restrict__ out, int64_t n,
                                   int first) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t seed = first ? kHashSeed : (uint32_t)prev[i];
    if (validity && !validity[i]) {
      out[i] = first ? 0 : (int64_t)seed;
      continue;
    }
    out[i] = (int64_t)spark_hash_bytes(bytes + offsets[i],
                                       offsets[i + 1] - offsets[i], seed);
  }
}

__global__ void bucket_ids_kernel(const int64_t* __restrict__ hashes,
                                  int32_t* __restrict__ out, int64_t n,
                                  uint32_t nbuckets) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    out[i] = (int32_t)((uint32_t)hashes[i] % nbuckets);
  }
}

// ===================================================================== //
// RLE / bit-packed hybrid expansion (dictionary indices, levels)
// ===================================================================== //

// runs: int64 [nruns][6] =
//   {out_off, n, is_literal, value_or_bitoff, bit_width, dict_elem_bias}
// repeat-run values arrive pre-biased; literal unpacks get bias added here.
// payload must be padded by >=8 bytes past the last literal bit.
__global__ void rle_expand_kernel(const uint8_t* __restrict__ payload,
                                  const int64_t* __restrict__ runs,
                                  int64_t nruns, int32_t* __restrict__ out,
                                  int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    // binary search run containing output index i
    int64_t lo = 0, hi = nruns - 1;
    while (lo < hi) {
      int64_t mid = (lo + hi + 1) >> 1;
      if (runs[mid * 6] <= i) lo = mid;
      else hi = mid - 1;
    }
    const int64_t* r = runs + lo * 6;
    int64_t k = i - r[0];
    if (r[2]) {  // literal bit-packed group
      int bit_width = (int)r[4];
      uint64_t mask = bit_width >= 32 ? 0xFFFFFFFFull : ((1ull << bit_width) - 1);
      uint64_t bitpos = (uint64_t)r[3] + (uint64_t)k * bit_width;
      uint64_t w;
      __builtin_memcpy(&w, payload + (bitpos >> 3), 8);
      out[i] = (int32_t)(((w >> (bitpos & 7)) & mask) + (uint64_t)r[5]);
    } else {
      out[i] = (int32_t)r[3];
    }
  }
}

// dictionary gather + validity scatter fused.
// positions: exclusive-scan of validity (dense index per valid row);
// validity==nullptr means all valid and positions==nullptr.
template <typename T>
__global__ void dict_gather_scatter_kernel(const T* __restrict__ dict,
                                           const int32_t* __restrict__ idx,
                                           const uint8_t* __restrict__ validity,
                                           const int64_t* __restrict__ positions,
                                           T* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (validity) {
      out[i] = validity[i] ? dict[idx[positions[i]]] : (T)0;
    } else {
      out[i] = dict[idx[i]];
    }
  }
}

// dense -> full-length scatter through validity (PLAIN nullable columns)
template <typename T>
__global__ void scatter_valid_kernel(const T* __restrict__ dense,
                                     const uint8_t* __restrict__ validity,
                                     const int64_t* __restrict__ positions,
                                     T* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    out[i] = validity[i] ? dense[positions[i]] : (T)0;
  }
}

// ===================================================================== //
// merge-path pairwise sorted merge (keys u64 + source index u64)
// ===================================================================== //
// A is the OLDER stream: on equal keys A's elements are emitted first so
// the newest row survives dedup-keep-last (UseLast semantics,
// reference sorted_stream_merger.rs / combiner.rs).

__device__ inline int64_t merge_path_search(const uint64_t* A, int64_t nA,
                                            const uint64_t* B, int64_t nB,
                                            int64_t d) {
  int64_t lo = d > nB ? d - nB : 0;
  int64_t hi = d < nA ? d : nA;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (A[mid] <= B[d - mid - 1]) lo = mid + 1;
    else hi = mid;
  }
  return lo;
}

#define MERGE_VT 8
#define MERGE_TILE (LS_THREADS * MERGE_VT)  // 2048

__global__ __launch_bounds__(LS_THREADS) void merge_pairs_kernel(
    const uint64_t* __restrict__ kA, const uint64_t* __restrict__ vA, int64_t nA,
    const uint64_t* __restrict__ kB, const uint64_t* __restrict__ vB, int64_t nB,
    uint64_t* __restrict__ kOut, uint64_t* __restrict__ vOut) {
  __shared__ uint64_t sk[MERGE_TILE + 2];   // A window then B window
  __shared__ int64_t s_bounds[4];

  int64_t total = nA + nB;
  for (int64_t tile = blockIdx.x;; tile += gridDim.x) {
    int64_t d0 = tile * MERGE_TILE;
    if (d0 >= total) break;
    int64_t d1 = d0 + MERGE_TILE;
    if (d1 > total) d1 = total;

    if (threadIdx.x == 0) {
      s_bounds[0] = merge_path_search(kA, nA, kB, nB, d0);
      s_bounds[1] = merge_path_search(kA, nA, kB, nB, d1);
    }
    __syncthreads();
    int64_t a0 = s_bounds[0], a1 = s_bounds[1];
    int64_t b0 = d0 - a0, b1 = d1 - a1;
    int aCount = (int)(a1 - a0);
    int bCount = (int)(b1 - b0);

    // stage key windows in LDS: [0,aCount) = A, [aCount, aCount+bCount) = B
    for (int i = threadIdx.x; i < aCount; i += blockDim.x) sk[i] = kA[a0 + i];
    for (int i = threadIdx.x; i < bCount; i += blockDim.x)
      sk[aCount + i] = kB[b0 + i];
    __syncthreads();

    // each thread merges MERGE_VT outputs starting at its local diagonal
    int local_d = threadIdx.x * MERGE_VT;
    int out_n = (int)(d1 - d0);
    if (local_d < out_n) {
      // local merge-path within LDS
      int lo = local_d > bCount ? local_d - bCount : 0;
      int hi = local_d < aCount ? local_d : aCount;
      while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (sk[mid] <= sk[aCount + local_d - mid - 1]) lo = mid + 1;
        else hi = mid;
      }
      int ai = lo;
      int bi = local_d - lo;
      int64_t base = d0 + local_d;
      int count = out_n - local_d < MERGE_VT ? out_n - local_d : MERGE_VT;
#pragma unroll
      for (int k = 0; k < MERGE_VT; k++) {
        if (k >= count) break;
        bool takeA;
        if (ai >= aCount) takeA = false;
        else if (bi >= bCount) takeA = true;
        else takeA = sk[ai] <= sk[aCount + bi];
        if (takeA) {
          kOut[base + k] = sk[ai];
          vOut[base + k] = vA[a0 + ai];
          ai++;
        } else {
          kOut[base + k] = sk[aCount + bi];
          vOut[base + k] = vB[b0 + bi];
          bi++;
        }
      }
    }
    __syncthreads();
  }
}

// keep-last mask over sorted keys: keep[i] = (i == n-1) || key[i] != key[i+1]
__global__ void keep_last_mask_kernel(const uint64_t* __restrict__ keys,
                                      uint8_t* __restrict__ keep, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    keep[i] = (i == n - 1) || (keys[i] != keys[i + 1]);
  }
}

// group-start mask: start[i] = (i==0) || key[i] != key[i-1]
__global__ void group_start_mask_kernel(const uint64_t* __restrict__ keys,
                                        uint8_t* __restrict__ start, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    start[i] = (i == 0) || (keys[i] != keys[i - 1]);
  }
}

// order-preserving u64 key packing
__global__ void pack_key_i64_kernel(const int64_t* __restrict__ x,
                                    uint64_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    out[i] = (uint64_t)x[i] ^ 0x8000000000000000ull;
  }
}

__global__ void pack_key_2xi32_kernel(const int32_t* __restrict__ hi,
                                      const int32_t* __restrict__ lo,
                                      uint64_t* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint64_t h = (uint32_t)hi[i] ^ 0x80000000u;
    uint64_t l = (uint32_t)lo[i] ^ 0x80000000u;
    out[i] = (h << 32) | l;
  }
}

// ===================================================================== //
// gather kernels (payload materialization after merge)
// ===================================================================== //

// fused multi-column fixed-width gather: up to 16 columns per launch.
// srcs/dsts are device pointer tables; elem sizes per column.
struct GatherTable {
  const void* src[16];
  void* dst[16];
  int esize[16];
  int ncols;
};

__global__ void gather_fixed_multi_kernel(GatherTable tbl,
                                          const int64_t* __restrict__ idx,
                                          int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t s = idx[i];
#pragma unroll 4
    for (int c = 0; c < tbl.ncols; c++) {
      switch (tbl.esize[c]) {
        case 1: ((uint8_t*)tbl.dst[c])[i] = ((const uint8_t*)tbl.src[c])[s]; break;
        case 2: ((uint16_t*)tbl.dst[c])[i] = ((const uint16_t*)tbl.src[c])[s]; break;
        case 4: ((uint32_t*)tbl.dst[c])[i] = ((const uint32_t*)tbl.src[c])[s]; break;
        default: ((uint64_t*)tbl.dst[c])[i] = ((const uint64_t*)tbl.src[c])[s]; break;
      }
    }
  }
}

// string gather: one wave per output row; lanes copy bytes cooperatively.
__global__ void gather_strings_kernel(const uint8_t* __restrict__ src_bytes,
                                      const int64_t* __restrict__ src_offsets,
                                      const int64_t* __restrict__ idx,
                                      const int64_t* __restrict__ dst_offsets,
                                      uint8_t* __restrict__ dst_bytes,
                                      int64_t n) {
  int64_t waves_per_grid = ((int64_t)gridDim.x * blockDim.x) >> 6;
  int64_t wave = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  int lane = threadIdx.x & 63;
  for (int64_t i = wave; i < n; i += waves_per_grid) {
    int64_t s = idx[i];
    int64_t so = src_offsets[s], eo = src_offsets[s + 1];
    int64_t d = dst_offsets[i];
    for (int64_t b = lane; b < eo - so; b += 64) {
      dst_bytes[d + b] = src_bytes[so + b];
    }
  }
}

// CDC filter: mask[i] = (value != pattern)
__global__ void bytes_ne_mask_kernel(const int64_t* __restrict__ offsets,
                                     const uint8_t* __restrict__ bytes,
                                     const uint8_t* __restrict__ pattern,
                                     int plen, uint8_t* __restrict__ out,
                                     int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t so = offsets[i], eo = offsets[i + 1];
    bool eq = (eo - so) == plen;
    if (eq) {
      for (int k = 0; k < plen; k++) {
        if (bytes[so + k] != pattern[k]) {
          eq = false;
          break;
        }
      }
    }
    out[i] = eq ? 0 : 1;
  }
}

// segmented sum over sorted groups: starts-based two-phase not needed —
// use atomics into per-group slot (group ids from scan of start mask).
__device__ inline void ls_atomic_add(float* p, float v) { atomicAdd(p, v); }
__device__ inline void ls_atomic_add(double* p, double v) { atomicAdd(p, v); }
__device__ inline void ls_atomic_add(int32_t* p, int32_t v) { atomicAdd(p, v); }
__device__ inline void ls_atomic_add(int64_t* p, int64_t v) {
  atomicAdd((unsigned long long*)p, (unsigned long long)v);
}

template <typename T, typename ACC>
__global__ void segmented_sum_kernel(const T* __restrict__ vals,
                                     const int64_t* __restrict__ group_of_row,
                                     const uint8_t* __restrict__ contrib,
                                     const uint8_t* __restrict__ validity,
                                     ACC* __restrict__ out_sum,
                                     int32_t* __restrict__ out_has_null,
                                     int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (contrib && !contrib[i]) continue;
    int64_t g = group_of_row[i];
    if (validity && !validity[i]) {
      out_has_null[g] = 1;
      continue;
    }
    ls_atomic_add(&out_sum[g], (ACC)vals[i]);
  }
}

// last contributing (and optionally non-null) row index per group.
// out_idx is zero-initialized and stores (i+1) — unsigned atomicMax can't
// start from -1 (0xFF..F would already be maximal). 0 means "no row".
__global__ void segmented_last_kernel(const int64_t* __restrict__ group_of_row,
                                      const uint8_t* __restrict__ contrib,
                                      const uint8_t* __restrict__ validity,
                                      int64_t* __restrict__ out_idx,  // init 0
                                      int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (contrib && !contrib[i]) continue;
    if (validity && !validity[i]) continue;
    atomicMax((unsigned long long*)&out_idx[group_of_row[i]],
              (unsigned long long)(i + 1));
  }
}

}  // namespace lakesoul

// ===================================================================== //
// host-side launchers (called from hip_module.cc)
// ===================================================================== //

namespace lakesoul {

void launch_hash_fixed(int dt, const void* data, const uint8_t* validity,
                       const int64_t* prev, int64_t* out, int64_t n, int first,
                       hipStream_t stream) {
  dim3 g(ls_blocks(n)), b(LS_THREADS);
  switch (dt) {
    case 0: hipLaunchKernelGGL((hash_fixed_kernel<0>), g, b, 0, stream, data, validity, prev, out, n, first); break;
    case 1: hipLaunchKernelGGL((hash_fixed_kernel<1>), g, b, 0, stream, data, validity, prev, out, n, first); break;
    case 2: hipLaunchKernelGGL((hash_fixed_kernel<2>), g, b, 0, stream, data, validity, prev, out, n, first); break;
    case 3: hipLaunchKernelGGL((hash_fixed_kernel<3>), g, b, 0, stream, data, validity, prev, out, n, first); break;
    case 4: hipLaunchKernelGGL((hash_fixed_kernel<4>), g, b, 0, stream, data, validity, prev, out, n, first); break;
    case 5: hipLaunchKernelGGL((hash_fixed_kernel<5>), g, b, 0, stream, data, validity, prev, out, n, first); break;
    default: hipLaunchKernelGGL((hash_fixed_kernel<6>), g, b, 0, stream, data, validity, prev, out, n, first); break;
  }
}

void launch_hash_string(const int32_t* offsets, const uint8_t* bytes,
                        const uint8_t* validity, const int64_t* prev,
                        int64_t* out, int64_t n, int first, hipStream_t s) {
  hipLaunchKernelGGL(hash_string_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS), 0, s,
                     offsets, bytes, validity, prev, out, n, first);
}

void launch_bucket_ids(const int64_t* hashes, int32_t* out, int64_t n,
                       uint32_t nb, hipStream_t s) {
  hipLaunchKernelGGL(bucket_ids_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS), 0, s,
                     hashes, out, n, nb);
}

void launch_rle_expand(const uint8_t* payload, const int64_t* runs,
                       int64_t nruns, int32_t* out, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(rle_expand_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS), 0, s,
                     payload, runs, nruns, out, n);
}

template <typename T>
void launch_dict_gather_scatter(const T* dict, const int32_t* idx,
                                const uint8_t* validity, const int64_t* pos,
                                T* out, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL((dict_gather_scatter_kernel<T>), dim3(ls_blocks(n)),
                     dim3(LS_THREADS), 0, s, dict, idx, validity, pos, out, n);
}
template void launch_dict_gather_scatter<uint32_t>(const uint32_t*, const int32_t*, const uint8_t*, const int64_t*, uint32_t*, int64_t, hipStream_t);
template void launch_dict_gather_scatter<uint64_t>(const uint64_t*, const int32_t*, const uint8_t*, const int64_t*, uint64_t*, int64_t, hipStream_t);
template void launch_dict_gather_scatter<uint8_t>(const uint8_t*, const int32_t*, const uint8_t*, const int64_t*, uint8_t*, int64_t, hipStream_t);

template <typename T>
void launch_scatter_valid(const T* dense, const uint8_t* validity,
                          const int64_t* pos, T* out, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL((scatter_valid_kernel<T>), dim3(ls_blocks(n)),
                     dim3(LS_THREADS), 0, s, dense, validity, pos, out, n);
}
template void launch_scatter_valid<uint8_t>(const uint8_t*, const uint8_t*, const int64_t*, uint8_t*, int64_t, hipStream_t);
template void launch_scatter_valid<uint16_t>(const uint16_t*, const uint8_t*, const int64_t*, uint16_t*, int64_t, hipStream_t);
template void launch_scatter_valid<uint32_t>(const uint32_t*, const uint8_t*, const int64_t*, uint32_t*, int64_t, hipStream_t);
template void launch_scatter_valid<uint64_t>(const uint64_t*, const uint8_t*, const int64_t*, uint64_t*, int64_t, hipStream_t);

void launch_merge_pairs(const uint64_t* kA, const uint64_t* vA, int64_t nA,
                        const uint64_t* kB, const uint64_t* vB, int64_t nB,
                        uint64_t* kOut, uint64_t* vOut, hipStream_t s) {
  int64_t ntiles = (nA + nB + MERGE_TILE - 1) / MERGE_TILE;
  int blocks = ntiles > LS_MAX_BLOCKS ? LS_MAX_BLOCKS : (int)ntiles;
  hipLaunchKernelGGL(merge_pairs_kernel, dim3(blocks), dim3(LS_THREADS), 0, s,
                     kA, vA, nA, kB, vB, nB, kOut, vOut);
}

void launch_keep_last(const uint64_t* keys, uint8_t* keep, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(keep_last_mask_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS), 0, s,
                     keys, keep, n);
}

void launch_group_start(const uint64_t* keys, uint8_t* start, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(group_start_mask_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS), 0, s,
                     keys, start, n);
}

void launch_pack_key_i64(const int64_t* x, uint64_t* out, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(pack_key_i64_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS), 0, s,
                     x, out, n);
}

void launch_pack_key_2xi32(const int32_t* hi, const int32_t* lo, uint64_t* out,
                           int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(pack_key_2xi32_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS), 0, s,
                     hi, lo, out, n);
}

void launch_gather_fixed_multi(const GatherTable& tbl, const int64_t* idx,
                               int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(gather_fixed_multi_kernel, dim3(ls_blocks(n)),
                     dim3(LS_THREADS), 0, s, tbl, idx, n);
}

void launch_gather_strings(const uint8_t* src_bytes, const int64_t* src_off,
                           const int64_t* idx, const int64_t* dst_off,
                           uint8_t* dst_bytes, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(gather_strings_kernel, dim3(ls_blocks(n, 1)),
                     dim3(LS_THREADS), 0, s, src_bytes, src_off, idx, dst_off,
                     dst_bytes, n);
}

void launch_bytes_ne_mask(const int64_t* offsets, const uint8_t* bytes,
                          const uint8_t* pattern, int plen, uint8_t* out,
                          int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(bytes_ne_mask_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS),
                     0, s, offsets, bytes, pattern, plen, out, n);
}

template <typename T, typename ACC>
void launch_segmented_sum(const T* vals, const int64_t* grp, const uint8_t* contrib,
                          const uint8_t* validity, ACC* out_sum,
                          int32_t* out_has_null, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL((segmented_sum_kernel<T, ACC>), dim3(ls_blocks(n)),
                     dim3(LS_THREADS), 0, s, vals, grp, contrib, validity,
                     out_sum, out_has_null, n);
}
template void launch_segmented_sum<float, float>(const float*, const int64_t*, const uint8_t*, const uint8_t*, float*, int32_t*, int64_t, hipStream_t);
template void launch_segmented_sum<double, double>(const double*, const int64_t*, const uint8_t*, const uint8_t*, double*, int32_t*, int64_t, hipStream_t);
template void launch_segmented_sum<int32_t, int32_t>(const int32_t*, const int64_t*, const uint8_t*, const uint8_t*, int32_t*, int32_t*, int64_t, hipStream_t);
template void launch_segmented_sum<int64_t, int64_t>(const int64_t*, const int64_t*, const uint8_t*, const uint8_t*, int64_t*, int32_t*, int64_t, hipStream_t);

void launch_segmented_last(const int64_t* grp, const uint8_t* contrib,
                           const uint8_t* validity, int64_t* out_idx, int64_t n,
                           hipStream_t s) {
  hipLaunchKernelGGL(segmented_last_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS),
                     0, s, grp, contrib, validity, out_idx, n);
}

}  // namespace lakesoul
namespace lakesoul {

// ------------------------------------------------------------------ //
// binary-code hamming scorer (vector index first pass — the RaBitQ
// 1-bit idea, lakesoul-vector quantizer.rs): codes are sign bits of
// rotated vectors packed into u64 words. Each thread owns one database
// row, keeps its W<=16 words in registers, loops the query block —
// codes read once from HBM per launch.
// ------------------------------------------------------------------ //

__global__ void hamming_scores_kernel(const uint64_t* __restrict__ codes,
                                      const uint64_t* __restrict__ qcodes,
                                      int32_t* __restrict__ out, int64_t n,
                                      int nq, int words) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  uint64_t c[16];
  for (; i < n; i += stride) {
    for (int w = 0; w < words; w++) c[w] = codes[i * words + w];
    for (int q = 0; q < nq; q++) {
      int d = 0;
      for (int w = 0; w < words; w++)
        d += __popcll(c[w] ^ qcodes[q * words + w]);
      out[i * nq + q] = d;
    }
  }
}

// 8-byte big-endian chunk key of each string (zero-padded past the
// end), sign-flipped so int64 ascending sort = unsigned byte order.
// LSD passes over chunks + a length pass give full lexicographic order
// on the GPU (string-PK merge; reference cursors compare byte-wise,
// sorted/cursor.rs).
__global__ void str_chunk_keys_kernel(const int64_t* __restrict__ offsets,
                                      const uint8_t* __restrict__ bytes,
                                      int64_t chunk, int64_t* __restrict__ out,
                                      int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t a = offsets[i], b = offsets[i + 1];
    int64_t start = a + chunk * 8;
    uint64_t key = 0;
    for (int k = 0; k < 8; k++) {
      uint64_t v = (start + k < b) ? bytes[start + k] : 0;
      key = (key << 8) | v;
    }
    out[i] = (int64_t)(key ^ 0x8000000000000000ull);
  }
}

void launch_str_chunk_keys(const int64_t* offsets, const uint8_t* bytes,
                           int64_t chunk, int64_t* out, int64_t n,
                           hipStream_t s) {
  hipLaunchKernelGGL(str_chunk_keys_kernel, dim3(ls_blocks(n)),
                     dim3(LS_THREADS), 0, s, offsets, bytes, chunk, out, n);
}

void launch_hamming_scores(const uint64_t* codes, const uint64_t* qcodes,
                           int32_t* out, int64_t n, int nq, int words,
                           hipStream_t s) {
  hipLaunchKernelGGL(hamming_scores_kernel, dim3(ls_blocks(n)), dim3(LS_THREADS),
                     0, s, codes, qcodes, out, n, nq, words);
}

}  // namespace lakesoul
