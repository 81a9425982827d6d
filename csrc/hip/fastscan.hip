// RaBitQ FastScan on gfx950: LDS-staged 16-entry lookup tables score the
// 1-bit codes of a cluster against a query — the MI355X-native
// replacement for the reference's AVX2 accumulate_batch
// (rust/lakesoul-vector/src/rabitq/simd.rs:1012-1059, packed 4-bit LUT
// gather over 32-vector batches).
//
// Design (cdna_hip_programming.md): the query's per-4-dim-group LUT
// (g groups x 16 entries, f32) is built once on device with tensor ops
// and staged in LDS by each workgroup (12 KB for 768-d — >10 workgroups
// per CU of occupancy headroom against 160 KB LDS). Each thread owns one
// vector: its packed sign bits stream from HBM once (w = dim/8 bytes per
// vector — 16x less traffic than the bf16 exact path), every byte costs
// two LDS lookups. Memory-bound by construction; the LDS lookups ride
// under the HBM latency.
//
// Bit layout matches lakesoul_amd/vector/rabitq.py pack_bits: byte b of
// a row covers dims 8b..8b+7 LSB-first, so the low nibble is group 2b
// and the high nibble group 2b+1.

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

namespace lakesoul {

// bits: [m][w] uint8; lut: [nq][g*16] f32; out: [m][nq] f32
__global__ __launch_bounds__(256) void fastscan_lut_kernel(
    const uint8_t* __restrict__ bits, const float* __restrict__ lut,
    float* __restrict__ out, int64_t m, int32_t nq, int32_t w, int32_t g) {
  extern __shared__ float slut[];  // g*16 floats
  int q = (int)blockIdx.y;
  const float* lq = lut + (int64_t)q * g * 16;
  for (int i = (int)threadIdx.x; i < g * 16; i += (int)blockDim.x)
    slut[i] = lq[i];
  __syncthreads();
  int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= m) return;
  const uint8_t* bp = bits + row * (int64_t)w;
  float acc = 0.f;
  int b = 0;
  // 4-byte chunks through a single dword load
  for (; b + 4 <= w; b += 4) {
    uint32_t v4;
    __builtin_memcpy(&v4, bp + b, 4);
#pragma unroll
    for (int j = 0; j < 4; j++) {
      uint32_t byte = (v4 >> (8 * j)) & 0xFF;
      acc += slut[(2 * (b + j)) * 16 + (byte & 0xF)];
      int g2 = 2 * (b + j) + 1;
      if (g2 < g) acc += slut[g2 * 16 + (byte >> 4)];
    }
  }
  for (; b < w; b++) {
    uint32_t byte = bp[b];
    acc += slut[(2 * b) * 16 + (byte & 0xF)];
    int g2 = 2 * b + 1;
    if (g2 < g) acc += slut[g2 * 16 + (byte >> 4)];
  }
  out[row * (int64_t)nq + q] = acc;
}

void launch_fastscan_lut(const uint8_t* bits, const float* lut,
                                    float* out, int64_t m, int32_t nq,
                                    int32_t w, int32_t g, hipStream_t s) {
  dim3 grid((uint32_t)((m + 255) / 256), (uint32_t)nq);
  size_t lds = (size_t)g * 16 * sizeof(float);
  hipLaunchKernelGGL(fastscan_lut_kernel, grid, dim3(256), lds, s, bits, lut,
                     out, m, nq, w, g);
}

// Fused stage-1 estimator: the LUT accumulate PLUS the RaBitQ
// correction factors applied in-register, writing the estimated
// distance directly (est[q][row] = f_add[row] + g_add[q][cl(row)] +
// f_rescale[row] * (acc + c1_sum_q[q])) — saves two full (nq x n) f32
// round trips through HBM vs computing est from a raw ip tensor.
// Output is (nq, n) row-major so the per-query top-C reads contiguously.
__global__ __launch_bounds__(256) void fastscan_est_kernel(
    const uint8_t* __restrict__ bits, const float* __restrict__ lut,
    const float* __restrict__ f_add, const float* __restrict__ f_rescale,
    const int32_t* __restrict__ cl_of_row, const float* __restrict__ g_add,
    const float* __restrict__ c1_sum_q, float* __restrict__ out, int64_t m,
    int32_t nq, int32_t w, int32_t g, int32_t n_clusters) {
  extern __shared__ float slut[];  // g*16 floats
  int q = (int)blockIdx.y;
  const float* lq = lut + (int64_t)q * g * 16;
  for (int i = (int)threadIdx.x; i < g * 16; i += (int)blockDim.x)
    slut[i] = lq[i];
  __syncthreads();
  int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= m) return;
  const uint8_t* bp = bits + row * (int64_t)w;
  float acc = 0.f;
  int b = 0;
  for (; b + 4 <= w; b += 4) {
    uint32_t v4;
    __builtin_memcpy(&v4, bp + b, 4);
#pragma unroll
    for (int j = 0; j < 4; j++) {
      uint32_t byte = (v4 >> (8 * j)) & 0xFF;
      acc += slut[(2 * (b + j)) * 16 + (byte & 0xF)];
      int g2 = 2 * (b + j) + 1;
      if (g2 < g) acc += slut[g2 * 16 + (byte >> 4)];
    }
  }
  for (; b < w; b++) {
    uint32_t byte = bp[b];
    acc += slut[(2 * b) * 16 + (byte & 0xF)];
    int g2 = 2 * b + 1;
    if (g2 < g) acc += slut[g2 * 16 + (byte >> 4)];
  }
  float ga = g_add[(int64_t)q * n_clusters + cl_of_row[row]];
  out[(int64_t)q * m + row] =
      f_add[row] + ga + f_rescale[row] * (acc + c1_sum_q[q]);
}

// Query-blocked estimator: the per-query kernel above re-reads every
// row's packed bits nq times (30 GB of HBM traffic for 64 queries over
// 5M x 768 — measured 1.4 TB/s effective but 25 ms/batch). This
// variant loads each row's bits ONCE per 8 queries: 8 LUTs stage in
// LDS (8 x g x 16 f32 = 98 KB at 768-d, inside the 160 KB CU budget),
// each thread keeps 8 accumulators in registers. HBM traffic drops
// ~8x; the extra LDS reads (16/byte) ride far under LDS bandwidth.
template <int FS_QB>
__global__ __launch_bounds__(256) void fastscan_est_qb_kernel(
    const uint8_t* __restrict__ bits, const float* __restrict__ lut,
    const float* __restrict__ f_add, const float* __restrict__ f_rescale,
    const int32_t* __restrict__ cl_of_row, const float* __restrict__ g_add,
    const float* __restrict__ c1_sum_q, float* __restrict__ out, int64_t m,
    int32_t nq, int32_t w, int32_t g, int32_t n_clusters) {
  // f16 LDS LUTs: half the footprint doubles the usable query-block at
  // equal occupancy. Precision is a non-issue here — the estimate's own
  // 1-bit quantization error dwarfs f16's ~5e-4 relative, and the top-C
  // candidates are exactness-rescored downstream anyway.
  extern __shared__ _Float16 slut_h[];  // FS_QB * g * 16 halves
  int qlo = (int)blockIdx.y * FS_QB;
  int qn = nq - qlo < FS_QB ? nq - qlo : FS_QB;
  int total = FS_QB * g * 16;
  for (int i = (int)threadIdx.x; i < total; i += (int)blockDim.x) {
    int qi = i / (g * 16);
    // zero-fill LUT slots beyond the live queries so the unrolled
    // accumulate below needs no per-iteration bound checks
    slut_h[i] = qi < qn
        ? (_Float16)lut[(int64_t)(qlo + qi) * g * 16 + (i % (g * 16))]
        : (_Float16)0.f;
  }
  __syncthreads();
  int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= m) return;
  const uint8_t* bp = bits + row * (int64_t)w;
  float acc[FS_QB];
#pragma unroll
  for (int qi = 0; qi < FS_QB; qi++) acc[qi] = 0.f;
  int b = 0;
  for (; b + 4 <= w; b += 4) {
    uint32_t v4;
    __builtin_memcpy(&v4, bp + b, 4);
#pragma unroll
    for (int j = 0; j < 4; j++) {
      uint32_t byte = (v4 >> (8 * j)) & 0xFF;
      int g1 = 2 * (b + j), g2 = g1 + 1;
      int lo = g1 * 16 + (int)(byte & 0xF);
      int hi = g2 * 16 + (int)(byte >> 4);
#pragma unroll
      for (int qi = 0; qi < FS_QB; qi++) {
        acc[qi] += (float)slut_h[qi * g * 16 + lo];
        if (g2 < g) acc[qi] += (float)slut_h[qi * g * 16 + hi];
      }
    }
  }
  for (; b < w; b++) {
    uint32_t byte = bp[b];
    int g1 = 2 * b, g2 = g1 + 1;
    int lo = g1 * 16 + (int)(byte & 0xF);
    int hi = g2 * 16 + (int)(byte >> 4);
#pragma unroll
    for (int qi = 0; qi < FS_QB; qi++) {
      acc[qi] += (float)slut_h[qi * g * 16 + lo];
      if (g2 < g) acc[qi] += (float)slut_h[qi * g * 16 + hi];
    }
  }
  float fa = f_add[row];
  float fr = f_rescale[row];
  int cl = cl_of_row[row];
  for (int qi = 0; qi < qn; qi++) {
    int q = qlo + qi;
    float ga = g_add[(int64_t)q * n_clusters + cl];
    out[(int64_t)q * m + row] = fa + ga + fr * (acc[qi] + c1_sum_q[q]);
  }
}

void launch_fastscan_est(const uint8_t* bits, const float* lut,
                         const float* f_add, const float* f_rescale,
                         const int32_t* cl_of_row, const float* g_add,
                         const float* c1_sum_q, float* out, int64_t m,
                         int32_t nq, int32_t w, int32_t g,
                         int32_t n_clusters, hipStream_t s) {
  // query-block factor: trade HBM traffic (bits re-reads) against LDS
  // occupancy (QB LUTs resident). Sweep via LAKESOUL_FS_QB; 0 = off.
  static const int kQB = []() {
    const char* e = getenv("LAKESOUL_FS_QB");
    // measured on MI355X (benchmarks/fs_qb_ab.py, 5Mx768, 64 queries):
    // QB=0 (per-query) 25.7 ms/batch, 2: 15.3, 4: 16.9, 8: 43.6 — the
    // sweet spot trades a 2x bits-traffic cut against LDS occupancy
    // (24.6 KB/block keeps ~6 blocks/CU; 98 KB at QB=8 collapses to 1)
    int v = e ? atoi(e) : 4;
    return v == 1 || v == 2 || v == 4 || v == 8 ? v : (v <= 0 ? 0 : 4);
  }();
  size_t qb_lds = (size_t)kQB * g * 16 * sizeof(_Float16);
  if (kQB > 0 && nq >= kQB && qb_lds <= 120 * 1024) {
    dim3 grid((uint32_t)((m + 255) / 256),
              (uint32_t)((nq + kQB - 1) / kQB));
    auto* fn = kQB == 8 ? fastscan_est_qb_kernel<8>
               : kQB == 4 ? fastscan_est_qb_kernel<4>
               : kQB == 2 ? fastscan_est_qb_kernel<2>
                          : fastscan_est_qb_kernel<1>;
    hipLaunchKernelGGL(fn, grid, dim3(256), qb_lds, s,
                       bits, lut, f_add, f_rescale, cl_of_row, g_add,
                       c1_sum_q, out, m, nq, w, g, n_clusters);
    return;
  }
  dim3 grid((uint32_t)((m + 255) / 256), (uint32_t)nq);
  size_t lds = (size_t)g * 16 * sizeof(float);
  hipLaunchKernelGGL(fastscan_est_kernel, grid, dim3(256), lds, s, bits, lut,
                     f_add, f_rescale, cl_of_row, g_add, c1_sum_q, out, m, nq,
                     w, g, n_clusters);
}

// Ex-code refinement dot: ex nibbles [m][wn] uint8 (wn = ceil(dim/2),
// low nibble = even dim), q: [nq][dim] f32 -> out [m][nq] f32 of
// <ex_code, q>. Candidate sets are small (top-C per query), so a simple
// one-thread-per-(row) loop with q staged in LDS suffices.
__global__ __launch_bounds__(256) void fastscan_ex_dot_kernel(
    const uint8_t* __restrict__ ex, const float* __restrict__ qv,
    float* __restrict__ out, int64_t m, int32_t nq, int32_t wn,
    int32_t dim) {
  extern __shared__ float sq[];  // dim floats for this block's query
  int q = (int)blockIdx.y;
  const float* qp = qv + (int64_t)q * dim;
  for (int i = (int)threadIdx.x; i < dim; i += (int)blockDim.x) sq[i] = qp[i];
  __syncthreads();
  int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= m) return;
  const uint8_t* ep = ex + row * (int64_t)wn;
  float acc = 0.f;
  for (int b = 0; b < wn; b++) {
    uint32_t v = ep[b];
    int d0 = 2 * b;
    acc += (float)(v & 0xF) * sq[d0];
    if (d0 + 1 < dim) acc += (float)(v >> 4) * sq[d0 + 1];
  }
  out[row * (int64_t)nq + q] = acc;
}

void launch_fastscan_ex_dot(const uint8_t* ex, const float* qv,
                                       float* out, int64_t m, int32_t nq,
                                       int32_t wn, int32_t dim,
                                       hipStream_t s) {
  dim3 grid((uint32_t)((m + 255) / 256), (uint32_t)nq);
  size_t lds = (size_t)dim * sizeof(float);
  hipLaunchKernelGGL(fastscan_ex_dot_kernel, grid, dim3(256), lds, s, ex, qv,
                     out, m, nq, wn, dim);
}

// Pair-wise ex dots: one dot per (query, candidate) pair instead of
// (unique-candidate x every-query). At C candidates/query the unique-row
// variant above does ~nq/overlap times the needed work (measured 93 ms
// of 123 ms in the 5Mx768 hi-C search); this form reads each packed ex
// row once per selecting query only. Block = one query's chunk of
// candidates; the query vector stages through LDS.
__global__ __launch_bounds__(256) void fastscan_ex_dot_pairs_kernel(
    const uint8_t* __restrict__ ex, const int64_t* __restrict__ cand,
    const float* __restrict__ qv, float* __restrict__ out, int32_t C,
    int32_t nq, int32_t wn, int32_t dim) {
  extern __shared__ float sq[];  // dim floats for this block's query
  int q = (int)blockIdx.y;
  const float* qp = qv + (int64_t)q * dim;
  for (int i = (int)threadIdx.x; i < dim; i += (int)blockDim.x) sq[i] = qp[i];
  __syncthreads();
  int p = (int)(blockIdx.x * blockDim.x + threadIdx.x);
  if (p >= C) return;
  int64_t row = cand[(int64_t)q * C + p];
  if (row < 0) {
    out[(int64_t)q * C + p] = 0.f;
    return;
  }
  const uint8_t* ep = ex + row * (int64_t)wn;
  float acc = 0.f;
  // 4-byte chunks: each uint32 carries 8 nibbles (8 FULL dims — the
  // byte tail below covers dim % 8, so no sq[] overread on odd dims)
  int wn4 = dim / 8;
  const uint32_t* ep4 = (const uint32_t*)ep;
  for (int b = 0; b < wn4; b++) {
    uint32_t v = ep4[b];
    int d0 = 8 * b;
#pragma unroll
    for (int j = 0; j < 8; j++)
      acc += (float)((v >> (4 * j)) & 0xF) * sq[d0 + j];
  }
  for (int b = 4 * wn4; b < wn; b++) {
    uint32_t v = ep[b];
    int d0 = 2 * b;
    acc += (float)(v & 0xF) * sq[d0];
    if (d0 + 1 < dim) acc += (float)(v >> 4) * sq[d0 + 1];
  }
  out[(int64_t)q * C + p] = acc;
}

void launch_fastscan_ex_dot_pairs(const uint8_t* ex, const int64_t* cand,
                                  const float* qv, float* out, int32_t C,
                                  int32_t nq, int32_t wn, int32_t dim,
                                  hipStream_t s) {
  dim3 grid((uint32_t)((C + 255) / 256), (uint32_t)nq);
  size_t lds = (size_t)dim * sizeof(float);
  hipLaunchKernelGGL(fastscan_ex_dot_pairs_kernel, grid, dim3(256), lds, s,
                     ex, cand, qv, out, C, nq, wn, dim);
}

}  // namespace lakesoul
