// ANN scoring on MFMA matrix cores (gfx950): S[n][q] = dot(X[n], Q[q])
// over bf16 vectors, fp32 accumulate — the MI355X-native replacement for
// the reference's AVX2/AVX-512 FastScan + dot kernels
// (rust/lakesoul-vector/src/simd.rs).
//
// Structure (cdna_hip_programming.md §5 anatomy, correctness-first):
// one workgroup = 4 waves, each wave owns a 16-row (vector) strip;
// mfma_f32_16x16x32_bf16 over the K dimension; query tile of 16 columns
// looped inside the kernel. X rows stream from HBM (each used once per
// query tile — memory-bound); Q is small and stays L2/L1-resident.
// C/D fragment mapping col=lane&15, row=(lane>>4)*4+reg (guide §3,
// verified on HW by the refcheck test against torch.matmul fp32).

#include <hip/hip_runtime.h>

#include <cstdint>

namespace lakesoul {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// X: [n][K] bf16 row-major; Q: [nq][K] bf16 row-major.
// out: [n][nq] f32, or [nq][n] when TRANSPOSED (query-major output lets
// the downstream per-query top-k read contiguously — torch.topk over
// the strided dim of (n, nq) measured 28 ms vs 2.5 ms contiguous on
// 5M x 64; benchmarks/topk_micro.py).
// K must be a multiple of 32; nq a multiple of 16 (caller pads).
template <bool TRANSPOSED>
__global__ __launch_bounds__(256) void ann_scores_kernel(
    const short* __restrict__ X, const short* __restrict__ Q,
    float* __restrict__ out, int64_t n, int32_t nq, int32_t K) {
  int wave = (int)(threadIdx.x >> 6);
  int lane = (int)(threadIdx.x & 63);
  int64_t row0 = ((int64_t)blockIdx.x * 4 + wave) * 16;
  if (row0 >= n) return;

  int r = lane & 15;        // row within the 16-strip (A), col for B/D
  int khalf = lane >> 4;    // 0..3 -> k sub-offset *8

  int64_t xrow = row0 + r;
  bool row_ok = xrow < n;
  const short* xp = X + (row_ok ? xrow * K : 0);

  // process up to 4 query tiles (64 queries) per X pass: X bytes are the
  // memory-bound term, so amortizing them 4x quadruples arithmetic
  // intensity (each X fragment feeds 4 MFMAs)
  for (int q0 = 0; q0 < nq; q0 += 64) {
    int ntiles = (nq - q0) / 16;
    if (ntiles > 4) ntiles = 4;
    f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f},
                    {0.f, 0.f, 0.f, 0.f},
                    {0.f, 0.f, 0.f, 0.f},
                    {0.f, 0.f, 0.f, 0.f}};
    for (int k0 = 0; k0 < K; k0 += 32) {
      bf16x8 a = row_ok ? *(const bf16x8*)(xp + k0 + khalf * 8)
                        : (bf16x8)(short)0;
#pragma unroll
      for (int t = 0; t < 4; t++) {
        if (t >= ntiles) break;
        const short* qp = Q + (int64_t)(q0 + t * 16 + r) * K;
        bf16x8 b = *(const bf16x8*)(qp + k0 + khalf * 8);
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[t], 0, 0, 0);
      }
    }
    // D mapping: col=lane&15 (query), row=(lane>>4)*4+reg (vector)
#pragma unroll
    for (int t = 0; t < 4; t++) {
      if (t >= ntiles) break;
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        int64_t orow = row0 + khalf * 4 + reg;
        if (orow < n) {
          if (TRANSPOSED)
            out[(int64_t)(q0 + t * 16 + r) * n + orow] = acc[t][reg];
          else
            out[orow * nq + q0 + t * 16 + r] = acc[t][reg];
        }
      }
    }
  }
}

void launch_ann_scores(const short* X, const short* Q, float* out, int64_t n,
                       int32_t nq, int32_t K, hipStream_t s) {
  int64_t blocks = (n + 63) / 64;
  hipLaunchKernelGGL(ann_scores_kernel<false>, dim3((uint32_t)blocks),
                     dim3(256), 0, s, X, Q, out, n, nq, K);
}

void launch_ann_scores_t(const short* X, const short* Q, float* out,
                         int64_t n, int32_t nq, int32_t K, hipStream_t s) {
  int64_t blocks = (n + 63) / 64;
  hipLaunchKernelGGL(ann_scores_kernel<true>, dim3((uint32_t)blocks),
                     dim3(256), 0, s, X, Q, out, n, nq, K);
}

}  // namespace lakesoul
