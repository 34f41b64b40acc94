// Pairwise distance / similarity GEMM on MFMA (K8 in SURVEY.md §2.9).
//
// Computes OUT[m, n] over row-major feature matrices A[M, D], B[N, D]:
//   mode 0:  A·Bᵀ                      (similarity, the CMC/mAP eval GEMM —
//                                       replaces the per-query GEMV loop of
//                                       ref:tools/evaluate.py:103-142)
//   mode 1:  1 − A·Bᵀ                  (cosine distance on pre-normalised
//                                       rows, ref:tools/distance.py:19-30)
//   mode 2:  aa[m] + bb[n] − 2·A·Bᵀ    (squared euclidean,
//                                       ref:tools/distance.py:9-16)
//
// Exact fp32 numerics via the f32-input MFMA (v_mfma_f32_32x32x2_f32): the
// result is bit-for-bit an fmaf chain (guide §3), so eval metrics match the
// fp32 reference exactly up to summation order.
//
// Structure (guide §5 canonical): 64×64 block tile, 4 waves × one 32×32
// sub-tile, LDS-staged K panels (BK=32, padded rows: conflict-free b32 reads).

#include "common.h"

namespace flreid {

using f32x16 = __attribute__((ext_vector_type(16))) float;

constexpr int BM = 128;     // block tile rows
constexpr int BN = 128;     // block tile cols
constexpr int BK = 32;      // K panel
constexpr int PAD = 1;      // LDS row padding (33 floats/row)

// 128×128 tile, 4 waves, each wave owns a 64×64 quadrant = 2×2 MFMA 32×32
// sub-tiles (4 independent accumulators pipeline the 64-cycle f32 MFMA).
// Arithmetic intensity 32 flops/byte — the 64×64 version measured HBM-bound
// at 117 TF; this tile halves the HBM traffic.
template <int MODE>
__global__ __launch_bounds__(256) void pairwise_mfma_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    const float* __restrict__ aa, const float* __restrict__ bb,
    float* __restrict__ OUT, int64_t M, int64_t N, int64_t D) {
  __shared__ float ldsA[BM][BK + PAD];
  __shared__ float ldsB[BN][BK + PAD];

  // note: an XCD-aware block swizzle (guide T1) measured neutral here —
  // x-major dispatch already gives consecutive blocks a shared B panel
  const int64_t row0 = (int64_t)blockIdx.x * BM;
  const int64_t col0 = (int64_t)blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;            // 4 waves
  const int wr = (wave >> 1) * 64;      // wave quadrant row offset (0/64)
  const int wc = (wave & 1) * 64;       // wave quadrant col offset (0/64)

  f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};

  // float4 staging: 8 lanes cover one 32-float K-row; interior blocks skip
  // every bounds check and the int64 per-element address math (PMC showed
  // guarded scalar loads made the kernel VALU-bound: 4.5 VALU/MFMA)
  const int lc4 = (tid & 7) * 4;        // load col 0,4,...,28
  const int lr0 = tid >> 3;             // load row 0..31 (32 rows/pass)
  const bool interior = (row0 + BM <= M) && (col0 + BN <= N) && (D % 4 == 0);

  const int fr = lane & 31;             // fragment row/col within 32-tile
  const int fk = lane >> 5;             // fragment k (0/1)

  for (int64_t k0 = 0; k0 < D; k0 += BK) {
    if (interior && k0 + BK <= D) {
      const float* arow = A + (row0 + lr0) * D + k0 + lc4;
      const float* brow = B + (col0 + lr0) * D + k0 + lc4;
#pragma unroll
      for (int r = 0; r < BM; r += 32) {
        const float4 va = *(const float4*)(arow + (int64_t)r * D);
        const float4 vb = *(const float4*)(brow + (int64_t)r * D);
        ldsA[lr0 + r][lc4 + 0] = va.x; ldsA[lr0 + r][lc4 + 1] = va.y;
        ldsA[lr0 + r][lc4 + 2] = va.z; ldsA[lr0 + r][lc4 + 3] = va.w;
        ldsB[lr0 + r][lc4 + 0] = vb.x; ldsB[lr0 + r][lc4 + 1] = vb.y;
        ldsB[lr0 + r][lc4 + 2] = vb.z; ldsB[lr0 + r][lc4 + 3] = vb.w;
      }
    } else {
#pragma unroll
      for (int r = 0; r < BM; r += 32) {
        const int lr = lr0 + r;
        const int64_t ar = row0 + lr;
        const int64_t bc = col0 + lr;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int64_t kk = k0 + lc4 + c;
          ldsA[lr][lc4 + c] = (ar < M && kk < D) ? A[ar * D + kk] : 0.0f;
          ldsB[lr][lc4 + c] = (bc < N && kk < D) ? B[bc * D + kk] : 0.0f;
        }
      }
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < BK; kk += 2) {
      const float a0 = ldsA[wr + fr][kk + fk];
      const float a1 = ldsA[wr + 32 + fr][kk + fk];
      const float b0 = ldsB[wc + fr][kk + fk];
      const float b1 = ldsB[wc + 32 + fr][kk + fk];
      acc00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
  }

  // C/D mapping (guide §3): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
#pragma unroll
  for (int ti = 0; ti < 2; ++ti) {
#pragma unroll
    for (int tj = 0; tj < 2; ++tj) {
      const f32x16 acc = ti == 0 ? (tj == 0 ? acc00 : acc01)
                                 : (tj == 0 ? acc10 : acc11);
      const int64_t jc = col0 + wc + tj * 32 + (lane & 31);
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int64_t ir = row0 + wr + ti * 32 + (reg & 3) + 8 * (reg >> 2)
                           + 4 * (lane >> 5);
        if (ir < M && jc < N) {
          float v = acc[reg];
          if (MODE == 1) v = 1.0f - v;
          if (MODE == 2) v = aa[ir] + bb[jc] - 2.0f * v;
          OUT[ir * N + jc] = v;
        }
      }
    }
  }
}

extern "C" void flreid_pairwise(const float* A, const float* B,
                                const float* aa, const float* bb, float* OUT,
                                int64_t M, int64_t N, int64_t D, int mode,
                                hipStream_t stream) {
  dim3 grid((unsigned)((M + BM - 1) / BM), (unsigned)((N + BN - 1) / BN));
  dim3 block(256);
  switch (mode) {
    case 0:
      hipLaunchKernelGGL((pairwise_mfma_kernel<0>), grid, block, 0, stream, A,
                         B, aa, bb, OUT, M, N, D);
      break;
    case 1:
      hipLaunchKernelGGL((pairwise_mfma_kernel<1>), grid, block, 0, stream, A,
                         B, aa, bb, OUT, M, N, D);
      break;
    case 2:
      hipLaunchKernelGGL((pairwise_mfma_kernel<2>), grid, block, 0, stream, A,
                         B, aa, bb, OUT, M, N, D);
      break;
    default:
      throw std::runtime_error("bad pairwise mode");
  }
  HIP_CHECK(hipGetLastError());
}

// row sums of squares (for the euclidean epilogue)
template <int BLOCK>
__global__ void rowsq_kernel(const float* __restrict__ x,
                             float* __restrict__ out, int64_t rows,
                             int64_t cols) {
  __shared__ float scratch[BLOCK / kWave];
  const int64_t row = blockIdx.x;
  if (row >= rows) return;
  const float* xr = x + row * cols;
  float ss = 0.f;
  for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
    const float v = xr[c];
    ss += v * v;
  }
  ss = block_reduce_sum<BLOCK>(ss, scratch);
  if (threadIdx.x == 0) out[row] = ss;
}

extern "C" void flreid_rowsq(const float* x, float* out, int64_t rows,
                             int64_t cols, hipStream_t stream) {
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL((rowsq_kernel<BLOCK>), dim3((unsigned)rows), dim3(BLOCK),
                     0, stream, x, out, rows, cols);
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
