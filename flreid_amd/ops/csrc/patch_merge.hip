// K4: fused Swin PatchMerging gather + LayerNorm.
//
// ref:models/swin_transformer.py:414-435: the eager chain materialises the
// 2×2 strided concat [B, L/4, 4C], then runs LayerNorm (stats + normalize)
// as separate kernels.  Here one pass gathers the four source tokens,
// computes the row statistics and writes the NORMALIZED bf16/f32 rows the
// reduction GEMM consumes — the concat tensor never exists.
//
// Concat order matches the model: [x(0::2,0::2), x(1::2,0::2),
// x(0::2,1::2), x(1::2,1::2)] along the channel dim.
//
// Geometry: one 64-lane wave per output row (4 rows per 256-thread block);
// lanes stride the 4C channels.  fp32 compute; mean/rstd saved for the
// backward, which recomputes x̂ from the gathered input and emits the
// scattered dx plus per-block dgamma/dbeta partials (summed by the
// wrapper).

#include "common.h"

namespace flreid {

__device__ __forceinline__ int64_t pm_src_index(int64_t b, int r, int c4,
                                                int C, int H, int W,
                                                int out_w, int ho, int wo) {
  // channel block q = c4 / C selects which of the 4 source tokens
  const int q = c4 / C;
  const int c = c4 - q * C;
  const int dh = (q == 1 || q == 3) ? 1 : 0;   // rows 1 for blocks 1,3
  const int dw = (q >= 2) ? 1 : 0;             // cols 1 for blocks 2,3
  const int h = ho * 2 + dh, w = wo * 2 + dw;
  return ((b * H + h) * (int64_t)W + w) * C + c;
}

template <typename T>
__global__ __launch_bounds__(256) void patch_merge_ln_fwd_kernel(
    const T* __restrict__ X, const float* __restrict__ GAMMA,
    const float* __restrict__ BETA, T* __restrict__ Y,
    float* __restrict__ MEAN, float* __restrict__ RSTD, int64_t rows,
    int C, int H, int W, float eps) {
  const int64_t row = (int64_t)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const int C4 = 4 * C;
  const int out_w = W >> 1;
  const int64_t ncols = (int64_t)(H >> 1) * out_w;
  const int64_t b = row / ncols;
  const int64_t rem = row - b * ncols;
  const int ho = (int)(rem / out_w), wo = (int)(rem % out_w);

  float sum = 0.f, sq = 0.f;
  for (int c4 = lane; c4 < C4; c4 += 64) {
    const float v = load_as_float(
        X, pm_src_index(b, 0, c4, C, H, W, out_w, ho, wo));
    sum += v;
    sq += v * v;
  }
  sum = wave_reduce_sum(sum);
  sq = wave_reduce_sum(sq);
  sum = __shfl(sum, 0, 64);
  sq = __shfl(sq, 0, 64);
  const float mean = sum / C4;
  const float var = sq / C4 - mean * mean;
  const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
  if (lane == 0) {
    MEAN[row] = mean;
    RSTD[row] = rstd;
  }
  T* yr = Y + row * C4;
  for (int c4 = lane; c4 < C4; c4 += 64) {
    const float v = load_as_float(
        X, pm_src_index(b, 0, c4, C, H, W, out_w, ho, wo));
    store_from_float(yr, c4, (v - mean) * rstd * GAMMA[c4] + BETA[c4]);
  }
}

template <typename T>
__global__ __launch_bounds__(256) void patch_merge_ln_bwd_kernel(
    const T* __restrict__ X, const float* __restrict__ GAMMA,
    const T* __restrict__ DY, const float* __restrict__ MEAN,
    const float* __restrict__ RSTD, T* __restrict__ DX,
    float* __restrict__ DG_PART, float* __restrict__ DB_PART, int64_t rows,
    int C, int H, int W) {
  __shared__ float sdg[3072];  // 4C ≤ 3072: per-block dgamma/dbeta partials
  __shared__ float sdb[3072];
  const int C4 = 4 * C;
  for (int i = threadIdx.x; i < C4; i += 256) {
    sdg[i] = 0.f;
    sdb[i] = 0.f;
  }
  __syncthreads();

  const int64_t row = (int64_t)blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (row < rows) {
    const int out_w = W >> 1;
    const int64_t ncols = (int64_t)(H >> 1) * out_w;
    const int64_t b = row / ncols;
    const int64_t rem = row - b * ncols;
    const int ho = (int)(rem / out_w), wo = (int)(rem % out_w);
    const float mean = MEAN[row], rstd = RSTD[row];
    const T* dyr = DY + row * C4;

    // two row reductions: m1 = mean(γ·dy), m2 = mean(γ·dy·x̂)
    float m1 = 0.f, m2 = 0.f;
    for (int c4 = lane; c4 < C4; c4 += 64) {
      const float xv = load_as_float(
          X, pm_src_index(b, 0, c4, C, H, W, out_w, ho, wo));
      const float xh = (xv - mean) * rstd;
      const float gdy = GAMMA[c4] * load_as_float(dyr, c4);
      m1 += gdy;
      m2 += gdy * xh;
    }
    m1 = wave_reduce_sum(m1);
    m2 = wave_reduce_sum(m2);
    m1 = __shfl(m1, 0, 64) / C4;
    m2 = __shfl(m2, 0, 64) / C4;

    for (int c4 = lane; c4 < C4; c4 += 64) {
      const int64_t src = pm_src_index(b, 0, c4, C, H, W, out_w, ho, wo);
      const float xv = load_as_float(X, src);
      const float xh = (xv - mean) * rstd;
      const float dyv = load_as_float(dyr, c4);
      const float gdy = GAMMA[c4] * dyv;
      store_from_float(DX, src, (gdy - m1 - xh * m2) * rstd);
      atomicAdd(&sdg[c4], dyv * xh);
      atomicAdd(&sdb[c4], dyv);
    }
  }
  __syncthreads();
  float* dg = DG_PART + (int64_t)blockIdx.x * C4;
  float* db = DB_PART + (int64_t)blockIdx.x * C4;
  for (int i = threadIdx.x; i < C4; i += 256) {
    dg[i] = sdg[i];
    db[i] = sdb[i];
  }
}

extern "C" void flreid_patch_merge_ln_fwd(const void* X, const float* gamma,
                                          const float* beta, void* Y,
                                          float* mean, float* rstd,
                                          int64_t rows, int C, int H, int W,
                                          float eps, int dtype,
                                          hipStream_t stream) {
  if (4 * C > 3072) throw std::runtime_error("patch_merge_ln: 4C > 3072");
  dim3 grid((unsigned)((rows + 3) / 4)), block(256);
  if (dtype == kF32) {
    hipLaunchKernelGGL((patch_merge_ln_fwd_kernel<float>), grid, block, 0,
                       stream, (const float*)X, gamma, beta, (float*)Y, mean,
                       rstd, rows, C, H, W, eps);
  } else {
    hipLaunchKernelGGL((patch_merge_ln_fwd_kernel<__hip_bfloat16>), grid,
                       block, 0, stream, (const __hip_bfloat16*)X, gamma,
                       beta, (__hip_bfloat16*)Y, mean, rstd, rows, C, H, W,
                       eps);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void flreid_patch_merge_ln_bwd(const void* X, const float* gamma,
                                          const void* DY, const float* mean,
                                          const float* rstd, void* DX,
                                          float* dg_part, float* db_part,
                                          int64_t rows, int C, int H, int W,
                                          int dtype, hipStream_t stream) {
  dim3 grid((unsigned)((rows + 3) / 4)), block(256);
  if (dtype == kF32) {
    hipLaunchKernelGGL((patch_merge_ln_bwd_kernel<float>), grid, block, 0,
                       stream, (const float*)X, gamma, (const float*)DY,
                       mean, rstd, (float*)DX, dg_part, db_part, rows, C, H,
                       W);
  } else {
    hipLaunchKernelGGL((patch_merge_ln_bwd_kernel<__hip_bfloat16>), grid,
                       block, 0, stream, (const __hip_bfloat16*)X, gamma,
                       (const __hip_bfloat16*)DY, mean, rstd,
                       (__hip_bfloat16*)DX, dg_part, db_part, rows, C, H, W);
  }
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
