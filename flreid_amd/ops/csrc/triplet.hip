// Fused batch-hard triplet loss (K5 in SURVEY.md §2.9).
//
// Replaces the eager chain pairwise-distance GEMM -> masked max/min ->
// margin-ranking loss (ref:criterions/triplet_loss.py:35-61,116-125 +
// ref:tools/distance.py:9-16) with one kernel per direction:
//   fwd: per anchor i, the squared-euclidean row d_ij = na_i + na_j − 2·f_i·f_j
//        lives in registers; hardest positive (max over same-label, FIRST
//        index on ties, matching torch.max) and hardest negative
//        (min over other-label + 1e9·pos) are reduced in-wave; per-anchor
//        hinge loss max(0, margin + ap − an).
//   bwd: analytic — active anchors route ±(2/N)·go through (i, p_i, n_i)
//        feature rows via atomics.
//
// fp32 features (the training path normalises/loss in fp32).

#include "common.h"

namespace flreid {

// one block per anchor row; 4 waves share the j loop
template <int BLOCK>
__global__ __launch_bounds__(256) void triplet_fwd_kernel(
    const float* __restrict__ F, const float* __restrict__ norms,
    const int64_t* __restrict__ labels, float* __restrict__ row_loss,
    int* __restrict__ p_idx, int* __restrict__ n_idx, int N, int D,
    float margin) {
  const int i = blockIdx.x;
  if (i >= N) return;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  constexpr int NW = BLOCK / 64;
  __shared__ float s_ap[NW], s_an[NW];
  __shared__ int s_pi[NW], s_ni[NW];

  const float* fi = F + (int64_t)i * D;
  const int64_t li = labels[i];
  const float ni_sq = norms[i];

  float best_ap = -INFINITY, best_an = INFINITY;
  int best_p = N, best_n = N;     // first-index tie-break via index compare

  for (int j = wave; j < N; j += NW) {
    const float* fj = F + (int64_t)j * D;
    float dot = 0.f;
    for (int d = lane; d < D; d += 64) {
      dot = fmaf(fi[d], fj[d], dot);
    }
    dot = wave_reduce_sum(dot);
    dot = __shfl(dot, 0, 64);
    const float dij = ni_sq + norms[j] - 2.0f * dot;
    const bool pos = labels[j] == li;
    // reference semantics: ap = max(d·is_pos), an = min(d·is_neg + 1e9·is_pos)
    const float apv = pos ? dij : 0.0f;
    const float anv = pos ? 1e9f : dij;
    if (apv > best_ap || (apv == best_ap && j < best_p)) {
      best_ap = apv;
      best_p = j;
    }
    if (anv < best_an || (anv == best_an && j < best_n)) {
      best_an = anv;
      best_n = j;
    }
  }

  if (lane == 0) {
    s_ap[wave] = best_ap;
    s_an[wave] = best_an;
    s_pi[wave] = best_p;
    s_ni[wave] = best_n;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float ap = -INFINITY, an = INFINITY;
    int pi = N, nidx = N;
#pragma unroll
    for (int w = 0; w < NW; ++w) {
      if (s_ap[w] > ap || (s_ap[w] == ap && s_pi[w] < pi)) {
        ap = s_ap[w];
        pi = s_pi[w];
      }
      if (s_an[w] < an || (s_an[w] == an && s_ni[w] < nidx)) {
        an = s_an[w];
        nidx = s_ni[w];
      }
    }
    row_loss[i] = fmaxf(0.0f, margin + ap - an);
    p_idx[i] = pi;
    n_idx[i] = nidx;
  }
}

// grad routing: block per ACTIVE anchor; 2/N·go through (i, p, n) rows
template <int BLOCK>
__global__ __launch_bounds__(256) void triplet_bwd_kernel(
    const float* __restrict__ F, const float* __restrict__ row_loss,
    const int* __restrict__ p_idx, const int* __restrict__ n_idx,
    float* __restrict__ grad, int N, int D, float coeff) {
  const int i = blockIdx.x;
  if (i >= N || row_loss[i] <= 0.0f) return;
  const int p = p_idx[i];
  const int n = n_idx[i];
  const float* fi = F + (int64_t)i * D;
  const float* fp = F + (int64_t)p * D;
  const float* fn = F + (int64_t)n * D;
  // d(ap)/dfi = 2(fi-fp), d(an)/dfi = 2(fi-fn); loss_i = m + ap - an
  for (int d = threadIdx.x; d < D; d += BLOCK) {
    const float vi = fi[d], vp = fp[d], vn = fn[d];
    atomicAdd(&grad[(int64_t)i * D + d], coeff * ((vi - vp) - (vi - vn)));
    atomicAdd(&grad[(int64_t)p * D + d], coeff * (vp - vi));
    atomicAdd(&grad[(int64_t)n * D + d], -coeff * (vn - vi));
  }
}

extern "C" void flreid_triplet_fwd(const float* F, const float* norms,
                                   const int64_t* labels, float* row_loss,
                                   int* p_idx, int* n_idx, int N, int D,
                                   float margin, hipStream_t stream) {
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL((triplet_fwd_kernel<BLOCK>), dim3(N), dim3(BLOCK), 0,
                     stream, F, norms, labels, row_loss, p_idx, n_idx, N, D,
                     margin);
  HIP_CHECK(hipGetLastError());
}

extern "C" void flreid_triplet_bwd(const float* F, const float* row_loss,
                                   const int* p_idx, const int* n_idx,
                                   float* grad, int N, int D, float coeff,
                                   hipStream_t stream) {
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL((triplet_bwd_kernel<BLOCK>), dim3(N), dim3(BLOCK), 0,
                     stream, F, row_loss, p_idx, n_idx, grad, N, D, coeff);
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
