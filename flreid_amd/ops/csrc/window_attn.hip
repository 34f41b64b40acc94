// Fused Swin window attention forward (K3 in SURVEY.md §2.9).
//
// One workgroup per (window, head): Q/K/V tiles staged in LDS, the 49×49
// score matrix never touches HBM (the eager path materialises it plus the
// bias/mask adds and softmax as five separate [B·nW, H, N, N] kernels —
// ref:models/swin_transformer.py:255-286).
//
//   S = scale·Q·Kᵀ + rel_pos_bias[h] (+ shift_mask[w])   (LDS-resident)
//   P = softmax_rows(S)
//   O = P·V
//
// Shapes: Q/K/V/O [BW, H, N, D] contiguous (N = ws² ≤ 64, D ≤ 64),
// bias [H, N, N], mask [nW, N, N] or null (window index = blockIdx.x % nW
// after splitting off the head).  fp32 compute; bf16 or fp32 I/O.
// Grid = BW·H workgroups ≫ 256 CUs at Swin shapes (stage 1 tiny: 12k).

#include "common.h"

namespace flreid {

constexpr int MAX_N = 64;   // tokens per window (49 for ws=7)
constexpr int MAX_D = 64;   // head dim (32 for Swin)

template <typename T, int BLOCK>
__global__ __launch_bounds__(256) void window_attn_fwd_kernel(
    const T* __restrict__ Q, const T* __restrict__ K, const T* __restrict__ V,
    const float* __restrict__ bias, const float* __restrict__ mask,
    T* __restrict__ O, int64_t BW, int H, int N, int D, int nW, float scale) {
  __shared__ float lq[MAX_N][MAX_D + 1];
  // K stored TRANSPOSED [d][j]: the QK^T inner loop then reads 64 lanes at
  // consecutive j (stride 1, conflict-free) instead of 49 distinct rows
  // (measured 1.9e8 SQ_LDS_BANK_CONFLICT on the row-major layout)
  __shared__ float lkT[MAX_D][MAX_N + 1];
  __shared__ float lv[MAX_N][MAX_D + 1];
  __shared__ float ls[MAX_N][MAX_N + 1];

  const int64_t wh = blockIdx.x;          // (window, head) flat index
  if (wh >= BW * H) return;
  const int64_t bw = wh / H;
  const int h = (int)(wh % H);
  const int w_idx = (int)(bw % nW);

  const int64_t base = (bw * H + h) * (int64_t)N * D;
  const int tid = threadIdx.x;

  // stage Q/K/V tiles
  for (int i = tid; i < N * D; i += BLOCK) {
    const int r = i / D, c = i % D;
    lq[r][c] = load_as_float(Q, base + i);
    lkT[c][r] = load_as_float(K, base + i);
    lv[r][c] = load_as_float(V, base + i);
  }
  __syncthreads();

  // S = scale·Q·Kᵀ + bias (+ mask)
  const float* brow = bias + (int64_t)h * N * N;
  const float* mrow = mask ? mask + (int64_t)w_idx * N * N : nullptr;
  for (int e = tid; e < N * N; e += BLOCK) {
    const int i = e / N, j = e % N;
    float acc = 0.f;
#pragma unroll 8
    for (int d = 0; d < D; ++d) {
      acc = fmaf(lq[i][d], lkT[d][j], acc);
    }
    acc = acc * scale + brow[e];
    if (mrow) acc += mrow[e];
    ls[i][j] = acc;
  }
  __syncthreads();

  // row softmax: 4 lanes per row, shfl_xor reduction within the quad
  const int r = tid >> 2;          // row
  const int sub = tid & 3;         // lane within quad
  if (r < N) {
    float m = -INFINITY;
    for (int j = sub; j < N; j += 4) m = fmaxf(m, ls[r][j]);
    m = fmaxf(m, __shfl_xor(m, 1, 64));
    m = fmaxf(m, __shfl_xor(m, 2, 64));
    float se = 0.f;
    for (int j = sub; j < N; j += 4) {
      const float p = __expf(ls[r][j] - m);
      ls[r][j] = p;
      se += p;
    }
    se += __shfl_xor(se, 1, 64);
    se += __shfl_xor(se, 2, 64);
    const float inv = 1.0f / se;
    for (int j = sub; j < N; j += 4) ls[r][j] *= inv;
  }
  __syncthreads();

  // O = P·V
  for (int e = tid; e < N * D; e += BLOCK) {
    const int i = e / D, d = e % D;
    float acc = 0.f;
    for (int j = 0; j < N; ++j) {
      acc = fmaf(ls[i][j], lv[j][d], acc);
    }
    store_from_float(O, base + e, acc);
  }
}

// ---------------------------------------------------------------------------
// Window-attention BACKWARD (K3 training path).
//
// One workgroup per (window, head), everything LDS-resident; S→P is
// RECOMPUTED from Q/K (cheaper than saving the [BW,H,N,N] probability
// tensor through autograd):
//   P  = softmax(scale·Q·Kᵀ + bias (+ mask))
//   dV = Pᵀ·dO
//   dP = dO·Vᵀ
//   dS = P ∘ (dP − rowsum(dP ∘ P))          (softmax backward)
//   dQ = scale·dS·K,  dK = scale·dSᵀ·Q
// dS is also written to HBM; the wrapper reduces it over windows for the
// relative-position-bias gradient (the bias gather backward stays in
// autograd).  ref:models/swin_transformer.py:255-286.
// ---------------------------------------------------------------------------

template <typename T, int BLOCK>
__global__ __launch_bounds__(256) void window_attn_bwd_kernel(
    const T* __restrict__ Q, const T* __restrict__ K, const T* __restrict__ V,
    const float* __restrict__ bias, const float* __restrict__ mask,
    const T* __restrict__ DO, T* __restrict__ DQ, T* __restrict__ DK,
    T* __restrict__ DV, float* __restrict__ DS, int64_t BW, int H, int N,
    int D, int nW, float scale) {
  __shared__ float lq[MAX_N][MAX_D + 1];
  __shared__ float lkT[MAX_D][MAX_N + 1];
  __shared__ float lv[MAX_N][MAX_D + 1];
  __shared__ float ldo[MAX_N][MAX_D + 1];
  __shared__ float ls[MAX_N][MAX_N + 1];     // S -> P
  __shared__ float ldp[MAX_N][MAX_N + 1];    // dP -> dS

  const int64_t wh = blockIdx.x;
  if (wh >= BW * H) return;
  const int64_t bw = wh / H;
  const int h = (int)(wh % H);
  const int w_idx = (int)(bw % nW);

  const int64_t base = (bw * H + h) * (int64_t)N * D;
  const int tid = threadIdx.x;

  for (int i = tid; i < N * D; i += BLOCK) {
    const int r = i / D, c = i % D;
    lq[r][c] = load_as_float(Q, base + i);
    lkT[c][r] = load_as_float(K, base + i);
    lv[r][c] = load_as_float(V, base + i);
    ldo[r][c] = load_as_float(DO, base + i);
  }
  __syncthreads();

  // recompute S
  const float* brow = bias + (int64_t)h * N * N;
  const float* mrow = mask ? mask + (int64_t)w_idx * N * N : nullptr;
  for (int e = tid; e < N * N; e += BLOCK) {
    const int i = e / N, j = e % N;
    float acc = 0.f;
#pragma unroll 8
    for (int d = 0; d < D; ++d) {
      acc = fmaf(lq[i][d], lkT[d][j], acc);
    }
    acc = acc * scale + brow[e];
    if (mrow) acc += mrow[e];
    ls[i][j] = acc;
  }
  __syncthreads();

  // softmax rows (as fwd)
  {
    const int r = tid >> 2, sub = tid & 3;
    if (r < N) {
      float m = -INFINITY;
      for (int j = sub; j < N; j += 4) m = fmaxf(m, ls[r][j]);
      m = fmaxf(m, __shfl_xor(m, 1, 64));
      m = fmaxf(m, __shfl_xor(m, 2, 64));
      float se = 0.f;
      for (int j = sub; j < N; j += 4) {
        const float p = __expf(ls[r][j] - m);
        ls[r][j] = p;
        se += p;
      }
      se += __shfl_xor(se, 1, 64);
      se += __shfl_xor(se, 2, 64);
      const float inv = 1.0f / se;
      for (int j = sub; j < N; j += 4) ls[r][j] *= inv;
    }
  }
  __syncthreads();

  // dV = Pᵀ·dO  (ls still holds P)
  for (int e = tid; e < N * D; e += BLOCK) {
    const int j = e / D, d = e % D;
    float acc = 0.f;
    for (int i = 0; i < N; ++i) {
      acc = fmaf(ls[i][j], ldo[i][d], acc);
    }
    store_from_float(DV, base + e, acc);
  }

  // dP = dO·Vᵀ
  for (int e = tid; e < N * N; e += BLOCK) {
    const int i = e / N, j = e % N;
    float acc = 0.f;
#pragma unroll 8
    for (int d = 0; d < D; ++d) {
      acc = fmaf(ldo[i][d], lv[j][d], acc);
    }
    ldp[i][j] = acc;
  }
  __syncthreads();

  // dS = P ∘ (dP − rowsum(dP ∘ P)); overwrite ldp
  {
    const int r = tid >> 2, sub = tid & 3;
    if (r < N) {
      float rd = 0.f;
      for (int j = sub; j < N; j += 4) rd += ldp[r][j] * ls[r][j];
      rd += __shfl_xor(rd, 1, 64);
      rd += __shfl_xor(rd, 2, 64);
      for (int j = sub; j < N; j += 4) {
        ldp[r][j] = ls[r][j] * (ldp[r][j] - rd);
      }
    }
  }
  __syncthreads();

  // write dS for the bias gradient reduce
  {
    float* ds_out = DS + wh * (int64_t)N * N;
    for (int e = tid; e < N * N; e += BLOCK) {
      ds_out[e] = ldp[e / N][e % N];
    }
  }

  // dQ = scale·dS·K   (K read from lkT transposed: [d][j])
  for (int e = tid; e < N * D; e += BLOCK) {
    const int i = e / D, d = e % D;
    float acc = 0.f;
    for (int j = 0; j < N; ++j) {
      acc = fmaf(ldp[i][j], lkT[d][j], acc);
    }
    store_from_float(DQ, base + e, acc * scale);
  }

  // dK = scale·dSᵀ·Q
  for (int e = tid; e < N * D; e += BLOCK) {
    const int j = e / D, d = e % D;
    float acc = 0.f;
    for (int i = 0; i < N; ++i) {
      acc = fmaf(ldp[i][j], lq[i][d], acc);
    }
    store_from_float(DK, base + e, acc * scale);
  }
}

extern "C" void flreid_window_attn_bwd(const void* Q, const void* K,
                                       const void* V, const float* bias,
                                       const float* mask, const void* DO,
                                       void* DQ, void* DK, void* DV,
                                       float* DS, int64_t BW, int H, int N,
                                       int D, int nW, float scale, int dtype,
                                       hipStream_t stream) {
  if (N > MAX_N || D > MAX_D) {
    throw std::runtime_error("window_attn_bwd: N or D exceeds tile limits");
  }
  constexpr int BLOCK = 256;
  dim3 grid((unsigned)(BW * H)), block(BLOCK);
  if (dtype == kF32) {
    hipLaunchKernelGGL((window_attn_bwd_kernel<float, BLOCK>), grid, block, 0,
                       stream, (const float*)Q, (const float*)K,
                       (const float*)V, bias, mask, (const float*)DO,
                       (float*)DQ, (float*)DK, (float*)DV, DS, BW, H, N, D,
                       nW, scale);
  } else {
    hipLaunchKernelGGL((window_attn_bwd_kernel<__hip_bfloat16, BLOCK>), grid,
                       block, 0, stream, (const __hip_bfloat16*)Q,
                       (const __hip_bfloat16*)K, (const __hip_bfloat16*)V,
                       bias, mask, (const __hip_bfloat16*)DO,
                       (__hip_bfloat16*)DQ, (__hip_bfloat16*)DK,
                       (__hip_bfloat16*)DV, DS, BW, H, N, D, nW, scale);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void flreid_window_attn_fwd(const void* Q, const void* K,
                                       const void* V, const float* bias,
                                       const float* mask, void* O, int64_t BW,
                                       int H, int N, int D, int nW,
                                       float scale, int dtype,
                                       hipStream_t stream) {
  if (N > MAX_N || D > MAX_D) {
    throw std::runtime_error("window_attn_fwd: N or D exceeds tile limits");
  }
  constexpr int BLOCK = 256;
  dim3 grid((unsigned)(BW * H)), block(BLOCK);
  if (dtype == kF32) {
    hipLaunchKernelGGL((window_attn_fwd_kernel<float, BLOCK>), grid, block, 0,
                       stream, (const float*)Q, (const float*)K,
                       (const float*)V, bias, mask, (float*)O, BW, H, N, D,
                       nW, scale);
  } else {
    hipLaunchKernelGGL((window_attn_fwd_kernel<__hip_bfloat16, BLOCK>), grid,
                       block, 0, stream, (const __hip_bfloat16*)Q,
                       (const __hip_bfloat16*)K, (const __hip_bfloat16*)V,
                       bias, mask, (__hip_bfloat16*)O, BW, H, N, D, nW, scale);
  }
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
