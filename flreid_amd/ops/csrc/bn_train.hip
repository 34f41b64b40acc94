// Fused training-mode BatchNorm2d over NHWC rows (K1-adjacent hot op).
//
// MIOpen's train BN on the FedSTIL head-epoch shapes (M = N·H·W = 2048 rows,
// C ∈ {512, 1024, 2048}) costs 5 kernels per layer per direction
// (MeanVariance, FinalMeanVariance, Norm / DScaleDBias, DX) plus
// SubTensorOp casts and an fp32 autocast round-trip of the activation —
// measured ≈900 µs per training step for the 10 layer4 BNs
// (profiles/README.md).  This replaces each direction with TWO kernels:
//
//   k1  partial:   grid (C/64, NSLAB) — every block reduces its row slab to
//                  64-channel partial sums (LDS tree), written densely to a
//                  workspace [NSLAB, C] (no atomics → deterministic, and no
//                  zeroing pass: every slot is overwritten every call).
//   k2  finalize:  same grid — every block re-reduces the (tiny) partial
//                  array for its channel group, then normalizes its own
//                  slab; the slab-0 block additionally writes save_mean /
//                  save_invstd and the running-stat update (fused — no
//                  SubTensorOp casts).  bf16 activations never round-trip
//                  through fp32 HBM copies.
//
// A first single-kernel version (one block per 64 channels doing the whole
// column reduction) filled only C/64 ≤ 32 of the 256 CUs and measured SLOWER
// than the MIOpen chain — the split-row grid is the fix: C/64 × NSLAB blocks
// cover the chip and every traversal runs at full bandwidth.
//
// Layout: x viewed as [M, C] NHWC rows; each thread owns 4 consecutive
// channels (8 B vector loads) × its block's row stripes.
//
// Semantics match torch.nn.BatchNorm2d(train): biased batch var for the
// normalisation, unbiased (·M/(M−1)) for the running-var update
// (ref models the same torch module family: ref:models/resnet.py).

#include "common.h"

namespace flreid {

template <typename T>
__device__ __forceinline__ void bn_load4(const T* p, float v[4]);

template <>
__device__ __forceinline__ void bn_load4<float>(const float* p, float v[4]) {
  const float4 t = *reinterpret_cast<const float4*>(p);
  v[0] = t.x; v[1] = t.y; v[2] = t.z; v[3] = t.w;
}

template <>
__device__ __forceinline__ void bn_load4<__hip_bfloat16>(
    const __hip_bfloat16* p, float v[4]) {
  const ushort4 t = *reinterpret_cast<const ushort4*>(p);
  v[0] = __bfloat162float(*(const __hip_bfloat16*)&t.x);
  v[1] = __bfloat162float(*(const __hip_bfloat16*)&t.y);
  v[2] = __bfloat162float(*(const __hip_bfloat16*)&t.z);
  v[3] = __bfloat162float(*(const __hip_bfloat16*)&t.w);
}

template <typename T>
__device__ __forceinline__ void bn_store4(T* p, const float v[4]);

template <>
__device__ __forceinline__ void bn_store4<float>(float* p, const float v[4]) {
  *reinterpret_cast<float4*>(p) = float4{v[0], v[1], v[2], v[3]};
}

template <>
__device__ __forceinline__ void bn_store4<__hip_bfloat16>(
    __hip_bfloat16* p, const float v[4]) {
  ushort4 t;
  __hip_bfloat16 b;
  b = __float2bfloat16(v[0]); t.x = *(const unsigned short*)&b;
  b = __float2bfloat16(v[1]); t.y = *(const unsigned short*)&b;
  b = __float2bfloat16(v[2]); t.z = *(const unsigned short*)&b;
  b = __float2bfloat16(v[3]); t.w = *(const unsigned short*)&b;
  *reinterpret_cast<ushort4*>(p) = t;
}

constexpr int BN_STRIPES = 16;   // row stripes per block
constexpr int BN_CG = 64;        // channels per block

// Two per-channel f32 accumulators reduced across a block's stripes:
// fwd uses (Σx, Σx²); bwd uses (Σdy, Σ dy·x̂).
template <typename T, bool IS_BWD>
__global__ __launch_bounds__(256) void bn_partial_kernel(
    const T* __restrict__ X, const T* __restrict__ DY,
    const float* __restrict__ smean, const float* __restrict__ sinv,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ part_a, float* __restrict__ part_b, int64_t M,
    int C, int relu) {
  const int c0 = blockIdx.x * BN_CG;
  const int q = threadIdx.x & 15;       // channel quad within the group
  const int r = threadIdx.x >> 4;       // stripe within the block
  const int c = c0 + q * 4;
  const int S = gridDim.y * BN_STRIPES; // total stripes across slabs
  const int g = blockIdx.y * BN_STRIPES + r;

  __shared__ float la[BN_STRIPES][BN_CG + 4];
  __shared__ float lb[BN_STRIPES][BN_CG + 4];

  float mn[4], iv[4], sc[4], sh[4];
  if (IS_BWD) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      mn[j] = smean[c + j];
      iv[j] = sinv[c + j];
      if (relu) {                    // recompute pre-relu output sign
        sc[j] = gamma[c + j] * iv[j];
        sh[j] = beta[c + j] - mn[j] * sc[j];
      }
    }
  }

  float sa[4] = {0.f, 0.f, 0.f, 0.f};
  float sb[4] = {0.f, 0.f, 0.f, 0.f};
  for (int64_t row = g; row < M; row += S) {
    float x[4];
    bn_load4(X + row * C + c, x);
    if (IS_BWD) {
      float dy[4];
      bn_load4(DY + row * C + c, dy);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        if (relu && x[j] * sc[j] + sh[j] <= 0.f) continue;  // dy masked by y>0
        sa[j] += dy[j];
        sb[j] += dy[j] * (x[j] - mn[j]) * iv[j];
      }
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        sa[j] += x[j];
        sb[j] += x[j] * x[j];
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    la[r][q * 4 + j] = sa[j];
    lb[r][q * 4 + j] = sb[j];
  }
  __syncthreads();
  if (threadIdx.x < BN_CG) {
    float ta = 0.f, tb = 0.f;
#pragma unroll
    for (int i = 0; i < BN_STRIPES; ++i) {
      ta += la[i][threadIdx.x];
      tb += lb[i][threadIdx.x];
    }
    const int64_t w = (int64_t)blockIdx.y * C + c0 + threadIdx.x;
    part_a[w] = ta;
    part_b[w] = tb;
  }
}

// ---------------------------------------------------------------- forward
template <typename T>
__global__ __launch_bounds__(256) void bn_finalize_fwd_kernel(
    const T* __restrict__ X, T* __restrict__ Y,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ rmean, float* __restrict__ rvar,
    float* __restrict__ smean, float* __restrict__ sinv,
    const float* __restrict__ part_a, const float* __restrict__ part_b,
    unsigned long long* __restrict__ nbt, int64_t M, int C, float momentum,
    float eps, float unbiased, int relu) {
  const int c0 = blockIdx.x * BN_CG;
  const int q = threadIdx.x & 15;
  const int r = threadIdx.x >> 4;
  const int c = c0 + q * 4;
  const int S = gridDim.y * BN_STRIPES;
  const int g = blockIdx.y * BN_STRIPES + r;
  const int nslab = gridDim.y;

  // num_batches_tracked update fused in (torch increments it per train call)
  if (nbt != nullptr && blockIdx.x == 0 && blockIdx.y == 0 &&
      threadIdx.x == 0) {
    ++(*nbt);
  }

  __shared__ float lmean[BN_CG], linv[BN_CG];
  if (threadIdx.x < BN_CG) {
    float ta = 0.f, tb = 0.f;
    for (int s = 0; s < nslab; ++s) {
      ta += part_a[(int64_t)s * C + c0 + threadIdx.x];
      tb += part_b[(int64_t)s * C + c0 + threadIdx.x];
    }
    const float inv_m = 1.f / (float)M;
    const float mean = ta * inv_m;
    const float var = fmaxf(tb * inv_m - mean * mean, 0.f);
    const float inv = rsqrtf(var + eps);
    lmean[threadIdx.x] = mean;
    linv[threadIdx.x] = inv;
    if (blockIdx.y == 0) {
      const int gc = c0 + threadIdx.x;
      smean[gc] = mean;
      sinv[gc] = inv;
      if (rmean != nullptr) {
        rmean[gc] = (1.f - momentum) * rmean[gc] + momentum * mean;
        rvar[gc] = (1.f - momentum) * rvar[gc] + momentum * var * unbiased;
      }
    }
  }
  __syncthreads();

  float scale[4], shift[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    scale[j] = gamma[c + j] * linv[q * 4 + j];
    shift[j] = beta[c + j] - lmean[q * 4 + j] * scale[j];
  }
  for (int64_t row = g; row < M; row += S) {
    float v[4];
    bn_load4(X + row * C + c, v);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v[j] = v[j] * scale[j] + shift[j];
      if (relu) v[j] = fmaxf(v[j], 0.f);
    }
    bn_store4(Y + row * C + c, v);
  }
}

// --------------------------------------------------------------- backward
// dx = γ·invstd·(dy − Σdy/M − x̂·Σ(dy·x̂)/M);  dγ = Σ(dy·x̂);  dβ = Σdy
template <typename T>
__global__ __launch_bounds__(256) void bn_finalize_bwd_kernel(
    const T* __restrict__ X, const T* __restrict__ DY, T* __restrict__ DX,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ smean, const float* __restrict__ sinv,
    float* __restrict__ dgamma, float* __restrict__ dbeta,
    const float* __restrict__ part_a, const float* __restrict__ part_b,
    int64_t M, int C, int relu) {
  const int c0 = blockIdx.x * BN_CG;
  const int q = threadIdx.x & 15;
  const int r = threadIdx.x >> 4;
  const int c = c0 + q * 4;
  const int S = gridDim.y * BN_STRIPES;
  const int g = blockIdx.y * BN_STRIPES + r;
  const int nslab = gridDim.y;

  __shared__ float lmdy[BN_CG], lmdyx[BN_CG];
  if (threadIdx.x < BN_CG) {
    float tdy = 0.f, tdyx = 0.f;
    for (int s = 0; s < nslab; ++s) {
      tdy += part_a[(int64_t)s * C + c0 + threadIdx.x];
      tdyx += part_b[(int64_t)s * C + c0 + threadIdx.x];
    }
    if (blockIdx.y == 0) {
      const int gc = c0 + threadIdx.x;
      dgamma[gc] = tdyx;
      dbeta[gc] = tdy;
    }
    const float inv_m = 1.f / (float)M;
    lmdy[threadIdx.x] = tdy * inv_m;
    lmdyx[threadIdx.x] = tdyx * inv_m;
  }
  __syncthreads();

  float g_iv[4], mn[4], iv[4], mdy[4], mdyx[4], sh[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    mn[j] = smean[c + j];
    iv[j] = sinv[c + j];
    g_iv[j] = gamma[c + j] * iv[j];
    mdy[j] = lmdy[q * 4 + j];
    mdyx[j] = lmdyx[q * 4 + j];
    if (relu) sh[j] = beta[c + j] - mn[j] * g_iv[j];
  }
  for (int64_t row = g; row < M; row += S) {
    float x[4], dy[4], dx[4];
    bn_load4(X + row * C + c, x);
    bn_load4(DY + row * C + c, dy);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float xhat = (x[j] - mn[j]) * iv[j];
      float d = dy[j];
      if (relu && x[j] * g_iv[j] + sh[j] <= 0.f) d = 0.f;
      dx[j] = g_iv[j] * (d - mdy[j] - xhat * mdyx[j]);
    }
    bn_store4(DX + row * C + c, dx);
  }
}

// ----------------------------------------------------------------- launch
static int bn_nslab(int64_t M, int C) {
  // target ≈256 blocks (one wave over the 8 XCDs); never more slabs than
  // 16-row stripes available
  const int cg = C / BN_CG;
  int ns = 256 / cg;
  const int max_rows = (int)((M + BN_STRIPES - 1) / BN_STRIPES);
  if (ns > max_rows) ns = max_rows;
  if (ns > 64) ns = 64;
  if (ns < 1) ns = 1;
  return ns;
}

extern "C" int flreid_bn_train_nslab(int64_t M, int C) {
  return bn_nslab(M, C);
}

extern "C" void flreid_bn_train_fwd(const void* X, void* Y, const float* gamma,
                                    const float* beta, float* rmean,
                                    float* rvar, float* smean, float* sinv,
                                    float* part_a, float* part_b,
                                    unsigned long long* nbt, int64_t M, int C,
                                    float momentum, float eps, float unbiased,
                                    int relu, int dtype, hipStream_t stream) {
  if (C % BN_CG != 0) throw std::runtime_error("bn_train: C % 64 != 0");
  dim3 grid(C / BN_CG, bn_nslab(M, C)), block(256);
  if (dtype == kBF16) {
    hipLaunchKernelGGL((bn_partial_kernel<__hip_bfloat16, false>), grid,
                       block, 0, stream, (const __hip_bfloat16*)X, nullptr,
                       nullptr, nullptr, nullptr, nullptr, part_a, part_b, M,
                       C, 0);
    hipLaunchKernelGGL(bn_finalize_fwd_kernel<__hip_bfloat16>, grid, block, 0,
                       stream, (const __hip_bfloat16*)X, (__hip_bfloat16*)Y,
                       gamma, beta, rmean, rvar, smean, sinv, part_a, part_b,
                       nbt, M, C, momentum, eps, unbiased, relu);
  } else {
    hipLaunchKernelGGL((bn_partial_kernel<float, false>), grid, block, 0,
                       stream, (const float*)X, nullptr, nullptr, nullptr,
                       nullptr, nullptr, part_a, part_b, M, C, 0);
    hipLaunchKernelGGL(bn_finalize_fwd_kernel<float>, grid, block, 0, stream,
                       (const float*)X, (float*)Y, gamma, beta, rmean, rvar,
                       smean, sinv, part_a, part_b, nbt, M, C, momentum, eps,
                       unbiased, relu);
  }
  HIP_CHECK(hipGetLastError());
}

extern "C" void flreid_bn_train_bwd(const void* X, const void* DY, void* DX,
                                    const float* gamma, const float* beta,
                                    const float* smean, const float* sinv,
                                    float* dgamma, float* dbeta,
                                    float* part_a, float* part_b, int64_t M,
                                    int C, int relu, int dtype,
                                    hipStream_t stream) {
  if (C % BN_CG != 0) throw std::runtime_error("bn_train: C % 64 != 0");
  dim3 grid(C / BN_CG, bn_nslab(M, C)), block(256);
  if (dtype == kBF16) {
    hipLaunchKernelGGL((bn_partial_kernel<__hip_bfloat16, true>), grid, block,
                       0, stream, (const __hip_bfloat16*)X,
                       (const __hip_bfloat16*)DY, smean, sinv, gamma, beta,
                       part_a, part_b, M, C, relu);
    hipLaunchKernelGGL(bn_finalize_bwd_kernel<__hip_bfloat16>, grid, block, 0,
                       stream, (const __hip_bfloat16*)X,
                       (const __hip_bfloat16*)DY, (__hip_bfloat16*)DX, gamma,
                       beta, smean, sinv, dgamma, dbeta, part_a, part_b, M, C,
                       relu);
  } else {
    hipLaunchKernelGGL((bn_partial_kernel<float, true>), grid, block, 0,
                       stream, (const float*)X, (const float*)DY, smean, sinv,
                       gamma, beta, part_a, part_b, M, C, relu);
    hipLaunchKernelGGL(bn_finalize_bwd_kernel<float>, grid, block, 0, stream,
                       (const float*)X, (const float*)DY, (float*)DX, gamma,
                       beta, smean, sinv, dgamma, dbeta, part_a, part_b, M, C,
                       relu);
  }
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
