// Fused adaptive-linear forward (K2 in SURVEY.md §2.9):
//   y[m, n] = Σ_k x[m, k] · (atten[k]·gw[n, k] + aw[n, k])  (+ bias[n])
//
// The FedSTIL composition runs in the WEIGHT FETCH: gw/aw stream from HBM
// exactly once, θ is composed into LDS as bf16 and never touches HBM.  The
// eager chain (compose kernel -> autocast cast -> hipBLASLt GEMM) moves
// ~2.5× the bytes (θ written fp32, re-read, re-written bf16, re-read).
//
// bf16 MFMA (v_mfma_f32_16x16x32_bf16), fp32 accumulate.  Shapes: x [M, K]
// bf16, gw/aw [N, K] fp32, atten [K] fp32 (or null -> plain linear),
// bias [N] fp32 (or null), out [M, N] bf16.  K % 32 == 0.
//
// Layout note: the 16x16x32 A/B fragment holds 8 bf16 per lane as TWO
// 4-element groups (k = (lane>>4)*4 + e for e<4, k = 16 + (lane>>4)*4 + e-4
// for e>=4) — verified against the eager reference on gfx950
// (tests/test_ops_gpu.py::test_adaptive_linear_fused).

#include "common.h"

namespace flreid {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int AG_BM = 64;     // M tile (4 m-fragments of 16)
constexpr int AG_BN = 32;     // N tile (2 waves × one 16-col fragment)
constexpr int AG_BK = 32;     // K tile
constexpr int AG_PAD = 2;     // bf16 pad per LDS row

template <bool SPLIT_K_GROUPS>
__global__ __launch_bounds__(128) void adaptive_linear_fwd_kernel(
    const __hip_bfloat16* __restrict__ X, const float* __restrict__ GW,
    const float* __restrict__ AW, const float* __restrict__ ATTEN,
    const float* __restrict__ BIAS, __hip_bfloat16* __restrict__ OUT,
    int M, int N, int K) {
  __shared__ __hip_bfloat16 lx[AG_BM][AG_BK + AG_PAD];
  __shared__ __hip_bfloat16 lth[AG_BN][AG_BK + AG_PAD];

  const int m0 = blockIdx.x * AG_BM;
  const int n0 = blockIdx.y * AG_BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;            // 2 waves, one 16-col frag each
  const int fn = wave * 16 + (lane & 15);   // fragment col within block

  f32x4 acc[4] = {{}, {}, {}, {}};

  for (int k0 = 0; k0 < K; k0 += AG_BK) {
    // stage x tile: 128 threads × 16 elems = 64×32 bf16 (bf16x4 loads)
    {
      const int lc4 = (tid & 7) * 4;    // 8 threads per 32-elem row
      const int lr0 = tid >> 3;         // 16 rows per pass
#pragma unroll
      for (int r = 0; r < AG_BM; r += 16) {
        const int mr = m0 + lr0 + r;
        if (mr < M) {
          const __hip_bfloat16* src = X + (int64_t)mr * K + k0 + lc4;
          lx[lr0 + r][lc4 + 0] = src[0];
          lx[lr0 + r][lc4 + 1] = src[1];
          lx[lr0 + r][lc4 + 2] = src[2];
          lx[lr0 + r][lc4 + 3] = src[3];
        } else {
          lx[lr0 + r][lc4 + 0] = __float2bfloat16(0.f);
          lx[lr0 + r][lc4 + 1] = __float2bfloat16(0.f);
          lx[lr0 + r][lc4 + 2] = __float2bfloat16(0.f);
          lx[lr0 + r][lc4 + 3] = __float2bfloat16(0.f);
        }
      }
    }
    // compose θ tile into LDS: 32 rows × 32 k = 1024 elems / 128 thr = 8 each
    {
      const int lc4 = (tid & 7) * 4;
      const int lr0 = tid >> 3;         // 16 rows per pass, 2 passes
#pragma unroll
      for (int r = 0; r < AG_BN; r += 16) {
        const int nr = n0 + lr0 + r;
        if (nr < N) {
          const int64_t base = (int64_t)nr * K + k0 + lc4;
#pragma unroll
          for (int c = 0; c < 4; ++c) {
            const float a = ATTEN ? ATTEN[k0 + lc4 + c] : 1.0f;
            lth[lr0 + r][lc4 + c] =
                __float2bfloat16(fmaf(a, GW[base + c], AW ? AW[base + c] : 0.f));
          }
        } else {
#pragma unroll
          for (int c = 0; c < 4; ++c) {
            lth[lr0 + r][lc4 + c] = __float2bfloat16(0.f);
          }
        }
      }
    }
    __syncthreads();

    // fragments: lane reads 8 bf16 of row (l&15) at the k-offsets of its
    // lane group; SPLIT_K_GROUPS selects the two-4-group vs consecutive-8 map
    const int kg = (lane >> 4) * 4;
    bf16x8 bfrag;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int kk = SPLIT_K_GROUPS ? (e < 4 ? kg + e : 16 + kg + e - 4)
                                    : kg * 2 + e;
      bfrag[e] = *reinterpret_cast<const __bf16*>(&lth[fn][kk]);
    }
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
      bf16x8 afrag;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int kk = SPLIT_K_GROUPS ? (e < 4 ? kg + e : 16 + kg + e - 4)
                                      : kg * 2 + e;
        afrag[e] = *reinterpret_cast<const __bf16*>(&lx[mf * 16 + (lane & 15)][kk]);
      }
      acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mf], 0, 0, 0);
    }
    __syncthreads();
  }

  // C/D map: col = lane&15, row = (lane>>4)*4 + reg
  const int nc = n0 + fn;
  if (nc >= N) return;
  const float bias_v = BIAS ? BIAS[nc] : 0.0f;
#pragma unroll
  for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int mr = m0 + mf * 16 + (lane >> 4) * 4 + reg;
      if (mr < M) {
        OUT[(int64_t)mr * N + nc] = __float2bfloat16(acc[mf][reg] + bias_v);
      }
    }
  }
}

extern "C" void flreid_adaptive_linear_fwd(
    const void* X, const float* GW, const float* AW, const float* ATTEN,
    const float* BIAS, void* OUT, int M, int N, int K, int split_layout,
    hipStream_t stream) {
  if (K % AG_BK != 0) {
    throw std::runtime_error("adaptive_linear_fwd: K must be a multiple of 32");
  }
  dim3 grid((M + AG_BM - 1) / AG_BM, (N + AG_BN - 1) / AG_BN);
  dim3 block(128);
  if (split_layout) {
    hipLaunchKernelGGL((adaptive_linear_fwd_kernel<true>), grid, block, 0,
                       stream, (const __hip_bfloat16*)X, GW, AW, ATTEN, BIAS,
                       (__hip_bfloat16*)OUT, M, N, K);
  } else {
    hipLaunchKernelGGL((adaptive_linear_fwd_kernel<false>), grid, block, 0,
                       stream, (const __hip_bfloat16*)X, GW, AW, ATTEN, BIAS,
                       (__hip_bfloat16*)OUT, M, N, K);
  }
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
