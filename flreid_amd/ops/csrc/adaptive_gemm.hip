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
// Pipeline: register-prefetch of the NEXT K-tile issued before the current
// tile's MFMAs (guide T14 — the loads' s_waitcnt lands at the LDS write
// after the compute), double-buffered LDS, one barrier per K-tile.  The
// un-pipelined version measured 0.85 TB/s effective (latency-bound).
//
// Fragment layout (verified on gfx950): lane holds 8 bf16 as TWO 4-element
// k-groups: k = (lane>>4)*4 + e (e<4) and k = 16 + (lane>>4)*4 + (e-4).

#include "common.h"

namespace flreid {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int AG_BM = 64;     // M tile (4 m-fragments of 16)
constexpr int AG_BN = 32;     // N tile (2 waves × one 16-col fragment)
constexpr int AG_BK = 32;     // K tile
constexpr int AG_PAD = 2;     // bf16 pad per LDS row

struct Prefetch {
  ushort4 xr[4];              // 4×(4 bf16) of the x tile
  float4 gr[2], ar[2];        // 2×float4 of gw / aw
};

__global__ __launch_bounds__(128) void adaptive_linear_fwd_kernel(
    const __hip_bfloat16* __restrict__ X, const float* __restrict__ GW,
    const float* __restrict__ AW, const float* __restrict__ ATTEN,
    const float* __restrict__ BIAS, __hip_bfloat16* __restrict__ OUT,
    int M, int N, int K) {
  __shared__ __hip_bfloat16 lx[2][AG_BM][AG_BK + AG_PAD];
  __shared__ __hip_bfloat16 lth[2][AG_BN][AG_BK + AG_PAD];

  const int m0 = blockIdx.x * AG_BM;
  const int n0 = blockIdx.y * AG_BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;            // 2 waves, one 16-col frag each
  const int fn = wave * 16 + (lane & 15);

  const int lc4 = (tid & 7) * 4;        // 8 threads per 32-elem row
  const int lr0 = tid >> 3;             // row 0..15 (16 rows per pass)
  const bool edge = (m0 + AG_BM > M) || (n0 + AG_BN > N);

  f32x4 acc[4] = {{}, {}, {}, {}};

  auto load_tile = [&](int k0, Prefetch& p) {
    if (!edge) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        p.xr[i] = *(const ushort4*)(X + (int64_t)(m0 + lr0 + 16 * i) * K + k0 + lc4);
      }
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int64_t base = (int64_t)(n0 + lr0 + 16 * i) * K + k0 + lc4;
        p.gr[i] = *(const float4*)(GW + base);
        p.ar[i] = AW ? *(const float4*)(AW + base) : float4{0.f, 0.f, 0.f, 0.f};
      }
    } else {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int mr = m0 + lr0 + 16 * i;
        ushort4 v = {0, 0, 0, 0};
        if (mr < M) {
          v = *(const ushort4*)(X + (int64_t)mr * K + k0 + lc4);
        }
        p.xr[i] = v;
      }
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int nr = n0 + lr0 + 16 * i;
        float4 g = {0.f, 0.f, 0.f, 0.f}, a = {0.f, 0.f, 0.f, 0.f};
        if (nr < N) {
          const int64_t base = (int64_t)nr * K + k0 + lc4;
          g = *(const float4*)(GW + base);
          if (AW) a = *(const float4*)(AW + base);
        }
        p.gr[i] = g;
        p.ar[i] = a;
      }
    }
  };

  auto store_tile = [&](int k0, const Prefetch& p, int buf) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __hip_bfloat16* dst = &lx[buf][lr0 + 16 * i][lc4];
      dst[0] = *(const __hip_bfloat16*)&p.xr[i].x;
      dst[1] = *(const __hip_bfloat16*)&p.xr[i].y;
      dst[2] = *(const __hip_bfloat16*)&p.xr[i].z;
      dst[3] = *(const __hip_bfloat16*)&p.xr[i].w;
    }
    float at[4] = {1.f, 1.f, 1.f, 1.f};
    if (ATTEN) {
      at[0] = ATTEN[k0 + lc4 + 0];
      at[1] = ATTEN[k0 + lc4 + 1];
      at[2] = ATTEN[k0 + lc4 + 2];
      at[3] = ATTEN[k0 + lc4 + 3];
    }
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      __hip_bfloat16* dst = &lth[buf][lr0 + 16 * i][lc4];
      dst[0] = __float2bfloat16(fmaf(at[0], p.gr[i].x, p.ar[i].x));
      dst[1] = __float2bfloat16(fmaf(at[1], p.gr[i].y, p.ar[i].y));
      dst[2] = __float2bfloat16(fmaf(at[2], p.gr[i].z, p.ar[i].z));
      dst[3] = __float2bfloat16(fmaf(at[3], p.gr[i].w, p.ar[i].w));
    }
  };

  const int NT = K / AG_BK;
  Prefetch cur_p, next_p;
  load_tile(0, cur_p);
  store_tile(0, cur_p, 0);
  __syncthreads();

  int buf = 0;
  const int kg = (lane >> 4) * 4;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) {
      load_tile((t + 1) * AG_BK, next_p);    // issue early; waited at store
    }

    bf16x8 bfrag;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int kk = e < 4 ? kg + e : 16 + kg + e - 4;
      bfrag[e] = *reinterpret_cast<const __bf16*>(&lth[buf][fn][kk]);
    }
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
      bf16x8 afrag;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int kk = e < 4 ? kg + e : 16 + kg + e - 4;
        afrag[e] = *reinterpret_cast<const __bf16*>(
            &lx[buf][mf * 16 + (lane & 15)][kk]);
      }
      acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mf],
                                                        0, 0, 0);
    }

    if (t + 1 < NT) {
      store_tile((t + 1) * AG_BK, next_p, buf ^ 1);
    }
    __syncthreads();
    buf ^= 1;
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
  (void)split_layout;   // layout resolved (split k-groups); kept for ABI
  if (K % AG_BK != 0) {
    throw std::runtime_error("adaptive_linear_fwd: K must be a multiple of 32");
  }
  dim3 grid((M + AG_BM - 1) / AG_BM, (N + AG_BN - 1) / AG_BN);
  dim3 block(128);
  hipLaunchKernelGGL((adaptive_linear_fwd_kernel), grid, block, 0, stream,
                     (const __hip_bfloat16*)X, GW, AW, ATTEN, BIAS,
                     (__hip_bfloat16*)OUT, M, N, K);
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
