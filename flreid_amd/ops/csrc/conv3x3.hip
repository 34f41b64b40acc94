// Hand-written 3×3 stride-1 pad-1 NHWC conv forward (K1 in SURVEY.md §2.9).
//
// Implicit GEMM as NINE SHIFTED GEMMs: for each kernel tap (r, s),
//   y[m, k] += Σ_c x[m + ((r−1)·W + (s−1))·C, c] · w[k, r, s, c]
// over flattened NHWC rows m = ((n·H + h)·W + w), with per-row border masks
// instead of an im2col buffer.  Same MFMA 16x16x32 bf16 + register-prefetch
// + double-buffered-LDS pipeline as adaptive_gemm.hip.
//
// Covers the frozen-backbone forward (prototype capture / validation — the
// FedSTIL hot eval path); training-side dgrad/wgrad stay on MIOpen this
// round.  x bf16 NHWC [NB, H, W, C], w fp32 channels-last [K, 3, 3, C]
// (torch's memory_format=channels_last conv weight), y bf16 NHWC.
// C % 32 == 0, K % 16 == 0.

#include "common.h"

namespace flreid {

using cbf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using cf32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int CV_BM = 128;   // output rows per block (8 m-fragments)
constexpr int CV_BN = 64;    // output channels per block (4 waves x 16)
constexpr int CV_BK = 32;    // input-channel tile
constexpr int CV_PAD = 2;

struct CvPrefetch {
  ushort4 xr[4];
  float4 wr[2];
};

__global__ __launch_bounds__(256) void conv3x3_fwd_kernel(
    const __hip_bfloat16* __restrict__ X, const float* __restrict__ W,
    __hip_bfloat16* __restrict__ Y, int NB, int H, int Wd, int C, int K) {
  __shared__ __hip_bfloat16 lx[2][CV_BM][CV_BK + CV_PAD];
  __shared__ __hip_bfloat16 lw[2][CV_BN][CV_BK + CV_PAD];
  static_assert(CV_BM == 128 && CV_BN == 64, "geometry assumptions");

  const int m0 = blockIdx.x * CV_BM;
  const int k0c = blockIdx.y * CV_BN;        // output-channel block
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;                 // 4 waves x one 16-col fragment
  const int fn = wave * 16 + (lane & 15);

  const int M = NB * H * Wd;
  const int lc4 = (tid & 7) * 4;
  const int lr0 = tid >> 3;                  // 32 rows / pass

  cf32x4 acc[8] = {{}, {}, {}, {}, {}, {}, {}, {}};

  // decode the 4 output rows this thread stages (shared across shifts)
  int row_h[4], row_w[4];
  int64_t row_base[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int m = m0 + lr0 + 32 * i;
    const int hw = m % (H * Wd);
    row_h[i] = hw / Wd;
    row_w[i] = hw % Wd;
    row_base[i] = (int64_t)m * C;            // NHWC flat row
  }

  const int n_ctiles = C / CV_BK;
  const int NT = 9 * n_ctiles;

  auto load_tile = [&](int t, CvPrefetch& p) {
    const int shift = t / n_ctiles;          // 0..8 tap index
    const int r = shift / 3 - 1;             // -1..1
    const int s = shift % 3 - 1;
    const int ck = (t % n_ctiles) * CV_BK;
    const int64_t xoff = ((int64_t)r * Wd + s) * C + ck + lc4;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int m = m0 + lr0 + 32 * i;
      const int hh = row_h[i] + r;
      const int ww = row_w[i] + s;
      const bool ok = (m < M) && hh >= 0 && hh < H && ww >= 0 && ww < Wd;
      p.xr[i] = ok ? *(const ushort4*)(X + row_base[i] + xoff)
                   : ushort4{0, 0, 0, 0};
    }
    // w[k, r, s, c]: rows k = k0c + lr0 (+32)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int kk = k0c + lr0 + 32 * i;
      float4 v = {0.f, 0.f, 0.f, 0.f};
      if (kk < K) {
        v = *(const float4*)(W + (((int64_t)kk * 3 + (r + 1)) * 3 + (s + 1)) * C
                             + ck + lc4);
      }
      p.wr[i] = v;
    }
  };

  auto store_tile = [&](const CvPrefetch& p, int buf) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __hip_bfloat16* dst = &lx[buf][lr0 + 32 * i][lc4];
      dst[0] = *(const __hip_bfloat16*)&p.xr[i].x;
      dst[1] = *(const __hip_bfloat16*)&p.xr[i].y;
      dst[2] = *(const __hip_bfloat16*)&p.xr[i].z;
      dst[3] = *(const __hip_bfloat16*)&p.xr[i].w;
    }
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      __hip_bfloat16* dst = &lw[buf][lr0 + 32 * i][lc4];
      dst[0] = __float2bfloat16(p.wr[i].x);
      dst[1] = __float2bfloat16(p.wr[i].y);
      dst[2] = __float2bfloat16(p.wr[i].z);
      dst[3] = __float2bfloat16(p.wr[i].w);
    }
  };

  CvPrefetch cur_p, next_p;
  load_tile(0, cur_p);
  store_tile(cur_p, 0);
  __syncthreads();

  int buf = 0;
  const int kg = (lane >> 4) * 4;
  for (int t = 0; t < NT; ++t) {
    if (t + 1 < NT) {
      load_tile(t + 1, next_p);
    }
    cbf16x8 bfrag;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int kk = e < 4 ? kg + e : 16 + kg + e - 4;
      bfrag[e] = *reinterpret_cast<const __bf16*>(&lw[buf][fn][kk]);
    }
#pragma unroll
    for (int mf = 0; mf < 8; ++mf) {
      cbf16x8 afrag;
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
      store_tile(next_p, buf ^ 1);
    }
    __syncthreads();
    buf ^= 1;
  }

  const int kc = k0c + fn;
  if (kc >= K) return;
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int mr = m0 + mf * 16 + (lane >> 4) * 4 + reg;
      if (mr < M) {
        Y[(int64_t)mr * K + kc] = __float2bfloat16(acc[mf][reg]);
      }
    }
  }
}

extern "C" void flreid_conv3x3_fwd(const void* X, const float* W, void* Y,
                                   int NB, int H, int Wd, int C, int K,
                                   hipStream_t stream) {
  if (C % CV_BK != 0 || K % 16 != 0) {
    throw std::runtime_error("conv3x3_fwd: C%32 or K%16 != 0");
  }
  const int M = NB * H * Wd;
  dim3 grid((M + CV_BM - 1) / CV_BM, (K + CV_BN - 1) / CV_BN);
  dim3 block(256);
  hipLaunchKernelGGL(conv3x3_fwd_kernel, grid, block, 0, stream,
                     (const __hip_bfloat16*)X, W, (__hip_bfloat16*)Y, NB, H,
                     Wd, C, K);
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
