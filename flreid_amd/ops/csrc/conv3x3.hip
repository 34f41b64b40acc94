// Hand-written 3×3 stride-1 pad-1 NHWC conv forward (K1 in SURVEY.md §2.9).
//
// Implicit GEMM, slab-staged: per input-channel tile the block stages its
// output rows' x slab WITH HALO ([rows+2][W+2] zero-padded image window)
// into LDS ONCE and reuses it for all nine taps — a tap is just a constant
// LDS offset (idx + r·(W+2) + s).  Only the weights stream per (tap, ck),
// double-buffered with a register ring two phases deep (HBM latency ≈ two
// 8-MFMA phases).  x is L2-resident at ReID shapes; w streams once.
//
// Evolution (PMC-driven, see profiles/): naive shifted-GEMM loop 135 TF
// (23 VALU/MFMA: per-tile div/mod + 64-bit address math + 2 B LDS stores) →
// hoisted addressing + 8 B stores 268 TF → this slab structure.
//
// Constraints: stride 1, pad 1, C % 32 == 0, K % 16 == 0, (H·W) % 128 == 0
// (a block never crosses an image boundary); the dispatch falls back
// otherwise.  x bf16 NHWC; w fp32 channels-last [K, 3, 3, C]; y bf16 NHWC.

#include "common.h"

namespace flreid {

using cbf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using cf32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int CV_BM = 128;   // output rows per block (8 m-fragments)
constexpr int CV_BN = 64;    // output channels per block (4 waves × 16)
constexpr int CV_BK = 32;    // input-channel tile
constexpr int CV_PAD = 4;    // bf16 pad -> 72 B row stride (8 B aligned)
constexpr int MAX_SLAB = 288;  // max (rows_img+2)(W+2) slab rows

struct WPrefetch {
  float4 wr[2];
};

__global__ __launch_bounds__(256) void conv3x3_fwd_kernel(
    const __hip_bfloat16* __restrict__ X, const float* __restrict__ W,
    __hip_bfloat16* __restrict__ Y, int NB, int H, int Wd, int C, int K,
    int slab_rows) {
  extern __shared__ __hip_bfloat16 smem[];
  // layout: slab [slab_rows][CV_BK+PAD] | w ring 3 × [CV_BN][CV_BK+PAD]
  const int row_stride = CV_BK + CV_PAD;
  __hip_bfloat16* slab = smem;
  __hip_bfloat16* wbuf = smem + (int64_t)slab_rows * row_stride;

  const int m0 = blockIdx.x * CV_BM;
  const int k0c = blockIdx.y * CV_BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int fn = wave * 16 + (lane & 15);

  const int lc4 = (tid & 7) * 4;
  const int lr0 = tid >> 3;                  // 32 slab rows per pass

  cf32x4 acc[8] = {{}, {}, {}, {}, {}, {}, {}, {}};

  // ---- hoisted slab-row decode (fixed across ck) -------------------------
  // block covers image rows [h0, h0 + CV_BM/Wd); slab adds ±1 row + w pad
  const int hw0 = m0 % (H * Wd);
  const int h0 = hw0 / Wd;                   // hw0 % Wd == 0 by constraint
  const int wp = Wd + 2;
  // this thread stages slab rows lr0, lr0+32, ... : precompute pointers
  const int n_passes = (slab_rows + 31) / 32;
  const __hip_bfloat16* srow_ptr[MAX_SLAB / 32 + 1];
  bool srow_ok[MAX_SLAB / 32 + 1];
#pragma unroll 4
  for (int p = 0; p < n_passes; ++p) {
    const int sr = lr0 + 32 * p;
    const int hh = h0 - 1 + sr / wp;
    const int ww = sr % wp - 1;
    const bool ok = sr < slab_rows && hh >= 0 && hh < H && ww >= 0 && ww < Wd;
    srow_ok[p] = ok;
    const int m_img = m0 / (H * Wd) * (H * Wd);     // image base row
    srow_ptr[p] = X + ((int64_t)m_img + (int64_t)hh * Wd + ww) * C + lc4;
  }

  const float* wrow_ptr[2];
  bool wlive[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int kk = k0c + lr0 + 32 * i;
    wlive[i] = kk < K;
    wrow_ptr[i] = W + (int64_t)(wlive[i] ? kk : 0) * 9 * C + lc4;
  }

  // fragment A row base: output row mr = m0 + mf*16 + (lane&15 of the
  // 16-row fragment...) -> slab index (h-h0+... ) precompute per mf
  int a_base[8];
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
    const int mr_loc = mf * 16 + (lane & 15);      // 0..127 within block
    a_base[mf] = (mr_loc / Wd) * wp + mr_loc % Wd; // + r*wp + s at tap time
  }

  const int n_ctiles = C / CV_BK;

  auto load_w = [&](int shift, int ck, WPrefetch& p) {
    const int wo = shift * C + ck;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      p.wr[i] = wlive[i] ? *(const float4*)(wrow_ptr[i] + wo)
                         : float4{0.f, 0.f, 0.f, 0.f};
    }
  };

  auto store_w = [&](const WPrefetch& p, int buf) {
    __hip_bfloat16* base = wbuf + (int64_t)buf * CV_BN * row_stride;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const __hip_bfloat16 b0 = __float2bfloat16(p.wr[i].x);
      const __hip_bfloat16 b1 = __float2bfloat16(p.wr[i].y);
      const __hip_bfloat16 b2 = __float2bfloat16(p.wr[i].z);
      const __hip_bfloat16 b3 = __float2bfloat16(p.wr[i].w);
      ushort4 v = {*(const unsigned short*)&b0, *(const unsigned short*)&b1,
                   *(const unsigned short*)&b2, *(const unsigned short*)&b3};
      *(ushort4*)&base[(lr0 + 32 * i) * row_stride + lc4] = v;
    }
  };

  auto stage_slab = [&](int ck) {
#pragma unroll 4
    for (int p = 0; p < n_passes; ++p) {
      const int sr = lr0 + 32 * p;
      if (sr < slab_rows) {
        const ushort4 v = srow_ok[p] ? *(const ushort4*)(srow_ptr[p] + ck)
                                     : ushort4{0, 0, 0, 0};
        *(ushort4*)&slab[sr * row_stride + lc4] = v;
      }
    }
  };

  // prologue: slab(ck=0), W phases 0 and 1
  WPrefetch wring[2];
  stage_slab(0);
  load_w(0, 0, wring[0]);
  store_w(wring[0], 0);
  load_w(1, 0, wring[1]);
  store_w(wring[1], 1);
  __syncthreads();

  const int kg = (lane >> 4) * 4;
  const int NT = 9 * n_ctiles;               // phases, shift-major per ck
  int shift = 0, ck = 0;
  // prefetch cursor two phases ahead
  int pf_shift = 2, pf_ck = 0;
  if (pf_shift >= 9) { pf_shift -= 9; pf_ck += CV_BK; }

  for (int t = 0; t < NT; ++t) {
    const int wslot = t % 3;
    // issue W prefetch 2 ahead into slot (t+2)%3
    const bool do_pf = t + 2 < NT;
    WPrefetch pf;
    if (do_pf) {
      load_w(pf_shift, pf_ck, pf);
    }

    // MFMA on (shift, ck): A from slab (+tap offset), B from wbuf[wslot]
    const __hip_bfloat16* wb = wbuf + (int64_t)wslot * CV_BN * row_stride;
    cbf16x8 bfrag;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int kk = e < 4 ? kg + e : 16 + kg + e - 4;
      bfrag[e] = *reinterpret_cast<const __bf16*>(&wb[fn * row_stride + kk]);
    }
    const int tap_off = (shift / 3) * wp + (shift % 3);
#pragma unroll
    for (int mf = 0; mf < 8; ++mf) {
      const __hip_bfloat16* ar = &slab[(a_base[mf] + tap_off) * row_stride];
      cbf16x8 afrag;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int kk = e < 4 ? kg + e : 16 + kg + e - 4;
        afrag[e] = *reinterpret_cast<const __bf16*>(&ar[kk]);
      }
      acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mf],
                                                        0, 0, 0);
    }

    // write the prefetched W into slot (t+2)%3: its last readers ran in
    // phase t-1, separated from here by that phase's barrier
    if (do_pf) {
      store_w(pf, (t + 2) % 3);
      pf_shift += 1;
      if (pf_shift == 9) { pf_shift = 0; pf_ck += CV_BK; }
    }
    // restage the slab at ck boundaries (needs a readers-done barrier first)
    if ((shift == 8) && (ck + CV_BK < C)) {
      __syncthreads();
      stage_slab(ck + CV_BK);
    }
    __syncthreads();                          // new tiles visible
    shift += 1;
    if (shift == 9) { shift = 0; ck += CV_BK; }
  }

  const int kc = k0c + fn;
  if (kc >= K) return;
  const int M = NB * H * Wd;
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
  if (C % CV_BK != 0 || K % 16 != 0 || (H * Wd) % CV_BM != 0) {
    throw std::runtime_error(
        "conv3x3_fwd: needs C%32==0, K%16==0, (H*W)%128==0");
  }
  const int rows_img = CV_BM / Wd;            // image rows per block
  const int slab_rows = (rows_img + 2) * (Wd + 2);
  if (slab_rows > MAX_SLAB) {
    throw std::runtime_error("conv3x3_fwd: W too large for the slab");
  }
  const int row_stride = CV_BK + CV_PAD;
  const size_t lds = ((size_t)slab_rows + 3 * CV_BN) * row_stride *
                     sizeof(__hip_bfloat16);
  const int M = NB * H * Wd;
  dim3 grid((M + CV_BM - 1) / CV_BM, (K + CV_BN - 1) / CV_BN);
  dim3 block(256);
  hipLaunchKernelGGL(conv3x3_fwd_kernel, grid, block, lds, stream,
                     (const __hip_bfloat16*)X, W, (__hip_bfloat16*)Y, NB, H,
                     Wd, C, K, slab_rows);
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
