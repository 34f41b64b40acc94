// Hand-written 3×3 s1p1 NHWC bf16 conv fwd, TWO images per block (K1).
//
// Evolution of the one-image halo kernel driven by PMC evidence
// (profiles/README.md K1 ladder): at one image per block the per-c-tile
// barrier + staging overhead amortises over only 72 MFMAs/wave
// (SQ_WAIT_ANY showed waves parked ~2.5× their MFMA-busy time).  Staging
// TWO images per block doubles the MFMA burst per barrier (144/wave),
// halves the per-image weight traffic (the 36 KB θ panel serves both),
// and puts the batch-64 training grid at exactly 256 blocks = one per CU.
//
//  - x: per-image zero-padded LDS halos [(H+2)(W+2)][48], double-buffered;
//    stride 48 elems = 24-dword = 8·odd bank stride -> conflict-free
//    ds_read_b128 lane groups; c-chunks stored in the MFMA fragment
//    permutation so one b128 read IS a fragment.
//  - θ (bf16, [K][3][3][C] channels-last, produced by compose2 in one
//    pass): single-buffered [9][64][48] panel, staged through registers,
//    two barriers per c-tile (compute -> overwrite -> visible).
//  - loads are unconditional from clamped addresses, invalid slots select
//    zero at the LDS store (branching around loads serialises them).

#include "common.h"

#include <algorithm>

namespace flreid {

using ibf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using if32x4 = __attribute__((ext_vector_type(4))) float;
using u16x8 = __attribute__((ext_vector_type(8))) unsigned short;

constexpr int CW_CS = 48;          // per-cell c-stride (elems)
constexpr int CW_CELLS = 18 * 10;  // max (H+2)*(W+2)
constexpr int CW_BK = 32;          // input-channel tile
constexpr int CW_BN = 64;          // output channels per block
constexpr int CW_IMGS = 2;         // images per block

__device__ __forceinline__ int cw_pbase(int j) {
  const int base[4] = {0, 16, 4, 20};
  return base[j];
}

template <int MF>
__global__ __launch_bounds__(256, 1) void conv3x3_imgw_fwd_kernel(
    const __hip_bfloat16* __restrict__ X, const __hip_bfloat16* __restrict__ W,
    __hip_bfloat16* __restrict__ Y, int NB, int H, int Wd, int C, int K) {
  __shared__ __hip_bfloat16 lx[2][CW_IMGS][CW_CELLS * CW_CS];
  __shared__ __hip_bfloat16 lw[9 * CW_BN * CW_CS];

  const int k0 = blockIdx.x * CW_BN;       // k-block first: XCD affinity
  const int img0 = blockIdx.y * CW_IMGS;
  const int HW = H * Wd;                   // == MF * 16
  const int Wp = Wd + 2;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int fn = wave * 16 + (lane & 15);
  const int kg = (lane >> 4);

  // ---- zero the x halo buffers once (pads stay zero forever) ----
  {
    const int total = (H + 2) * Wp * CW_CS;
    for (int i = tid; i < total; i += 256) {
#pragma unroll
      for (int b = 0; b < 2; ++b) {
#pragma unroll
        for (int g = 0; g < CW_IMGS; ++g) {
          lx[b][g][i] = __float2bfloat16(0.f);
        }
      }
    }
  }

  // ---- x staging: 2 slots per thread PER IMAGE ----
  const int xs_slots = HW * 4;
  bool x_valid[CW_IMGS][2];
  int64_t x_gaddr[CW_IMGS][2];
  int x_laddr[2];
  {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int slot = tid + 256 * i;
      const int cell = slot >> 2, j = slot & 3;
      const bool in_img = slot < xs_slots;
      if (in_img) {
        const int h = cell / Wd, w = cell % Wd;
        x_laddr[i] = ((h + 1) * Wp + (w + 1)) * CW_CS + cw_pbase(j);
      } else {
        x_laddr[i] = cw_pbase(j);    // pad cell (0,0): zero writes only
      }
#pragma unroll
      for (int g = 0; g < CW_IMGS; ++g) {
        const int img = img0 + g;
        x_valid[g][i] = in_img && img < NB;
        x_gaddr[g][i] = x_valid[g][i]
            ? (int64_t)img * HW * C + (int64_t)cell * C + j * 8
            : (int64_t)0;
      }
    }
  }

  // ---- θ staging: 9 slots (one per tap): k = tid>>2, j = tid&3 ----
  const int wk = tid >> 2, wj = tid & 3;
  const bool w_ok = (k0 + wk) < K;
  const int64_t w_gbase =
      ((int64_t)(w_ok ? (k0 + wk) : 0) * 9) * C + wj * 8;
  const int w_lbase = wk * CW_CS + cw_pbase(wj);

  const int NT = C / CW_BK;

  u16x8 xr[CW_IMGS][2];
  u16x8 wr[9];

  auto load_x = [&](int ct) {
    const int64_t coff = (int64_t)ct * CW_BK;
#pragma unroll
    for (int g = 0; g < CW_IMGS; ++g) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        xr[g][i] = *(const u16x8*)(X + x_gaddr[g][i] + coff);
      }
    }
  };

  auto load_w = [&](int ct) {
    const int64_t coff = (int64_t)ct * CW_BK;
#pragma unroll
    for (int t = 0; t < 9; ++t) {
      wr[t] = *(const u16x8*)(W + w_gbase + (int64_t)t * C + coff);
    }
  };

  auto store_x = [&](int buf) {
#pragma unroll
    for (int g = 0; g < CW_IMGS; ++g) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        __hip_bfloat16* dst = &lx[buf][g][x_laddr[i]];
        const uint64_t lo =
            x_valid[g][i] ? *(const uint64_t*)&xr[g][i] : 0ull;
        const uint64_t hi =
            x_valid[g][i] ? *(((const uint64_t*)&xr[g][i]) + 1) : 0ull;
        *(uint64_t*)dst = lo;
        *(uint64_t*)(dst + 8) = hi;
      }
    }
  };

  auto store_w = [&]() {
#pragma unroll
    for (int t = 0; t < 9; ++t) {
      __hip_bfloat16* dst = &lw[t * CW_BN * CW_CS + w_lbase];
      const uint64_t lo = w_ok ? *(const uint64_t*)&wr[t] : 0ull;
      const uint64_t hi = w_ok ? *(((const uint64_t*)&wr[t]) + 1) : 0ull;
      *(uint64_t*)dst = lo;
      *(uint64_t*)(dst + 8) = hi;
    }
  };

  int mcell[MF];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int m = mf * 16 + (lane & 15);
    const int h = m / Wd, w = m % Wd;
    mcell[mf] = ((h + 1) * Wp + (w + 1)) * CW_CS + 8 * kg;
  }

  if32x4 acc[CW_IMGS][MF] = {};

  // prologue: tile 0 fully staged, tile 1's x in flight
  load_x(0);
  load_w(0);
  __syncthreads();     // zero pass done
  store_x(0);
  store_w();
  if (NT > 1) load_x(1);
  __syncthreads();

  for (int ct = 0; ct < NT; ++ct) {
    if (ct + 1 < NT) {
      store_x((ct + 1) & 1);
      load_w(ct + 1);          // issued early; stored after the compute
      if (ct + 2 < NT) load_x(ct + 2);
    }
    const __hip_bfloat16* wb = lw;
    __builtin_amdgcn_s_setprio(1);   // keep the MFMA burst ahead of the
                                     // co-resident staging waves (guide T5)
#pragma unroll
    for (int r = 0; r < 3; ++r) {
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        const int tap = r * 3 + s;
        const ibf16x8 bfrag = *(const ibf16x8*)(
            wb + tap * CW_BN * CW_CS + fn * CW_CS + 8 * kg);
        const int toff = ((r - 1) * Wp + (s - 1)) * CW_CS;
#pragma unroll
        for (int g = 0; g < CW_IMGS; ++g) {
          const __hip_bfloat16* xb = lx[ct & 1][g];
#pragma unroll
          for (int mf = 0; mf < MF; ++mf) {
            const ibf16x8 afrag = *(const ibf16x8*)(xb + mcell[mf] + toff);
            acc[g][mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag, bfrag, acc[g][mf], 0, 0, 0);
          }
        }
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();           // everyone done reading lw
    if (ct + 1 < NT) {
      store_w();
    }
    __syncthreads();           // next tile's x and θ visible
  }

  const int kc = k0 + fn;
  if (kc >= K) return;
#pragma unroll
  for (int g = 0; g < CW_IMGS; ++g) {
    const int img = img0 + g;
    if (img >= NB) break;
    const int64_t out_base = (int64_t)img * HW * K + kc;
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = mf * 16 + (lane >> 4) * 4 + reg;
        Y[out_base + (int64_t)m * K] = __float2bfloat16(acc[g][mf][reg]);
      }
    }
  }
}

extern "C" void flreid_conv3x3_img_fwd_ldsw(const void* X, const void* W,
                                            void* Y, int NB, int H, int Wd,
                                            int C, int K,
                                            hipStream_t stream) {
  const int HW = H * Wd;
  if (HW > 128 || (HW & 15) || C % CW_BK || K % 16 || H + 2 > 18 ||
      Wd + 2 > 10) {
    throw std::runtime_error("conv3x3_img_fwd_ldsw: shape out of regime");
  }
  dim3 grid((K + CW_BN - 1) / CW_BN, (NB + CW_IMGS - 1) / CW_IMGS);
  const int mf = HW >> 4;
  switch (mf) {
#define FLREID_CW_CASE(MF)                                                   \
  case MF:                                                                   \
    hipLaunchKernelGGL(conv3x3_imgw_fwd_kernel<MF>, grid, dim3(256), 0,      \
                       stream, (const __hip_bfloat16*)X,                     \
                       (const __hip_bfloat16*)W, (__hip_bfloat16*)Y, NB, H,  \
                       Wd, C, K);                                            \
    break;
    FLREID_CW_CASE(1)
    FLREID_CW_CASE(2)
    FLREID_CW_CASE(3)
    FLREID_CW_CASE(4)
    FLREID_CW_CASE(5)
    FLREID_CW_CASE(6)
    FLREID_CW_CASE(7)
    FLREID_CW_CASE(8)
#undef FLREID_CW_CASE
    default:
      throw std::runtime_error("conv3x3_img_fwd_ldsw: bad MF");
  }
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
