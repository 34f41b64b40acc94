// v1 A/B variant of the specialized conv fwd (kept for measurement): both
// x AND the plain bf16 θ [K][3][3][C] staged through LDS (121 KB -> one
// block/CU) with de-serialized clamped loads.  The production kernel
// (conv3x3_img.hip) streams weights per-lane from the pre-tiled layout
// instead; benchmarks/conv_img_probe.py A/Bs the two.

#include "common.h"

#include <algorithm>

namespace flreid {

using ibf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using if32x4 = __attribute__((ext_vector_type(4))) float;
using u16x8 = __attribute__((ext_vector_type(8))) unsigned short;

constexpr int CW_CS = 48;        // padded per-cell c-stride: 24-dword
                                 // bank stride = 8*odd -> conflict-free
                                 // ds_read_b128 lane groups (PMC: 9.7
                                 // conflict-cycles/MFMA at stride 40)
constexpr int CW_CELLS = 18 * 10;  // max (H+2)*(W+2)
constexpr int CW_BK = 32;        // input-channel tile
constexpr int CW_BN = 64;        // output channels per block

// LDS c-permutation: fragment phys chunk base for c-chunk j (c = 8j..8j+7)
__device__ __forceinline__ int cw_pbase(int j) {
  // j=0 -> {0..3, 8..11}; j=1 -> {16..19, 24..27}; j=2 -> {4..7, 12..15};
  // j=3 -> {20..23, 28..31}   (see p(c) above)
  const int base[4] = {0, 16, 4, 20};
  return base[j];
}

template <int MF>
__global__ __launch_bounds__(256, 1) void conv3x3_imgw_fwd_kernel(
    const __hip_bfloat16* __restrict__ X, const __hip_bfloat16* __restrict__ W,
    __hip_bfloat16* __restrict__ Y, int H, int Wd, int C, int K) {
  __shared__ __hip_bfloat16 lx[2][CW_CELLS * CW_CS];
  __shared__ __hip_bfloat16 lw[2][9 * CW_BN * CW_CS];

  const int k0 = blockIdx.x * CW_BN;      // k-block first: XCD affinity
  const int img = blockIdx.y;
  const int HW = H * Wd;                  // == MF * 16
  const int Wp = Wd + 2;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int fn = wave * 16 + (lane & 15); // this wave's output channel row
  const int kg = (lane >> 4);             // k-group 0..3 (chunk index)

  // ---- zero the x halo buffers once (pads stay zero forever) ----
  {
    const int total = (H + 2) * Wp * CW_CS;
    for (int i = tid; i < total; i += 256) {
      lx[0][i] = __float2bfloat16(0.f);
      lx[1][i] = __float2bfloat16(0.f);
    }
  }

  // ---- per-thread staging slots ----
  // Loads are UNCONDITIONAL from clamped addresses; invalid slots select
  // zero at the LDS store.  Branching around each load makes hipcc emit a
  // vmcnt(0) wait per load — serial L2 round trips (guide §5 trap c).
  // x: slot = cell*4 + j  (cell = output pixel, j = 16B c-chunk)
  const int xs_slots = HW * 4;
  bool x_valid[2];
  int64_t x_gaddr[2];
  int x_laddr[2];
  {
    const int64_t img_base = (int64_t)img * HW * C;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int slot = tid + 256 * i;
      const int cell = slot >> 2, j = slot & 3;
      x_valid[i] = slot < xs_slots;
      if (x_valid[i]) {
        const int h = cell / Wd, w = cell % Wd;
        x_gaddr[i] = img_base + (int64_t)cell * C + j * 8;
        x_laddr[i] = ((h + 1) * Wp + (w + 1)) * CW_CS + cw_pbase(j);
      } else {
        x_gaddr[i] = img_base;   // clamped safe address
        x_laddr[i] = cw_pbase(j);  // pad cell (0,0): zero writes keep it zero
      }
    }
  }
  // w: 9 slots, one per tap: k = tid>>2, j = tid&3
  const int wk = tid >> 2, wj = tid & 3;
  const bool w_ok = (k0 + wk) < K;
  const int64_t w_gbase =
      ((int64_t)(w_ok ? (k0 + wk) : 0) * 9) * C + wj * 8;  // clamped
  const int w_lbase = wk * CW_CS + cw_pbase(wj);

  const int NT = C / CW_BK;

  u16x8 xr[2];
  u16x8 wr[9];

  auto load_tile = [&](int ct) {
    const int64_t coff = (int64_t)ct * CW_BK;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      xr[i] = *(const u16x8*)(X + x_gaddr[i] + coff);
    }
#pragma unroll
    for (int t = 0; t < 9; ++t) {
      wr[t] = *(const u16x8*)(W + w_gbase + (int64_t)t * C + coff);
    }
  };

  auto store_tile = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      __hip_bfloat16* dst = &lx[buf][x_laddr[i]];
      const uint64_t lo = x_valid[i] ? *(const uint64_t*)&xr[i] : 0ull;
      const uint64_t hi =
          x_valid[i] ? *(((const uint64_t*)&xr[i]) + 1) : 0ull;
      *(uint64_t*)dst = lo;          // phys p..p+3
      *(uint64_t*)(dst + 8) = hi;    // p+8..p+11
    }
#pragma unroll
    for (int t = 0; t < 9; ++t) {
      __hip_bfloat16* dst = &lw[buf][t * CW_BN * CW_CS + w_lbase];
      const uint64_t lo = w_ok ? *(const uint64_t*)&wr[t] : 0ull;
      const uint64_t hi = w_ok ? *(((const uint64_t*)&wr[t]) + 1) : 0ull;
      *(uint64_t*)dst = lo;
      *(uint64_t*)(dst + 8) = hi;
    }
  };

  // per-(thread, mf) halo cell offsets for the afrag reads
  int mcell[MF];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int m = mf * 16 + (lane & 15);
    const int h = m / Wd, w = m % Wd;
    mcell[mf] = ((h + 1) * Wp + (w + 1)) * CW_CS + 8 * kg;
  }

  if32x4 acc[MF] = {};

  load_tile(0);
  __syncthreads();   // after the zero pass
  store_tile(0);
  if (NT > 1) load_tile(1);
  __syncthreads();

  for (int ct = 0; ct < NT; ++ct) {
    if (ct + 1 < NT) {
      store_tile((ct + 1) & 1);
      if (ct + 2 < NT) load_tile(ct + 2);
    }
    const __hip_bfloat16* xb = lx[ct & 1];
    const __hip_bfloat16* wb = lw[ct & 1];
#pragma unroll
    for (int r = 0; r < 3; ++r) {
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        const int tap = r * 3 + s;
        const ibf16x8 bfrag = *(const ibf16x8*)(
            wb + tap * CW_BN * CW_CS + fn * CW_CS + 8 * kg);
        const int toff = ((r - 1) * Wp + (s - 1)) * CW_CS;
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
          const ibf16x8 afrag = *(const ibf16x8*)(xb + mcell[mf] + toff);
          acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                            acc[mf], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  const int kc = k0 + fn;
  if (kc >= K) return;
  const int64_t out_base = (int64_t)img * HW * K + kc;
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int m = mf * 16 + (lane >> 4) * 4 + reg;
      Y[out_base + (int64_t)m * K] = __float2bfloat16(acc[mf][reg]);
    }
  }
}

extern "C" void flreid_conv3x3_img_fwd_ldsw(const void* X, const void* W, void* Y,
                                       int NB, int H, int Wd, int C, int K,
                                       hipStream_t stream) {
  const int HW = H * Wd;
  if (HW > 128 || (HW & 15) || C % CW_BK || K % 16 || H + 2 > 18 ||
      Wd + 2 > 10) {
    throw std::runtime_error("conv3x3_img_fwd_ldsw: shape out of regime");
  }
  dim3 grid((K + CW_BN - 1) / CW_BN, NB);
  const int mf = HW >> 4;
  switch (mf) {
#define FLREID_CW_CASE(MF)                                                   \
  case MF:                                                                   \
    hipLaunchKernelGGL(conv3x3_imgw_fwd_kernel<MF>, grid, dim3(256), 0,       \
                       stream, (const __hip_bfloat16*)X,                     \
                       (const __hip_bfloat16*)W, (__hip_bfloat16*)Y, H, Wd,  \
                       C, K);                                                \
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
