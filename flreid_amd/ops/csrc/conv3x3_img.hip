// K1 production conv suite: 3×3 stride-1 pad-1 NHWC bf16, specialized for
// the ReID layer-4 regime (H·W ≤ 128: one IMAGE per block).
//
// The ref hot path (ref:models/resnet.py:93-141, the _Bottleneck conv chain
// at 16×8 spatial with last_stride=1) runs these shapes for every training
// step of every method and for the frozen-backbone eval forwards.
//
// Design vs the generic nine-shifted-GEMM kernel (conv3x3.hip):
//  - one image per block: the 3×3 taps re-read the SAME image rows, so the
//    image's C-tile is staged ONCE into a zero-padded LDS halo
//    [(H+2)·(W+2)][32] and all nine taps read LDS — 9× less global x traffic
//    and NO border masking in the MFMA loop (the pad rows/cols are zero);
//  - bf16 weights (the composed θ is produced in bf16 by the compose
//    kernel) — half the weight bytes of the fp32-weight generic kernel;
//  - the LDS c-dimension is stored PERMUTED (p(c) = ((c%16)>>2)*8 +
//    (c>>4)*4 + (c&3)) so one MFMA fragment (the 8 bf16 a lane feeds to
//    v_mfma_f32_16x16x32_bf16: k = kg+e and 16+kg+e') is 16 CONTIGUOUS
//    bytes -> one ds_read_b128 per fragment instead of 8 scalar reads;
//  - cell stride 48 elems (96 B = 24-dword = 8·odd bank stride) keeps
//    b128 rows 16B-aligned and every read lane group conflict-free;
//  - T14 pipeline: per c-tile, one barrier; global loads for tile t+2 are
//    issued before the MFMA burst on tile t (72 MFMAs/wave per barrier);
//  - grid = (K/64, n_images): with K/64 == 8 the k-block equals the XCD id
//    (blocks dispatch round-robin by linear id), so each XCD's L2 holds one
//    295 KB weight slice instead of all 2.4 MB.
//
// dgrad = this same kernel applied to (dy, flipped-transposed θ): dx =
// conv3x3_s1p1(dy, wT) with wT[c][r][s][k] = w[k][2-r][2-s][c]
// (conv3x3_tile mode 1 emits that tile straight from gw/atten/aw).
//
// wgrad (conv3x3_wgrad_kernel): per-tap M-reduction GEMM on
// PRE-TRANSPOSED operands, dθ[k,c,(r,s)] = Σ_m dyT[k,m]·xT[c,shift_rs(m)],
// fp32 accumulate into exclusive split-M partials; border masking applied
// at the staging loads.

#include "common.h"

#include <algorithm>

namespace flreid {

using ibf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using if32x4 = __attribute__((ext_vector_type(4))) float;
using u16x8 = __attribute__((ext_vector_type(8))) unsigned short;

constexpr int CI_CS = 48;        // padded per-cell c-stride (elems): 96 B
                                 // rows -> 24-dword bank stride = 8·odd:
                                 // conflict-free ds_read_b128 lane groups
constexpr int CI_CELLS = 18 * 10;  // max (H+2)*(W+2)
constexpr int CI_BK = 32;        // input-channel tile
constexpr int CI_BN = 64;        // output channels per block

// LDS c-permutation: fragment phys chunk base for c-chunk j (c = 8j..8j+7)
__device__ __forceinline__ int ci_pbase(int j) {
  // j=0 -> {0..3, 8..11}; j=1 -> {16..19, 24..27}; j=2 -> {4..7, 12..15};
  // j=3 -> {20..23, 28..31}   (p(c) = ((c%16)>>2)*8 + (c>>4)*4 + (c&3))
  const int base[4] = {0, 16, 4, 20};
  return base[j];
}

// -------------------------------------------------------------------------
// Weight tiling: θ is emitted by the composition kernel in a CONV-TILED
// bf16 layout [C/32][9][4kg][K][8] — per (c-tile, tap) panel the fragment
// chunks are ordered (kg, k): lane (fn, kg) reads its 16-B B-fragment at
// panel + (kg·K + fn)·8, so a wave's 64 lanes cover two contiguous 512-B
// runs per quarter-wave group and weights never touch LDS.  mode 1
// produces the dgrad tile with C/K swapped and flipped taps
// (wT[c][2-r][2-s][k]).
// -------------------------------------------------------------------------

__device__ __forceinline__ int ci_invp(int p) {
  const int g = p >> 3, i = p & 7;
  return (i < 4) ? (4 * g + i) : (16 + 4 * g + i - 4);
}

template <typename Tin>
__global__ __launch_bounds__(256) void conv3x3_tile_kernel(
    const Tin* __restrict__ GW, const float* __restrict__ ATTEN,
    const Tin* __restrict__ AW, __hip_bfloat16* __restrict__ OUT,
    int C, int K, int mode) {
  const int64_t total = (int64_t)C * 9 * K;
  for (int64_t o = (int64_t)blockIdx.x * 256 + threadIdx.x; o < total;
       o += (int64_t)gridDim.x * 256) {
    const int e = (int)(o & 7);           // element within the 16-B chunk
    const int64_t o2 = o >> 3;
    int k, c, tap;
    if (mode == 0) {                      // [C/32][9][4kg][K][8]
      k = (int)(o2 % K);
      const int kg = (int)((o2 / K) & 3);
      tap = (int)((o2 / ((int64_t)K * 4)) % 9);
      c = (int)(o2 / ((int64_t)K * 4 * 9)) * 32 + ci_invp(kg * 8 + e);
    } else {                              // [K/32][9][4kg][C][8], flipped
      c = (int)(o2 % C);
      const int kg = (int)((o2 / C) & 3);
      const int ft = (int)((o2 / ((int64_t)C * 4)) % 9);
      tap = (2 - ft / 3) * 3 + (2 - ft % 3);
      k = (int)(o2 / ((int64_t)C * 4 * 9)) * 32 + ci_invp(kg * 8 + e);
    }
    const int64_t src = ((int64_t)k * 9 + tap) * C + c;
    const float a = ATTEN ? ATTEN[tap % 3] : 1.0f;
    const float v = fmaf(a, load_as_float(GW, src),
                         AW ? load_as_float(AW, src) : 0.0f);
    OUT[o] = __float2bfloat16(v);
  }
}

extern "C" void flreid_conv3x3_tile(const void* GW, const float* ATTEN,
                                    const void* AW, void* OUT, int C, int K,
                                    int in_dtype, int mode,
                                    hipStream_t stream) {
  const int64_t total = (int64_t)C * 9 * K;
  const int blocks = (int)std::min<int64_t>((total + 255) / 256, 8192);
  if (in_dtype == kF32) {
    hipLaunchKernelGGL((conv3x3_tile_kernel<float>), dim3(blocks), dim3(256),
                       0, stream, (const float*)GW, ATTEN, (const float*)AW,
                       (__hip_bfloat16*)OUT, C, K, mode);
  } else {
    hipLaunchKernelGGL((conv3x3_tile_kernel<__hip_bfloat16>), dim3(blocks),
                       dim3(256), 0, stream, (const __hip_bfloat16*)GW,
                       ATTEN, (const __hip_bfloat16*)AW,
                       (__hip_bfloat16*)OUT, C, K, mode);
  }
  HIP_CHECK(hipGetLastError());
}

template <int MF>
__global__ __launch_bounds__(256) void conv3x3_img_fwd_kernel(
    const __hip_bfloat16* __restrict__ X, const __hip_bfloat16* __restrict__ WT,
    __hip_bfloat16* __restrict__ Y, int H, int Wd, int C, int K) {
  __shared__ __hip_bfloat16 lx[2][CI_CELLS * CI_CS];

  const int k0 = blockIdx.x * CI_BN;      // k-block first: XCD affinity
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
    const int total = (H + 2) * Wp * CI_CS;
    for (int i = tid; i < total; i += 256) {
      lx[0][i] = __float2bfloat16(0.f);
      lx[1][i] = __float2bfloat16(0.f);
    }
  }

  // ---- x staging slots (unconditional clamped loads; invalid slots
  // select zero at the LDS store — guide §5 trap c) ----
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
        x_laddr[i] = ((h + 1) * Wp + (w + 1)) * CI_CS + ci_pbase(j);
      } else {
        x_gaddr[i] = img_base;     // clamped safe address
        x_laddr[i] = ci_pbase(j);  // pad cell (0,0): zero writes keep it zero
      }
    }
  }

  // per-lane weight-fragment base in the tiled layout (elements):
  // WT[(((ct*9 + tap)*4 + kg)*K + k0+fn)*8] — monotonic in lane, two
  // contiguous 512-B runs per quarter-wave
  const int64_t w_lane = ((int64_t)kg * K + (k0 + fn)) * 8;
  const int64_t w_ctstride = (int64_t)9 * K * 32;
  const int64_t w_tapstride = (int64_t)K * 32;
  const bool w_ok = (k0 + fn) < K;
  const int64_t w_base = w_ok ? w_lane : (int64_t)kg * K * 8;

  const int NT = C / CI_BK;

  u16x8 xr[2];
  u16x8 wfr[2][9];                        // weight fragments, 2 c-tiles deep

  auto load_x = [&](int ct) {
    const int64_t coff = (int64_t)ct * CI_BK;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      xr[i] = *(const u16x8*)(X + x_gaddr[i] + coff);
    }
  };

  auto load_w = [&](int ct, int ring) {
    const __hip_bfloat16* wp = WT + (int64_t)ct * w_ctstride + w_base;
#pragma unroll
    for (int t = 0; t < 9; ++t) {
      wfr[ring][t] = *(const u16x8*)(wp + (int64_t)t * w_tapstride);
    }
  };

  auto store_x = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      __hip_bfloat16* dst = &lx[buf][x_laddr[i]];
      const uint64_t lo = x_valid[i] ? *(const uint64_t*)&xr[i] : 0ull;
      const uint64_t hi =
          x_valid[i] ? *(((const uint64_t*)&xr[i]) + 1) : 0ull;
      *(uint64_t*)dst = lo;          // phys p..p+3
      *(uint64_t*)(dst + 8) = hi;    // p+8..p+11
    }
  };

  // per-(thread, mf) halo cell offsets for the afrag reads
  int mcell[MF];
#pragma unroll
  for (int mf = 0; mf < MF; ++mf) {
    const int m = mf * 16 + (lane & 15);
    const int h = m / Wd, w = m % Wd;
    mcell[mf] = ((h + 1) * Wp + (w + 1)) * CI_CS + 8 * kg;
  }

  if32x4 acc[MF] = {};

  load_x(0);
  load_w(0, 0);
  __syncthreads();   // after the zero pass
  store_x(0);
  if (NT > 1) {
    load_x(1);
    load_w(1, 1);
  }
  __syncthreads();

  for (int ct = 0; ct < NT; ++ct) {
    if (ct + 1 < NT) {
      store_x((ct + 1) & 1);
      if (ct + 2 < NT) load_x(ct + 2);
    }
    const __hip_bfloat16* xb = lx[ct & 1];
#pragma unroll
    for (int r = 0; r < 3; ++r) {
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        const int tap = r * 3 + s;
        const ibf16x8 bfrag = *(const ibf16x8*)&wfr[ct & 1][tap];
        const int toff = ((r - 1) * Wp + (s - 1)) * CI_CS;
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
          const ibf16x8 afrag = *(const ibf16x8*)(xb + mcell[mf] + toff);
          acc[mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                            acc[mf], 0, 0, 0);
        }
      }
    }
    if (ct + 2 < NT) load_w(ct + 2, ct & 1);   // refill the ring just freed
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

extern "C" void flreid_conv3x3_img_fwd(const void* X, const void* W, void* Y,
                                       int NB, int H, int Wd, int C, int K,
                                       hipStream_t stream) {
  const int HW = H * Wd;
  if (HW > 128 || (HW & 15) || C % CI_BK || K % 16 || H + 2 > 18 ||
      Wd + 2 > 10) {
    throw std::runtime_error("conv3x3_img_fwd: shape out of regime");
  }
  dim3 grid((K + CI_BN - 1) / CI_BN, NB);
  const int mf = HW >> 4;
  switch (mf) {
#define FLREID_CI_CASE(MF)                                                   \
  case MF:                                                                   \
    hipLaunchKernelGGL(conv3x3_img_fwd_kernel<MF>, grid, dim3(256), 0,       \
                       stream, (const __hip_bfloat16*)X,                     \
                       (const __hip_bfloat16*)W, (__hip_bfloat16*)Y, H, Wd,  \
                       C, K);                                                \
    break;
    FLREID_CI_CASE(1)
    FLREID_CI_CASE(2)
    FLREID_CI_CASE(3)
    FLREID_CI_CASE(4)
    FLREID_CI_CASE(5)
    FLREID_CI_CASE(6)
    FLREID_CI_CASE(7)
    FLREID_CI_CASE(8)
#undef FLREID_CI_CASE
    default:
      throw std::runtime_error("conv3x3_img_fwd: bad MF");
  }
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// 64×64 LDS-tiled bf16 transpose (wgrad operand prep): OUT[n][m] = IN[m][n]
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void transpose_bf16_kernel(
    const __hip_bfloat16* __restrict__ IN, __hip_bfloat16* __restrict__ OUT,
    int64_t M, int64_t N) {
  __shared__ __hip_bfloat16 tile[64][72];   // 8-elem pad
  const int64_t m0 = (int64_t)blockIdx.x * 64;
  const int64_t n0 = (int64_t)blockIdx.y * 64;
  const int tid = threadIdx.x;
  const int tr = tid >> 3;          // 0..31
  const int tc8 = (tid & 7) * 8;    // 16-B column chunk
  for (int rr = 0; rr < 64; rr += 32) {
    const int64_t m = m0 + tr + rr;
    u16x8 v = {};
    if (m < M && n0 + tc8 < N) {
      v = *(const u16x8*)(IN + m * N + n0 + tc8);
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      tile[tr + rr][tc8 + e] = *((const __hip_bfloat16*)&v + e);
    }
  }
  __syncthreads();
  for (int rr = 0; rr < 64; rr += 32) {
    const int64_t n = n0 + tr + rr;
    if (n >= N) continue;
    u16x8 v;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      *((__hip_bfloat16*)&v + e) = tile[tc8 + e][tr + rr];
    }
    if (m0 + tc8 < M) {
      *(u16x8*)(OUT + n * M + m0 + tc8) = v;
    }
  }
}

extern "C" void flreid_transpose_bf16(const void* IN, void* OUT, int64_t M,
                                      int64_t N, hipStream_t stream) {
  if (M % 8 || N % 8) {
    // the kernel moves 16-B chunks guarded only at their start; a
    // non-multiple-of-8 edge would overhang into the next row
    throw std::runtime_error("transpose_bf16: M%8 or N%8 != 0");
  }
  dim3 grid((unsigned)((M + 63) / 64), (unsigned)((N + 63) / 64));
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, stream,
                     (const __hip_bfloat16*)IN, (__hip_bfloat16*)OUT, M, N);
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// wgrad v3: dθ[k][r][s][c] = Σ_m dyT[k][m] · xT[c][shift_rs(m)]
// ---------------------------------------------------------------------------
// Operates on PRE-TRANSPOSED operands (transpose_bf16 above): both LDS
// panels stage coalesced 16-B row chunks with vectorized b64 writes — the
// in-kernel transposed b16 scatter of the first version measured 75 LDS
// bank-conflict cycles per MFMA (PMC) and is gone.  The tap's m-shift and
// border mask are applied at the xT staging (one (n,h,w) decode per 8-m
// chunk, incremental carry within it).

constexpr int WG_BK = 64;    // k rows per block
constexpr int WG_BC = 128;   // c rows per block
constexpr int WG_BM = 64;    // m per iteration (two 32-m MFMA halves)
constexpr int WG_MS = 80;    // LDS m-stride (elems)
constexpr int WG_SPLITM = 2;

__global__ __launch_bounds__(256) void conv3x3_wgrad_kernel(
    const __hip_bfloat16* __restrict__ DYT, const __hip_bfloat16* __restrict__ XT,
    float* __restrict__ DW, int NB, int H, int Wd, int C, int K) {
  __shared__ __hip_bfloat16 ldy[2][WG_BK * WG_MS];
  __shared__ __hip_bfloat16 lxc[2][WG_BC * WG_MS];

  const int kb = blockIdx.x;          // k-block first: XCD dyT-affinity
  const int cb = blockIdx.y;
  const int tap = blockIdx.z % 9;
  const int sp = blockIdx.z / 9;
  const int r = tap / 3 - 1, s = tap % 3 - 1;
  const int k0 = kb * WG_BK, c0 = cb * WG_BC;
  const int HW = H * Wd;
  const int64_t M = (int64_t)NB * HW;
  const int shiftM = r * Wd + s;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wk = wave & 1, wc = wave >> 1;
  const int kg = lane >> 4;

  // A (dyT): 2 slots/thread: row = slot>>3 (0..63), j = slot&7 (8-m chunk)
  // B (xT): 4 slots/thread: row = slot>>3 (0..127), j = slot&7
  u16x8 ar_[2], br_[4];
  unsigned short bm_[4];              // per-element border masks for B

  // DYT/XT carry ≥128-elem pads on both ends (the wrapper allocates them),
  // so shifted/tail chunk addresses are loaded UNCLAMPED — a clamp would
  // misalign the chunk's element<->m correspondence.  Out-of-range lanes
  // read pad/neighbour-row bytes and are zero-masked at the LDS store.
  auto load_tile = [&](int64_t m0) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int slot = tid + 256 * i;
      const int row = slot >> 3, j = slot & 7;
      const int64_t kk = (int64_t)(k0 + row < K ? k0 + row : 0);
      const int64_t mc = m0 + 8 * j;
      ar_[i] = *(const u16x8*)(DYT + kk * M + mc);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int slot = tid + 256 * i;
      const int row = slot >> 3, j = slot & 7;
      const int64_t cc = (int64_t)(c0 + row < C ? c0 + row : 0);
      const int64_t mbase = m0 + 8 * j;
      const int64_t mc = mbase + shiftM;
      br_[i] = *(const u16x8*)(XT + cc * M + mc);
      // border mask: decode (n,h,w) of the first m, carry within the chunk
      unsigned short mask = 0;
      int mm = (int)(mbase < M ? mbase : M - 1);
      int n = mm / HW;
      int rem = mm - n * HW;
      int h = rem / Wd;
      int w = rem - h * Wd;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const bool ok = (mbase + e < M) && (h + r >= 0) && (h + r < H) &&
                        (w + s >= 0) && (w + s < Wd) &&
                        (mbase + e + shiftM >= 0);
        mask |= (ok ? 1 : 0) << e;
        if (++w == Wd) { w = 0; if (++h == H) h = 0; }
      }
      bm_[i] = mask;
    }
  };

  auto store_tile = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int slot = tid + 256 * i;
      const int row = slot >> 3, j = slot & 7;
      const int pm = (j >> 2) * 32 + ci_pbase(j & 3);
      __hip_bfloat16* dst = &ldy[buf][row * WG_MS + pm];
      const uint64_t zero_row = (k0 + row < K) ? ~0ull : 0ull;
      *(uint64_t*)dst = *(const uint64_t*)&ar_[i] & zero_row;
      *(uint64_t*)(dst + 8) = *(((const uint64_t*)&ar_[i]) + 1) & zero_row;
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int slot = tid + 256 * i;
      const int row = slot >> 3, j = slot & 7;
      const int pm = (j >> 2) * 32 + ci_pbase(j & 3);
      u16x8 v = br_[i];
      const unsigned short mask = bm_[i];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        if (!((mask >> e) & 1)) *((__hip_bfloat16*)&v + e) = __float2bfloat16(0.f);
      }
      __hip_bfloat16* dst = &lxc[buf][row * WG_MS + pm];
      *(uint64_t*)dst = *(const uint64_t*)&v;
      *(uint64_t*)(dst + 8) = *(((const uint64_t*)&v) + 1);
    }
  };

  if32x4 acc[2][4] = {};

  const int NC = (int)((M + WG_BM - 1) / WG_BM);
  const int per = (NC + WG_SPLITM - 1) / WG_SPLITM;
  const int ch0 = sp * per;
  const int ch1 = min(NC, ch0 + per);
  float* dst_part = DW + (int64_t)sp * K * 9 * C;
  if (ch0 >= ch1) {
    // empty split: its partial tile must still be zero-filled
#pragma unroll
    for (int fk = 0; fk < 2; ++fk)
#pragma unroll
      for (int fc = 0; fc < 4; ++fc)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int k = k0 + wk * 32 + fk * 16 + (lane >> 4) * 4 + reg;
          const int c = c0 + wc * 64 + fc * 16 + (lane & 15);
          if (k < K && c < C) dst_part[((int64_t)k * 9 + tap) * C + c] = 0.f;
        }
    return;
  }

  load_tile((int64_t)ch0 * WG_BM);
  store_tile(0);
  if (ch0 + 1 < ch1) load_tile((int64_t)(ch0 + 1) * WG_BM);
  __syncthreads();

  for (int it = ch0; it < ch1; ++it) {
    const int buf = (it - ch0) & 1;
    if (it + 1 < ch1) {
      store_tile(buf ^ 1);
      if (it + 2 < ch1) load_tile((int64_t)(it + 2) * WG_BM);
    }
    const __hip_bfloat16* db = ldy[buf];
    const __hip_bfloat16* xb = lxc[buf];
#pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
      ibf16x8 afrag[2];
#pragma unroll
      for (int fk = 0; fk < 2; ++fk) {
        afrag[fk] = *(const ibf16x8*)(
            db + (wk * 32 + fk * 16 + (lane & 15)) * WG_MS + hh * 32 + 8 * kg);
      }
#pragma unroll
      for (int fc = 0; fc < 4; ++fc) {
        const ibf16x8 bfrag = *(const ibf16x8*)(
            xb + (wc * 64 + fc * 16 + (lane & 15)) * WG_MS + hh * 32 + 8 * kg);
#pragma unroll
        for (int fk = 0; fk < 2; ++fk) {
          acc[fk][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[fk], bfrag, acc[fk][fc], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int fk = 0; fk < 2; ++fk) {
#pragma unroll
    for (int fc = 0; fc < 4; ++fc) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int k = k0 + wk * 32 + fk * 16 + (lane >> 4) * 4 + reg;
        const int c = c0 + wc * 64 + fc * 16 + (lane & 15);
        if (k < K && c < C) {
          dst_part[((int64_t)k * 9 + tap) * C + c] = acc[fk][fc][reg];
        }
      }
    }
  }
}

extern "C" void flreid_conv3x3_wgrad(const void* DYT, const void* XT,
                                     float* DW, int NB, int H, int Wd, int C,
                                     int K, hipStream_t stream) {
  if (C % 8 || K % 8) {
    throw std::runtime_error("conv3x3_wgrad: C%8 or K%8 != 0");
  }
  // DW = [WG_SPLITM][K][9][C] fp32 partials; caller sums over splits.
  // DYT/XT are the TRANSPOSED [K][M] / [C][M] operands.
  dim3 grid((K + WG_BK - 1) / WG_BK, (C + WG_BC - 1) / WG_BC, 9 * WG_SPLITM);
  hipLaunchKernelGGL(conv3x3_wgrad_kernel, grid, dim3(256), 0, stream,
                     (const __hip_bfloat16*)DYT, (const __hip_bfloat16*)XT,
                     DW, NB, H, Wd, C, K);
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
