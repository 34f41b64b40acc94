// Common helpers for flreid MI355X (gfx950 / CDNA4) kernels.
//
// Conventions:
//  - wavefront = 64 lanes (CDNA), blocks are multiples of 64 threads;
//  - fp32 compute with bf16 I/O where the model runs bf16 autocast;
//  - every kernel is launched on the torch stream passed from python.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>
#include <stdexcept>
#include <string>

#define HIP_CHECK(expr)                                                      \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      throw std::runtime_error(std::string("HIP error: ") +                  \
                               hipGetErrorString(_e) + " at " + __FILE__ +   \
                               ":" + std::to_string(__LINE__));              \
    }                                                                        \
  } while (0)

namespace flreid {

constexpr int kWave = 64;

// dtype tags shared with the python wrapper
enum DType : int { kF32 = 0, kBF16 = 1 };

__device__ __forceinline__ float load_as_float(const float* p, int64_t i) {
  return p[i];
}
__device__ __forceinline__ float load_as_float(const __hip_bfloat16* p, int64_t i) {
  return __bfloat162float(p[i]);
}
__device__ __forceinline__ void store_from_float(float* p, int64_t i, float v) {
  p[i] = v;
}
__device__ __forceinline__ void store_from_float(__hip_bfloat16* p, int64_t i, float v) {
  p[i] = __float2bfloat16(v);
}

// full-wave (64-lane) sum reduction
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v += __shfl_down(v, off, 64);
  }
  return v;
}

// block reduction through LDS (blockDim.x threads, <= 1024)
template <int BLOCK>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  constexpr int NW = BLOCK / kWave;
  float total = 0.f;
  if (threadIdx.x < NW) total = lds_scratch[threadIdx.x];
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1) {
      total += __shfl_down(total, off, 64);
    }
    if (lane == 0) lds_scratch[0] = total;
  }
  __syncthreads();
  float out = lds_scratch[0];
  __syncthreads();
  return out;
}

__device__ __forceinline__ float block_reduce_max(float v, float* lds_scratch,
                                                  int block) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_down(v, off, 64));
  }
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  const int nw = block / kWave;
  float total = -INFINITY;
  if (threadIdx.x < nw) total = lds_scratch[threadIdx.x];
  if (wid == 0) {
    for (int off = nw / 2; off > 0; off >>= 1) {
      total = fmaxf(total, __shfl_down(total, off, 64));
    }
    if (lane == 0) lds_scratch[0] = total;
  }
  __syncthreads();
  float out = lds_scratch[0];
  __syncthreads();
  return out;
}

}  // namespace flreid
