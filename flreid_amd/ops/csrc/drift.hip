// Fused multi-tensor L1 drift: Σᵢ ‖curᵢ − initᵢ‖₁ over a parameter list
// (FedSTIL round-start drift regulariser, ref:methods/fedstil.py:639-644,
// computed EVERY head-training step).
//
// torch's _foreach path costs ~5 full passes per step (materialised diffs,
// norm, sign, mul) across ~22 tensors / ~32 M fp32 elements.  Here: ONE
// read-only pass for the forward (block partials, summed by a tiny torch
// reduction) and ONE pass for the backward (writes sign(cur−init)·g into a
// single flat grad buffer the python wrapper splits into per-tensor views).
//
// The tensor list crosses as two small device tables built once per round
// (parameter storages are updated in-place across rounds, so the tables and
// any hipGraph that baked these pointers stay valid):
//   ptrs   int64 [n_pairs, 2]  — (cur, init) device addresses
//   chunks int32 [n_chunks, 4] — (pair, elem_offset, len, flat_offset)
// one 64 K-element chunk per workgroup.

#include "common.h"

namespace flreid {

constexpr int DRIFT_BLOCK = 256;

__global__ __launch_bounds__(DRIFT_BLOCK) void drift_fwd_kernel(
    const int64_t* __restrict__ ptrs, const int* __restrict__ chunks,
    float* __restrict__ partials) {
  const int* ch = chunks + (int64_t)blockIdx.x * 4;
  const float* cur =
      reinterpret_cast<const float*>(ptrs[ch[0] * 2]) + ch[1];
  const float* init =
      reinterpret_cast<const float*>(ptrs[ch[0] * 2 + 1]) + ch[1];
  const int len = ch[2];

  float acc = 0.f;
  const int q4 = len >> 2;            // full float4 quads
  const float4* c4 = reinterpret_cast<const float4*>(cur);
  const float4* i4 = reinterpret_cast<const float4*>(init);
  for (int i = threadIdx.x; i < q4; i += DRIFT_BLOCK) {
    const float4 a = c4[i], b = i4[i];
    acc += fabsf(a.x - b.x) + fabsf(a.y - b.y) + fabsf(a.z - b.z) +
           fabsf(a.w - b.w);
  }
  for (int i = (q4 << 2) + threadIdx.x; i < len; i += DRIFT_BLOCK) {
    acc += fabsf(cur[i] - init[i]);
  }
  __shared__ float lds[DRIFT_BLOCK / kWave];
  acc = block_reduce_sum<DRIFT_BLOCK>(acc, lds);
  if (threadIdx.x == 0) partials[blockIdx.x] = acc;
}

__global__ __launch_bounds__(DRIFT_BLOCK) void drift_bwd_kernel(
    const int64_t* __restrict__ ptrs, const int* __restrict__ chunks,
    const float* __restrict__ gscale, float* __restrict__ flat_grad) {
  const int* ch = chunks + (int64_t)blockIdx.x * 4;
  const float* cur =
      reinterpret_cast<const float*>(ptrs[ch[0] * 2]) + ch[1];
  const float* init =
      reinterpret_cast<const float*>(ptrs[ch[0] * 2 + 1]) + ch[1];
  const int len = ch[2];
  float* out = flat_grad + ch[3];
  const float g = *gscale;

  for (int i = threadIdx.x; i < len; i += DRIFT_BLOCK) {
    const float d = cur[i] - init[i];
    // subgradient 0 at 0, matching torch.sign
    out[i] = d > 0.f ? g : (d < 0.f ? -g : 0.f);
  }
}

extern "C" void flreid_drift_fwd(const int64_t* ptrs, const int* chunks,
                                 float* partials, int n_chunks,
                                 hipStream_t stream) {
  hipLaunchKernelGGL(drift_fwd_kernel, dim3(n_chunks), dim3(DRIFT_BLOCK), 0,
                     stream, ptrs, chunks, partials);
  HIP_CHECK(hipGetLastError());
}

extern "C" void flreid_drift_bwd(const int64_t* ptrs, const int* chunks,
                                 const float* gscale, float* flat_grad,
                                 int n_chunks, hipStream_t stream) {
  hipLaunchKernelGGL(drift_bwd_kernel, dim3(n_chunks), dim3(DRIFT_BLOCK), 0,
                     stream, ptrs, chunks, gscale, flat_grad);
  HIP_CHECK(hipGetLastError());
}

}  // namespace flreid
