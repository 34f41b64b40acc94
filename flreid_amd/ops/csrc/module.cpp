// Python bindings for the flreid MI355X kernels.
//
// Deliberately torch-header-free: tensors cross as raw device pointers plus
// the caller's torch HIP stream (uintptr_t), so the extension compiles with
// plain hipcc for gfx950 — no hipify, no CUDA-compat layer — and launches
// stay ordered with PyTorch work on the same stream.

#include <pybind11/pybind11.h>

#include <cstdint>

#include <hip/hip_runtime.h>

namespace flreid {
extern "C" void flreid_l2norm_rows(const void*, void*, int64_t, int64_t, int,
                                   float, hipStream_t);
extern "C" void flreid_ce_smooth(const void*, const int64_t*, float*, void*,
                                 int64_t, int64_t, int, float, hipStream_t);
extern "C" void flreid_compose(const void*, const float*, const void*, void*,
                               int64_t, int64_t, int, hipStream_t);
extern "C" void flreid_importance(float*, const void*, int64_t, int, int,
                                  float, hipStream_t);
extern "C" void flreid_pairwise(const float*, const float*, const float*,
                                const float*, float*, int64_t, int64_t,
                                int64_t, int, hipStream_t);
extern "C" void flreid_rowsq(const float*, float*, int64_t, int64_t,
                             hipStream_t);
extern "C" void flreid_window_attn_fwd(const void*, const void*, const void*,
                                       const float*, const float*, void*,
                                       int64_t, int, int, int, int, float,
                                       int, hipStream_t);
extern "C" void flreid_window_attn_bwd(const void*, const void*, const void*,
                                       const float*, const float*,
                                       const void*, void*, void*, void*,
                                       float*, int64_t, int, int, int, int,
                                       float, int, hipStream_t);
extern "C" void flreid_triplet_fwd(const float*, const float*, const int64_t*,
                                   float*, int*, int*, int, int, float,
                                   hipStream_t);
extern "C" void flreid_triplet_bwd(const float*, const float*, const int*,
                                   const int*, float*, int, int, float,
                                   hipStream_t);
extern "C" void flreid_adaptive_linear_fwd(const void*, const float*,
                                           const float*, const float*,
                                           const float*, void*, int, int,
                                           int, int, hipStream_t);
extern "C" void flreid_bn_eval(const void*, void*, const float*, const float*,
                               const float*, const float*, int64_t, int,
                               int64_t, float, int, int, int, hipStream_t);
extern "C" void flreid_conv3x3_fwd(const void*, const float*, void*, int, int,
                                   int, int, int, hipStream_t);
extern "C" int flreid_bn_train_nslab(int64_t, int);
extern "C" void flreid_bn_train_fwd(const void*, void*, const float*,
                                    const float*, float*, float*, float*,
                                    float*, float*, float*,
                                    unsigned long long*, int64_t, int, float,
                                    float, float, int, int, hipStream_t);
extern "C" void flreid_bn_train_bwd(const void*, const void*, void*,
                                    const float*, const float*, const float*,
                                    const float*, float*, float*, float*,
                                    float*, int64_t, int, int, int,
                                    hipStream_t);
extern "C" void flreid_compose2(const void*, const float*, const void*,
                                void*, int64_t, int64_t, int64_t, int, int,
                                hipStream_t);
extern "C" void flreid_conv3x3_img_fwd(const void*, const void*, void*, int,
                                       int, int, int, int, hipStream_t);
extern "C" void flreid_transpose_bf16(const void*, void*, int64_t, int64_t,
                                      hipStream_t);
extern "C" void flreid_conv3x3_tile(const void*, const float*, const void*,
                                    void*, int, int, int, int, hipStream_t);
extern "C" void flreid_conv3x3_img_fwd_ldsw(const void*, const void*, void*,
                                            int, int, int, int, int,
                                            hipStream_t);
extern "C" void flreid_conv3x3_wgrad(const void*, const void*, float*, int,
                                     int, int, int, int, hipStream_t);
extern "C" void flreid_patch_merge_ln_fwd(const void*, const float*,
                                          const float*, void*, float*,
                                          float*, int64_t, int, int, int,
                                          float, int, hipStream_t);
extern "C" void flreid_patch_merge_ln_bwd(const void*, const float*,
                                          const void*, const float*,
                                          const float*, void*, float*,
                                          float*, int64_t, int, int, int,
                                          int, hipStream_t);
extern "C" void flreid_kd_fwd(const float*, const float*, float*, float*,
                              int64_t, int64_t, float, hipStream_t);
extern "C" void flreid_icarl_distill(const float*, const int64_t*,
                                     const float*, float*, float*, int64_t,
                                     int64_t, int64_t, hipStream_t);
extern "C" void flreid_drift_fwd(const int64_t*, const int*, float*, int,
                                 hipStream_t);
extern "C" void flreid_drift_bwd(const int64_t*, const int*, const float*,
                                 float*, int, hipStream_t);
}  // namespace flreid

namespace py = pybind11;

static hipStream_t as_stream(uintptr_t s) {
  return reinterpret_cast<hipStream_t>(s);
}

PYBIND11_MODULE(_flreid_hip, m) {
  m.doc() = "flreid MI355X (gfx950) kernels";

  m.def("l2norm_rows",
        [](uintptr_t x, uintptr_t y, int64_t rows, int64_t cols, int dtype,
           float eps, uintptr_t stream) {
          flreid::flreid_l2norm_rows((const void*)x, (void*)y, rows, cols,
                                     dtype, eps, as_stream(stream));
        });

  m.def("ce_smooth",
        [](uintptr_t score, uintptr_t target, uintptr_t row_loss,
           uintptr_t grad, int64_t B, int64_t C, int dtype, float eps,
           uintptr_t stream) {
          flreid::flreid_ce_smooth((const void*)score, (const int64_t*)target,
                                   (float*)row_loss, (void*)grad, B, C, dtype,
                                   eps, as_stream(stream));
        });

  m.def("compose",
        [](uintptr_t gw, uintptr_t atten, uintptr_t aw, uintptr_t out,
           int64_t numel, int64_t L, int dtype, uintptr_t stream) {
          flreid::flreid_compose((const void*)gw, (const float*)atten,
                                 (const void*)aw, (void*)out, numel, L, dtype,
                                 as_stream(stream));
        });

  m.def("importance",
        [](uintptr_t F, uintptr_t g, int64_t numel, int dtype, bool sq,
           float scale, uintptr_t stream) {
          flreid::flreid_importance((float*)F, (const void*)g, numel, dtype,
                                    sq ? 1 : 0, scale, as_stream(stream));
        });

  m.def("pairwise",
        [](uintptr_t A, uintptr_t B, uintptr_t aa, uintptr_t bb, uintptr_t out,
           int64_t M, int64_t N, int64_t D, int mode, uintptr_t stream) {
          flreid::flreid_pairwise((const float*)A, (const float*)B,
                                  (const float*)aa, (const float*)bb,
                                  (float*)out, M, N, D, mode,
                                  as_stream(stream));
        });

  m.def("rowsq", [](uintptr_t x, uintptr_t out, int64_t rows, int64_t cols,
                    uintptr_t stream) {
    flreid::flreid_rowsq((const float*)x, (float*)out, rows, cols,
                         as_stream(stream));
  });

  m.def("window_attn_fwd",
        [](uintptr_t q, uintptr_t k, uintptr_t v, uintptr_t bias,
           uintptr_t mask, uintptr_t out, int64_t BW, int H, int N, int D,
           int nW, float scale, int dtype, uintptr_t stream) {
          flreid::flreid_window_attn_fwd(
              (const void*)q, (const void*)k, (const void*)v,
              (const float*)bias, (const float*)mask, (void*)out, BW, H, N, D,
              nW, scale, dtype, as_stream(stream));
        });

  m.def("window_attn_bwd",
        [](uintptr_t q, uintptr_t k, uintptr_t v, uintptr_t bias,
           uintptr_t mask, uintptr_t dout, uintptr_t dq, uintptr_t dk,
           uintptr_t dv, uintptr_t ds, int64_t BW, int H, int N, int D,
           int nW, float scale, int dtype, uintptr_t stream) {
          flreid::flreid_window_attn_bwd(
              (const void*)q, (const void*)k, (const void*)v,
              (const float*)bias, (const float*)mask, (const void*)dout,
              (void*)dq, (void*)dk, (void*)dv, (float*)ds, BW, H, N, D, nW,
              scale, dtype, as_stream(stream));
        });

  m.def("triplet_fwd",
        [](uintptr_t F, uintptr_t norms, uintptr_t labels, uintptr_t row_loss,
           uintptr_t p_idx, uintptr_t n_idx, int N, int D, float margin,
           uintptr_t stream) {
          flreid::flreid_triplet_fwd((const float*)F, (const float*)norms,
                                     (const int64_t*)labels, (float*)row_loss,
                                     (int*)p_idx, (int*)n_idx, N, D, margin,
                                     as_stream(stream));
        });

  m.def("adaptive_linear_fwd",
        [](uintptr_t X, uintptr_t GW, uintptr_t AW, uintptr_t ATTEN,
           uintptr_t BIAS, uintptr_t OUT, int M, int N, int K,
           int split_layout, uintptr_t stream) {
          flreid::flreid_adaptive_linear_fwd(
              (const void*)X, (const float*)GW, (const float*)AW,
              (const float*)ATTEN, (const float*)BIAS, (void*)OUT, M, N, K,
              split_layout, as_stream(stream));
        });

  m.def("bn_eval",
        [](uintptr_t x, uintptr_t y, uintptr_t gamma, uintptr_t beta,
           uintptr_t mean, uintptr_t var, int64_t numel, int C, int64_t HW,
           float eps, int nhwc, int relu, int dtype, uintptr_t stream) {
          flreid::flreid_bn_eval((const void*)x, (void*)y,
                                 (const float*)gamma, (const float*)beta,
                                 (const float*)mean, (const float*)var, numel,
                                 C, HW, eps, nhwc, relu, dtype,
                                 as_stream(stream));
        });

  m.def("conv3x3_fwd",
        [](uintptr_t x, uintptr_t w, uintptr_t y, int NB, int H, int Wd,
           int C, int K, uintptr_t stream) {
          flreid::flreid_conv3x3_fwd((const void*)x, (const float*)w,
                                     (void*)y, NB, H, Wd, C, K,
                                     as_stream(stream));
        });

  m.def("bn_train_nslab", [](int64_t M, int C) {
    return flreid::flreid_bn_train_nslab(M, C);
  });

  m.def("bn_train_fwd",
        [](uintptr_t x, uintptr_t y, uintptr_t gamma, uintptr_t beta,
           uintptr_t rmean, uintptr_t rvar, uintptr_t smean, uintptr_t sinv,
           uintptr_t part_a, uintptr_t part_b, uintptr_t nbt, int64_t M,
           int C, float momentum, float eps, float unbiased, int relu,
           int dtype, uintptr_t stream) {
          flreid::flreid_bn_train_fwd((const void*)x, (void*)y,
                                      (const float*)gamma, (const float*)beta,
                                      (float*)rmean, (float*)rvar,
                                      (float*)smean, (float*)sinv,
                                      (float*)part_a, (float*)part_b,
                                      (unsigned long long*)nbt, M, C,
                                      momentum, eps, unbiased, relu, dtype,
                                      as_stream(stream));
        });

  m.def("bn_train_bwd",
        [](uintptr_t x, uintptr_t dy, uintptr_t dx, uintptr_t gamma,
           uintptr_t beta, uintptr_t smean, uintptr_t sinv, uintptr_t dgamma,
           uintptr_t dbeta, uintptr_t part_a, uintptr_t part_b, int64_t M,
           int C, int relu, int dtype, uintptr_t stream) {
          flreid::flreid_bn_train_bwd((const void*)x, (const void*)dy,
                                      (void*)dx, (const float*)gamma,
                                      (const float*)beta, (const float*)smean,
                                      (const float*)sinv, (float*)dgamma,
                                      (float*)dbeta, (float*)part_a,
                                      (float*)part_b, M, C, relu, dtype,
                                      as_stream(stream));
        });

  m.def("compose2",
        [](uintptr_t gw, uintptr_t atten, uintptr_t aw, uintptr_t out,
           int64_t numel, int64_t L, int64_t inner, int in_dtype,
           int out_dtype, uintptr_t stream) {
          flreid::flreid_compose2((const void*)gw, (const float*)atten,
                                  (const void*)aw, (void*)out, numel, L,
                                  inner, in_dtype, out_dtype,
                                  as_stream(stream));
        });

  m.def("conv3x3_img_fwd",
        [](uintptr_t x, uintptr_t w, uintptr_t y, int NB, int H, int Wd,
           int C, int K, uintptr_t stream) {
          flreid::flreid_conv3x3_img_fwd((const void*)x, (const void*)w,
                                         (void*)y, NB, H, Wd, C, K,
                                         as_stream(stream));
        });

  m.def("transpose_bf16",
        [](uintptr_t in, uintptr_t out, int64_t M, int64_t N,
           uintptr_t stream) {
          flreid::flreid_transpose_bf16((const void*)in, (void*)out, M, N,
                                        as_stream(stream));
        });

  m.def("conv3x3_img_fwd_ldsw",
        [](uintptr_t x, uintptr_t w, uintptr_t y, int NB, int H, int Wd,
           int C, int K, uintptr_t stream) {
          flreid::flreid_conv3x3_img_fwd_ldsw((const void*)x, (const void*)w,
                                              (void*)y, NB, H, Wd, C, K,
                                              as_stream(stream));
        });

  m.def("conv3x3_tile",
        [](uintptr_t gw, uintptr_t atten, uintptr_t aw, uintptr_t out,
           int C, int K, int in_dtype, int mode, uintptr_t stream) {
          flreid::flreid_conv3x3_tile((const void*)gw, (const float*)atten,
                                      (const void*)aw, (void*)out, C, K,
                                      in_dtype, mode, as_stream(stream));
        });

  m.def("conv3x3_wgrad",
        [](uintptr_t dy, uintptr_t x, uintptr_t dw, int NB, int H, int Wd,
           int C, int K, uintptr_t stream) {
          flreid::flreid_conv3x3_wgrad((const void*)dy, (const void*)x,
                                       (float*)dw, NB, H, Wd, C, K,
                                       as_stream(stream));
        });

  m.def("patch_merge_ln_fwd",
        [](uintptr_t x, uintptr_t gamma, uintptr_t beta, uintptr_t y,
           uintptr_t mean, uintptr_t rstd, int64_t rows, int C, int H,
           int W, float eps, int dtype, uintptr_t stream) {
          flreid::flreid_patch_merge_ln_fwd(
              (const void*)x, (const float*)gamma, (const float*)beta,
              (void*)y, (float*)mean, (float*)rstd, rows, C, H, W, eps,
              dtype, as_stream(stream));
        });

  m.def("patch_merge_ln_bwd",
        [](uintptr_t x, uintptr_t gamma, uintptr_t dy, uintptr_t mean,
           uintptr_t rstd, uintptr_t dx, uintptr_t dg, uintptr_t db,
           int64_t rows, int C, int H, int W, int dtype, uintptr_t stream) {
          flreid::flreid_patch_merge_ln_bwd(
              (const void*)x, (const float*)gamma, (const void*)dy,
              (const float*)mean, (const float*)rstd, (void*)dx, (float*)dg,
              (float*)db, rows, C, H, W, dtype, as_stream(stream));
        });

  m.def("kd_fwd",
        [](uintptr_t zs, uintptr_t zt, uintptr_t row_loss, uintptr_t grad,
           int64_t B, int64_t C, float temperature, uintptr_t stream) {
          flreid::flreid_kd_fwd((const float*)zs, (const float*)zt,
                                (float*)row_loss, (float*)grad, B, C,
                                temperature, as_stream(stream));
        });

  m.def("icarl_distill",
        [](uintptr_t score, uintptr_t target, uintptr_t prev,
           uintptr_t row_loss, uintptr_t grad, int64_t B, int64_t C,
           int64_t P, uintptr_t stream) {
          flreid::flreid_icarl_distill((const float*)score,
                                       (const int64_t*)target,
                                       (const float*)prev, (float*)row_loss,
                                       (float*)grad, B, C, P,
                                       as_stream(stream));
        });

  m.def("drift_fwd",
        [](uintptr_t ptrs, uintptr_t chunks, uintptr_t partials, int n_chunks,
           uintptr_t stream) {
          flreid::flreid_drift_fwd((const int64_t*)ptrs, (const int*)chunks,
                                   (float*)partials, n_chunks,
                                   as_stream(stream));
        });

  m.def("drift_bwd",
        [](uintptr_t ptrs, uintptr_t chunks, uintptr_t gscale,
           uintptr_t flat_grad, int n_chunks, uintptr_t stream) {
          flreid::flreid_drift_bwd((const int64_t*)ptrs, (const int*)chunks,
                                   (const float*)gscale, (float*)flat_grad,
                                   n_chunks, as_stream(stream));
        });

  m.def("triplet_bwd",
        [](uintptr_t F, uintptr_t row_loss, uintptr_t p_idx, uintptr_t n_idx,
           uintptr_t grad, int N, int D, float coeff, uintptr_t stream) {
          flreid::flreid_triplet_bwd((const float*)F, (const float*)row_loss,
                                     (const int*)p_idx, (const int*)n_idx,
                                     (float*)grad, N, D, coeff,
                                     as_stream(stream));
        });
}
