// SPDX-License-Identifier: Apache-2.0
// CSR SpMV for gfx950 — the hot kernel of the framework.
//
// Semantics: y[i] (+)= sum_jp vals[jp] * x[indices[jp]] over local rows
// (the reference computes this through cuSPARSE on a localized CSR,
// spmv.cu:62-157; here it is a hand-written CDNA4 kernel).
//
// Strategy: "vector CSR" — a power-of-two sub-wave of W lanes per row,
// W picked by the host from the mean row length.  Within a row the W lanes
// read vals/indices coalesced; sub-wave reduction via __shfl_down (64-wide
// waves, cdna_hip_programming.md §1).  fp64 CSR SpMV is memory-bound
// (12-16 B/nnz), so the design minimizes streamed bytes:
//   - int32 column indices when the matrix allows (DISPATCH_IDX_T)
//   - plain (cached) loads: a measured nt-load variant was 25-40%% SLOWER
//     (profiles/spmv_sweep_r01.txt) — nt defeats the L2/L3 help on the
//     vals/indices stream here, unlike the decode-weight case
//   - grid-stride with a capped grid (G11), no LDS staging.

#include "common.h"

namespace {

template <typename T>
__device__ inline T nt_load(const T* p) {
  return __builtin_nontemporal_load(p);
}
template <typename T>
__device__ inline Cplx<T> nt_load(const Cplx<T>* p) {
  return {__builtin_nontemporal_load(&p->re),
          __builtin_nontemporal_load(&p->im)};
}

template <typename T, typename I, int W, bool NT>
__global__ __launch_bounds__(LS_THREADS) void spmv_vector_kernel(
    const i64* __restrict__ indptr, const I* __restrict__ indices,
    const T* __restrict__ vals, const T* __restrict__ x, T* __restrict__ y,
    i64 n_rows, int accumulate) {
  constexpr int ROWS_PER_BLOCK = LS_THREADS / W;
  const int group = threadIdx.x / W;
  const int lane = threadIdx.x % W;
  const i64 stride = (i64)gridDim.x * ROWS_PER_BLOCK;
  for (i64 row = (i64)blockIdx.x * ROWS_PER_BLOCK + group; row < n_rows;
       row += stride) {
    const i64 s = indptr[row];
    const i64 e = indptr[row + 1];
    T acc = ls_zero<T>();
    for (i64 jp = s + lane; jp < e; jp += W) {
      if constexpr (NT)
        acc += nt_load(vals + jp) * x[nt_load(indices + jp)];
      else
        acc += vals[jp] * x[indices[jp]];
    }
    acc = group_reduce_sum<T, W>(acc);
    if (lane == 0) {
      if (accumulate)
        y[row] += acc;
      else
        y[row] = acc;
    }
  }
}

template <typename T, typename I>
void spmv_launch(const i64* indptr, const I* indices, const T* vals,
                 const T* x, T* y, i64 n_rows, i64 nnz, bool accumulate,
                 int w_override, bool nt, hipStream_t stream) {
  const double mean = n_rows > 0 ? (double)nnz / (double)n_rows : 0.0;
  // W = largest power of two <= mean/2 (measured on MI355X: ~2 elements
  // per lane beats 1; profiles/spmv_sweep_r01.txt — W=2 gives 504 GF/s on
  // 5-pt Poisson fp64 vs 442 at W=4).
  int W = 1;
  while (W < 64 && (double)(W * 4) <= mean) W *= 2;
  if (w_override > 0) W = w_override;
  auto launch = [&](auto wtag) {
    constexpr int WS = decltype(wtag)::value;
    constexpr int RPB = LS_THREADS / WS;
    int grid = grid_1d(n_rows, RPB, 8192);
    if (nt)
      hipLaunchKernelGGL((spmv_vector_kernel<T, I, WS, true>), dim3(grid),
                         dim3(LS_THREADS), 0, stream, indptr, indices, vals,
                         x, y, n_rows, accumulate ? 1 : 0);
    else
      hipLaunchKernelGGL((spmv_vector_kernel<T, I, WS, false>), dim3(grid),
                         dim3(LS_THREADS), 0, stream, indptr, indices, vals,
                         x, y, n_rows, accumulate ? 1 : 0);
  };
  switch (W) {
    case 1: launch(std::integral_constant<int, 1>{}); break;
    case 2: launch(std::integral_constant<int, 2>{}); break;
    case 4: launch(std::integral_constant<int, 4>{}); break;
    case 8: launch(std::integral_constant<int, 8>{}); break;
    case 16: launch(std::integral_constant<int, 16>{}); break;
    case 32: launch(std::integral_constant<int, 32>{}); break;
    default: launch(std::integral_constant<int, 64>{}); break;
  }
  ls_check(hipGetLastError(), "spmv");
}

}  // namespace

void ls_spmv(uintptr_t indptr, uintptr_t indices, uintptr_t vals,
             uintptr_t x, uintptr_t y, i64 n_rows, i64 nnz, int dtype,
             int idx_dtype, bool accumulate, uintptr_t stream,
             int w_override, bool nt) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  DISPATCH_VAL_T(dtype, DISPATCH_IDX_T(idx_dtype, (
      spmv_launch<val_t, idx_t>(
          reinterpret_cast<const i64*>(indptr),
          reinterpret_cast<const idx_t*>(indices),
          reinterpret_cast<const val_t*>(vals),
          reinterpret_cast<const val_t*>(x), reinterpret_cast<val_t*>(y),
          n_rows, nnz, accumulate, w_override, nt, s))));
}
