// SPDX-License-Identifier: Apache-2.0
// Format conversion + diagonal kernels for gfx950.
// (reference: csr_to_dense.cu:26-47, dense_to_csr.cu:26-149,
//  get_diagonal.cu:26-44 — re-designed as wave-group grid-stride kernels)

#include "common.h"

namespace {

template <typename T, typename I, int W>
__global__ __launch_bounds__(LS_THREADS) void csr_to_dense_kernel(
    const i64* __restrict__ indptr, const I* __restrict__ indices,
    const T* __restrict__ vals, T* __restrict__ dense, i64 n_rows,
    i64 n_cols) {
  constexpr int RPB = LS_THREADS / W;
  const int group = threadIdx.x / W;
  const int lane = threadIdx.x % W;
  const i64 stride = (i64)gridDim.x * RPB;
  for (i64 row = (i64)blockIdx.x * RPB + group; row < n_rows; row += stride) {
    T* out = dense + row * n_cols;
    for (i64 jp = indptr[row] + lane; jp < indptr[row + 1]; jp += W) {
      out[indices[jp]] = vals[jp];
    }
  }
}

template <typename T>
__global__ __launch_bounds__(LS_THREADS) void dense_to_csr_nnz_kernel(
    const T* __restrict__ dense, i64* __restrict__ row_nnz, i64 n_rows,
    i64 n_cols) {
  constexpr int W = 16;
  constexpr int RPB = LS_THREADS / W;
  const int group = threadIdx.x / W;
  const int lane = threadIdx.x % W;
  const i64 stride = (i64)gridDim.x * RPB;
  for (i64 row = (i64)blockIdx.x * RPB + group; row < n_rows; row += stride) {
    const T* in = dense + row * n_cols;
    i64 cnt = 0;
    for (i64 c = lane; c < n_cols; c += W) {
      T v = in[c];
      if constexpr (is_cplx<T>::value) {
        if (v.re != 0 || v.im != 0) ++cnt;
      } else {
        if (v != 0) ++cnt;
      }
    }
    cnt = group_reduce_sum<i64, W>(cnt);
    if (lane == 0) row_nnz[row] = cnt;
  }
}

// fill: one thread per row (sequential cursor keeps columns sorted, like
// the reference's denseToCSRKernel, dense_to_csr.cu:128-149)
template <typename T, typename I>
__global__ __launch_bounds__(LS_THREADS) void dense_to_csr_fill_kernel(
    const T* __restrict__ dense, const i64* __restrict__ indptr,
    I* __restrict__ indices, T* __restrict__ vals, i64 n_rows,
    i64 n_cols) {
  const i64 stride = (i64)gridDim.x * blockDim.x;
  for (i64 row = (i64)blockIdx.x * blockDim.x + threadIdx.x; row < n_rows;
       row += stride) {
    const T* in = dense + row * n_cols;
    i64 out = indptr[row];
    for (i64 c = 0; c < n_cols; ++c) {
      T v = in[c];
      bool nz;
      if constexpr (is_cplx<T>::value)
        nz = (v.re != 0 || v.im != 0);
      else
        nz = (v != 0);
      if (nz) {
        indices[out] = (I)c;
        vals[out] = v;
        ++out;
      }
    }
  }
}

template <typename T, typename I>
__global__ __launch_bounds__(LS_THREADS) void diagonal_kernel(
    const i64* __restrict__ indptr, const I* __restrict__ indices,
    const T* __restrict__ vals, T* __restrict__ out, i64 n_rows,
    i64 row_offset) {
  constexpr int W = 4;
  constexpr int RPB = LS_THREADS / W;
  const int group = threadIdx.x / W;
  const int lane = threadIdx.x % W;
  const i64 stride = (i64)gridDim.x * RPB;
  for (i64 row = (i64)blockIdx.x * RPB + group; row < n_rows; row += stride) {
    const i64 want = row + row_offset;
    for (i64 jp = indptr[row] + lane; jp < indptr[row + 1]; jp += W) {
      if ((i64)indices[jp] == want) out[row] = vals[jp];
    }
  }
}

}  // namespace

void ls_csr_to_dense(uintptr_t indptr, uintptr_t indices, uintptr_t vals,
                     uintptr_t dense, i64 n_rows, i64 n_cols, int dtype,
                     int idx_dtype, uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  int grid = grid_1d(n_rows, LS_THREADS / 4, 8192);
  DISPATCH_VAL_T(dtype, DISPATCH_IDX_T(idx_dtype, hipLaunchKernelGGL(
      (csr_to_dense_kernel<val_t, idx_t, 4>), dim3(grid), dim3(LS_THREADS),
      0, s, reinterpret_cast<const i64*>(indptr),
      reinterpret_cast<const idx_t*>(indices),
      reinterpret_cast<const val_t*>(vals),
      reinterpret_cast<val_t*>(dense), n_rows, n_cols)));
  ls_check(hipGetLastError(), "csr_to_dense");
}

void ls_dense_to_csr_nnz(uintptr_t dense, uintptr_t row_nnz, i64 n_rows,
                         i64 n_cols, int dtype, uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  int grid = grid_1d(n_rows, LS_THREADS / 16, 8192);
  DISPATCH_VAL_T(dtype, hipLaunchKernelGGL(
      (dense_to_csr_nnz_kernel<val_t>), dim3(grid), dim3(LS_THREADS), 0, s,
      reinterpret_cast<const val_t*>(dense),
      reinterpret_cast<i64*>(row_nnz), n_rows, n_cols));
  ls_check(hipGetLastError(), "dense_to_csr_nnz");
}

void ls_dense_to_csr_fill(uintptr_t dense, uintptr_t indptr,
                          uintptr_t indices, uintptr_t vals, i64 n_rows,
                          i64 n_cols, int dtype, int idx_dtype,
                          uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  int grid = grid_1d(n_rows, LS_THREADS, 8192);
  DISPATCH_VAL_T(dtype, DISPATCH_IDX_T(idx_dtype, hipLaunchKernelGGL(
      (dense_to_csr_fill_kernel<val_t, idx_t>), dim3(grid),
      dim3(LS_THREADS), 0, s, reinterpret_cast<const val_t*>(dense),
      reinterpret_cast<const i64*>(indptr),
      reinterpret_cast<idx_t*>(indices), reinterpret_cast<val_t*>(vals),
      n_rows, n_cols)));
  ls_check(hipGetLastError(), "dense_to_csr_fill");
}

void ls_diagonal(uintptr_t indptr, uintptr_t indices, uintptr_t vals,
                 uintptr_t out, i64 n_rows, i64 row_offset, int dtype,
                 int idx_dtype, uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  int grid = grid_1d(n_rows, LS_THREADS / 4, 8192);
  DISPATCH_VAL_T(dtype, DISPATCH_IDX_T(idx_dtype, hipLaunchKernelGGL(
      (diagonal_kernel<val_t, idx_t>), dim3(grid), dim3(LS_THREADS), 0, s,
      reinterpret_cast<const i64*>(indptr),
      reinterpret_cast<const idx_t*>(indices),
      reinterpret_cast<const val_t*>(vals),
      reinterpret_cast<val_t*>(out), n_rows, row_offset)));
  ls_check(hipGetLastError(), "diagonal");
}
