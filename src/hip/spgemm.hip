// SPDX-License-Identifier: Apache-2.0
// Hash-based Gustavson SpGEMM for gfx950.
//
// Replaces the reference's cuSPARSE 3-phase path
// (spgemm_csr_csr_csr.cu:196-332) with a 2-phase LDS-hash design:
//
//   phase 0  row_upper_bound: ub[i] = sum of B-row lengths over A's row i
//            (cheap; drives the size binning below)
//   phase 1  symbolic: per C-row distinct-column count via hash insert
//   phase 2  numeric:  hash insert + accumulate, then compact
//
// Rows are binned by ub so the hash table fits LDS (160 KiB/CU):
//   cfg0: ub <=  128 — 16 lanes/row, 16 rows/block, 256-slot table
//   cfg1: ub <= 1024 — 1 wave/row,    2 rows/block, 2048-slot table
//   cfg2: ub <= 4096 — 1 block/row,                 8192-slot table
//   global bin (ub > 4096): per-row power-of-two table in HBM workspace
//     (the ALG3-style memory fallback; reference picks cuSPARSE ALG1/ALG3
//      by the LS_FAST_SPGEMM switch — here the binning is automatic).
//
// Column ids are hashed as uint32 (requires B.cols < 2^32 — asserted on
// the Python side).  Tables are linear-probed, empty slot = 0xFFFFFFFF.
// Output columns are left UNSORTED within a row; the Python layer runs one
// fused canonical sort when needed.

#include "common.h"

namespace {

constexpr uint32_t EMPTY = 0xFFFFFFFFu;

__device__ inline uint32_t hash1(uint32_t col, uint32_t mask) {
  return (col * 2654435761u) & mask;
}

// Insert col into an LDS/global table region; returns slot index or
// 0xFFFFFFFF if it was already present.
__device__ inline uint32_t tbl_insert(uint32_t* tbl, uint32_t mask,
                                      uint32_t col) {
  uint32_t h = hash1(col, mask);
  while (true) {
    uint32_t cur = tbl[h];
    if (cur == col) return EMPTY;
    if (cur == EMPTY) {
      uint32_t prev = atomicCAS(&tbl[h], EMPTY, col);
      if (prev == EMPTY) return h;
      if (prev == col) return EMPTY;
    }
    h = (h + 1) & mask;
  }
}

// Find the slot of an existing col (numeric accumulate path).
__device__ inline uint32_t tbl_find_or_insert(uint32_t* tbl, uint32_t mask,
                                              uint32_t col, bool* fresh) {
  uint32_t h = hash1(col, mask);
  while (true) {
    uint32_t cur = tbl[h];
    if (cur == col) {
      *fresh = false;
      return h;
    }
    if (cur == EMPTY) {
      uint32_t prev = atomicCAS(&tbl[h], EMPTY, col);
      if (prev == EMPTY) {
        *fresh = true;
        return h;
      }
      if (prev == col) {
        *fresh = false;
        return h;
      }
    }
    h = (h + 1) & mask;
  }
}

__device__ inline void atomic_add_val(float* p, float v) { atomicAdd(p, v); }
__device__ inline void atomic_add_val(double* p, double v) { atomicAdd(p, v); }
template <typename T>
__device__ inline void atomic_add_val(Cplx<T>* p, Cplx<T> v) {
  atomicAdd(&p->re, v.re);
  atomicAdd(&p->im, v.im);
}

// ---------------------------------------------------------------------------
// phase 0: upper bound per row
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(LS_THREADS) void row_ub_kernel(
    const i64* __restrict__ A_indptr, const i64* __restrict__ A_indices,
    const i64* __restrict__ B_indptr, i64* __restrict__ ub, i64 n_rows) {
  constexpr int W = 4;
  constexpr int RPB = LS_THREADS / W;
  const int group = threadIdx.x / W;
  const int lane = threadIdx.x % W;
  const i64 stride = (i64)gridDim.x * RPB;
  for (i64 row = (i64)blockIdx.x * RPB + group; row < n_rows; row += stride) {
    i64 acc = 0;
    for (i64 jp = A_indptr[row] + lane; jp < A_indptr[row + 1]; jp += W) {
      const i64 k = A_indices[jp];
      acc += B_indptr[k + 1] - B_indptr[k];
    }
    acc = group_reduce_sum<i64, W>(acc);
    if (lane == 0) ub[row] = acc;
  }
}

// ---------------------------------------------------------------------------
// LDS-hash symbolic/numeric, parameterized on (BLOCK, ROWS_PER_BLOCK, TBL).
// One row per thread-group (W = BLOCK/ROWS threads); exactly one row per
// group per launch index (no grid-stride) so __syncthreads() is uniform.
// ---------------------------------------------------------------------------
template <int BLOCK, int ROWS, int TBL>
__global__ __launch_bounds__(BLOCK) void spgemm_symbolic_lds_kernel(
    const i64* __restrict__ rows_list, i64 n_bin_rows,
    const i64* __restrict__ A_indptr, const i64* __restrict__ A_indices,
    const i64* __restrict__ B_indptr, const i64* __restrict__ B_indices,
    i64* __restrict__ row_nnz) {
  constexpr int W = BLOCK / ROWS;
  __shared__ uint32_t tbl[ROWS][TBL];
  __shared__ int cnt[ROWS];
  const int group = threadIdx.x / W;
  const int lane = threadIdx.x % W;
  const i64 idx = (i64)blockIdx.x * ROWS + group;
  for (int t = lane; t < TBL; t += W) tbl[group][t] = EMPTY;
  if (lane == 0) cnt[group] = 0;
  __syncthreads();
  if (idx < n_bin_rows) {
    const i64 row = rows_list[idx];
    int local = 0;
    for (i64 jp = A_indptr[row] + lane; jp < A_indptr[row + 1]; jp += W) {
      const i64 k = A_indices[jp];
      for (i64 kp = B_indptr[k]; kp < B_indptr[k + 1]; ++kp) {
        if (tbl_insert(tbl[group], TBL - 1,
                       (uint32_t)B_indices[kp]) != EMPTY)
          ++local;
      }
    }
    atomicAdd(&cnt[group], local);
  }
  __syncthreads();
  if (idx < n_bin_rows && lane == 0) row_nnz[rows_list[idx]] = cnt[group];
}

template <typename T, int BLOCK, int ROWS, int TBL>
__global__ __launch_bounds__(BLOCK) void spgemm_numeric_lds_kernel(
    const i64* __restrict__ rows_list, i64 n_bin_rows,
    const i64* __restrict__ A_indptr, const i64* __restrict__ A_indices,
    const T* __restrict__ A_vals, const i64* __restrict__ B_indptr,
    const i64* __restrict__ B_indices, const T* __restrict__ B_vals,
    const i64* __restrict__ C_indptr, i64* __restrict__ C_indices,
    T* __restrict__ C_vals) {
  constexpr int W = BLOCK / ROWS;
  __shared__ uint32_t keys[ROWS][TBL];
  __shared__ T vals[ROWS][TBL];
  __shared__ int cursor[ROWS];
  const int group = threadIdx.x / W;
  const int lane = threadIdx.x % W;
  const i64 idx = (i64)blockIdx.x * ROWS + group;
  // pre-zero both tables: any lane may then atomicAdd into a slot another
  // lane just claimed, with no init race
  for (int t = lane; t < TBL; t += W) {
    keys[group][t] = EMPTY;
    vals[group][t] = ls_zero<T>();
  }
  if (lane == 0) cursor[group] = 0;
  __syncthreads();
  if (idx < n_bin_rows) {
    const i64 row = rows_list[idx];
    for (i64 jp = A_indptr[row] + lane; jp < A_indptr[row + 1]; jp += W) {
      const i64 k = A_indices[jp];
      const T a = A_vals[jp];
      for (i64 kp = B_indptr[k]; kp < B_indptr[k + 1]; ++kp) {
        bool fresh;
        const uint32_t slot = tbl_find_or_insert(
            keys[group], TBL - 1, (uint32_t)B_indices[kp], &fresh);
        atomic_add_val(&vals[group][slot], a * B_vals[kp]);
      }
    }
  }
  __syncthreads();
  if (idx < n_bin_rows) {
    const i64 row = rows_list[idx];
    const i64 base = C_indptr[row];
    for (int t = lane; t < TBL; t += W) {
      const uint32_t c = keys[group][t];
      if (c != EMPTY) {
        const int at = atomicAdd(&cursor[group], 1);
        C_indices[base + at] = (i64)c;
        C_vals[base + at] = vals[group][t];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Global-memory hash bin (ub > LDS capacity): one 256-thread block per row,
// per-row power-of-two table region in workspace (keys pre-filled 0xFF,
// vals pre-zeroed by the host).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(LS_THREADS) void spgemm_symbolic_global_kernel(
    const i64* __restrict__ rows_list, i64 n_bin_rows,
    const i64* __restrict__ A_indptr, const i64* __restrict__ A_indices,
    const i64* __restrict__ B_indptr, const i64* __restrict__ B_indices,
    uint32_t* __restrict__ tbl_ws, const i64* __restrict__ tbl_off,
    const i64* __restrict__ tbl_size, i64* __restrict__ row_nnz) {
  const i64 idx = blockIdx.x;
  if (idx >= n_bin_rows) return;
  const i64 row = rows_list[idx];
  uint32_t* tbl = tbl_ws + tbl_off[idx];
  const uint32_t mask = (uint32_t)tbl_size[idx] - 1;
  __shared__ int cnt;
  if (threadIdx.x == 0) cnt = 0;
  __syncthreads();
  int local = 0;
  for (i64 jp = A_indptr[row] + threadIdx.x; jp < A_indptr[row + 1];
       jp += blockDim.x) {
    const i64 k = A_indices[jp];
    for (i64 kp = B_indptr[k]; kp < B_indptr[k + 1]; ++kp) {
      if (tbl_insert(tbl, mask, (uint32_t)B_indices[kp]) != EMPTY) ++local;
    }
  }
  atomicAdd(&cnt, local);
  __syncthreads();
  if (threadIdx.x == 0) row_nnz[row] = cnt;
}

template <typename T>
__global__ __launch_bounds__(LS_THREADS) void spgemm_numeric_global_kernel(
    const i64* __restrict__ rows_list, i64 n_bin_rows,
    const i64* __restrict__ A_indptr, const i64* __restrict__ A_indices,
    const T* __restrict__ A_vals, const i64* __restrict__ B_indptr,
    const i64* __restrict__ B_indices, const T* __restrict__ B_vals,
    uint32_t* __restrict__ key_ws, T* __restrict__ val_ws,
    const i64* __restrict__ tbl_off, const i64* __restrict__ tbl_size,
    const i64* __restrict__ C_indptr, i64* __restrict__ C_indices,
    T* __restrict__ C_vals) {
  const i64 idx = blockIdx.x;
  if (idx >= n_bin_rows) return;
  const i64 row = rows_list[idx];
  uint32_t* keys = key_ws + tbl_off[idx];
  T* vals = val_ws + tbl_off[idx];
  const uint32_t mask = (uint32_t)tbl_size[idx] - 1;
  __shared__ int cursor;
  if (threadIdx.x == 0) cursor = 0;
  __syncthreads();
  for (i64 jp = A_indptr[row] + threadIdx.x; jp < A_indptr[row + 1];
       jp += blockDim.x) {
    const i64 k = A_indices[jp];
    const T a = A_vals[jp];
    for (i64 kp = B_indptr[k]; kp < B_indptr[k + 1]; ++kp) {
      bool fresh;
      const uint32_t slot = tbl_find_or_insert(
          keys, mask, (uint32_t)B_indices[kp], &fresh);
      // val_ws is pre-zeroed by the host, so no init race exists here.
      atomic_add_val(&vals[slot], a * B_vals[kp]);
    }
  }
  __syncthreads();
  const i64 base = C_indptr[row];
  for (i64 t = threadIdx.x; t <= (i64)mask; t += blockDim.x) {
    const uint32_t c = keys[t];
    if (c != EMPTY) {
      const int at = atomicAdd(&cursor, 1);
      C_indices[base + at] = (i64)c;
      C_vals[base + at] = vals[t];
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Host entry points
// ---------------------------------------------------------------------------
void ls_spgemm_row_ub(uintptr_t A_indptr, uintptr_t A_indices,
                      uintptr_t B_indptr, uintptr_t ub, i64 n_rows,
                      uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  int grid = grid_1d(n_rows, LS_THREADS / 4, 8192);
  hipLaunchKernelGGL(row_ub_kernel, dim3(grid), dim3(LS_THREADS), 0, s,
                     reinterpret_cast<const i64*>(A_indptr),
                     reinterpret_cast<const i64*>(A_indices),
                     reinterpret_cast<const i64*>(B_indptr),
                     reinterpret_cast<i64*>(ub), n_rows);
  ls_check(hipGetLastError(), "spgemm_row_ub");
}

void ls_spgemm_symbolic_lds(int cfg, uintptr_t rows_list, i64 n_bin_rows,
                            uintptr_t A_indptr, uintptr_t A_indices,
                            uintptr_t B_indptr, uintptr_t B_indices,
                            uintptr_t row_nnz, uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  auto rl = reinterpret_cast<const i64*>(rows_list);
  auto ap = reinterpret_cast<const i64*>(A_indptr);
  auto ai = reinterpret_cast<const i64*>(A_indices);
  auto bp = reinterpret_cast<const i64*>(B_indptr);
  auto bi = reinterpret_cast<const i64*>(B_indices);
  auto rn = reinterpret_cast<i64*>(row_nnz);
  if (cfg == 0) {
    const i64 grid = (n_bin_rows + 15) / 16;
    hipLaunchKernelGGL((spgemm_symbolic_lds_kernel<256, 16, 256>),
                       dim3(grid), dim3(256), 0, s, rl, n_bin_rows, ap, ai,
                       bp, bi, rn);
  } else if (cfg == 1) {
    const i64 grid = (n_bin_rows + 1) / 2;
    hipLaunchKernelGGL((spgemm_symbolic_lds_kernel<128, 2, 2048>),
                       dim3(grid), dim3(128), 0, s, rl, n_bin_rows, ap, ai,
                       bp, bi, rn);
  } else {
    hipLaunchKernelGGL((spgemm_symbolic_lds_kernel<256, 1, 8192>),
                       dim3(n_bin_rows), dim3(256), 0, s, rl, n_bin_rows,
                       ap, ai, bp, bi, rn);
  }
  ls_check(hipGetLastError(), "spgemm_symbolic_lds");
}

template <typename T>
static void numeric_lds_dispatch(int cfg, const i64* rl, i64 n,
                                 const i64* ap, const i64* ai, const T* av,
                                 const i64* bp, const i64* bi, const T* bv,
                                 const i64* cp, i64* ci, T* cv,
                                 hipStream_t s) {
  if (cfg == 0) {
    const i64 grid = (n + 15) / 16;
    hipLaunchKernelGGL((spgemm_numeric_lds_kernel<T, 256, 16, 256>),
                       dim3(grid), dim3(256), 0, s, rl, n, ap, ai, av, bp,
                       bi, bv, cp, ci, cv);
  } else if (cfg == 1) {
    const i64 grid = (n + 1) / 2;
    hipLaunchKernelGGL((spgemm_numeric_lds_kernel<T, 128, 2, 2048>),
                       dim3(grid), dim3(128), 0, s, rl, n, ap, ai, av, bp,
                       bi, bv, cp, ci, cv);
  } else {
    // 16-byte values (c128) at TBL=8192 would need 160 KiB + change of
    // LDS; drop to 4096 slots there (bin cutoff is ub <= 4096, so the
    // table can be exactly full — linear probing still terminates).
    constexpr int TBL2 = (sizeof(T) > 8) ? 4096 : 8192;
    hipLaunchKernelGGL((spgemm_numeric_lds_kernel<T, 256, 1, TBL2>),
                       dim3(n), dim3(256), 0, s, rl, n, ap, ai, av, bp, bi,
                       bv, cp, ci, cv);
  }
}

void ls_spgemm_numeric_lds(int cfg, uintptr_t rows_list, i64 n_bin_rows,
                           uintptr_t A_indptr, uintptr_t A_indices,
                           uintptr_t A_vals, uintptr_t B_indptr,
                           uintptr_t B_indices, uintptr_t B_vals,
                           uintptr_t C_indptr, uintptr_t C_indices,
                           uintptr_t C_vals, int dtype, uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  DISPATCH_VAL_T(dtype, numeric_lds_dispatch<val_t>(
      cfg, reinterpret_cast<const i64*>(rows_list), n_bin_rows,
      reinterpret_cast<const i64*>(A_indptr),
      reinterpret_cast<const i64*>(A_indices),
      reinterpret_cast<const val_t*>(A_vals),
      reinterpret_cast<const i64*>(B_indptr),
      reinterpret_cast<const i64*>(B_indices),
      reinterpret_cast<const val_t*>(B_vals),
      reinterpret_cast<const i64*>(C_indptr),
      reinterpret_cast<i64*>(C_indices), reinterpret_cast<val_t*>(C_vals),
      s));
  ls_check(hipGetLastError(), "spgemm_numeric_lds");
}

void ls_spgemm_symbolic_global(uintptr_t rows_list, i64 n_bin_rows,
                               uintptr_t A_indptr, uintptr_t A_indices,
                               uintptr_t B_indptr, uintptr_t B_indices,
                               uintptr_t tbl_ws, uintptr_t tbl_off,
                               uintptr_t tbl_size, uintptr_t row_nnz,
                               uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(spgemm_symbolic_global_kernel, dim3(n_bin_rows),
                     dim3(LS_THREADS), 0, s,
                     reinterpret_cast<const i64*>(rows_list), n_bin_rows,
                     reinterpret_cast<const i64*>(A_indptr),
                     reinterpret_cast<const i64*>(A_indices),
                     reinterpret_cast<const i64*>(B_indptr),
                     reinterpret_cast<const i64*>(B_indices),
                     reinterpret_cast<uint32_t*>(tbl_ws),
                     reinterpret_cast<const i64*>(tbl_off),
                     reinterpret_cast<const i64*>(tbl_size),
                     reinterpret_cast<i64*>(row_nnz));
  ls_check(hipGetLastError(), "spgemm_symbolic_global");
}

void ls_spgemm_numeric_global(uintptr_t rows_list, i64 n_bin_rows,
                              uintptr_t A_indptr, uintptr_t A_indices,
                              uintptr_t A_vals, uintptr_t B_indptr,
                              uintptr_t B_indices, uintptr_t B_vals,
                              uintptr_t key_ws, uintptr_t val_ws,
                              uintptr_t tbl_off, uintptr_t tbl_size,
                              uintptr_t C_indptr, uintptr_t C_indices,
                              uintptr_t C_vals, int dtype, uintptr_t stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  DISPATCH_VAL_T(dtype, hipLaunchKernelGGL(
      (spgemm_numeric_global_kernel<val_t>), dim3(n_bin_rows),
      dim3(LS_THREADS), 0, s, reinterpret_cast<const i64*>(rows_list),
      n_bin_rows, reinterpret_cast<const i64*>(A_indptr),
      reinterpret_cast<const i64*>(A_indices),
      reinterpret_cast<const val_t*>(A_vals),
      reinterpret_cast<const i64*>(B_indptr),
      reinterpret_cast<const i64*>(B_indices),
      reinterpret_cast<const val_t*>(B_vals),
      reinterpret_cast<uint32_t*>(key_ws),
      reinterpret_cast<val_t*>(val_ws),
      reinterpret_cast<const i64*>(tbl_off),
      reinterpret_cast<const i64*>(tbl_size),
      reinterpret_cast<const i64*>(C_indptr),
      reinterpret_cast<i64*>(C_indices),
      reinterpret_cast<val_t*>(C_vals)));
  ls_check(hipGetLastError(), "spgemm_numeric_global");
}
