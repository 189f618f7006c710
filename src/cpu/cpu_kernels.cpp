// SPDX-License-Identifier: Apache-2.0
// CPU/OpenMP kernels for legate_sparse (MI355X-native framework).
//
// These are the CPU counterparts of the reference's cpu/omp task variants:
//   - CSR SpMV row loop        (reference src/sparse/array/csr/spmv.cc:36-43,
//                               spmv_omp.cc:36-45)
//   - Gustavson SpGEMM 2-phase (reference spgemm_csr_csr_csr.cc:38-160,
//                               spgemm_csr_csr_csr_omp.cc:53-167)
// re-designed: plain indptr instead of Legion pos rects, int32/int64
// indices, dense per-thread accumulators, OpenMP dynamic row scheduling.
//
// All pointers arrive as uintptr_t from torch tensors (contiguous,
// host-resident). value dtype codes: 0=f32, 1=f64, 2=c64, 3=c128;
// index dtype codes: 0=int32, 1=int64.

#include <pybind11/pybind11.h>
#include <algorithm>
#include <cstdint>
#include <complex>
#include <vector>
#include <stdexcept>

#ifdef _OPENMP
#include <omp.h>
#endif

namespace py = pybind11;

using i64 = int64_t;

template <typename T, typename I>
static void spmv_impl(const i64* indptr, const I* indices, const T* vals,
                      const T* x, T* y, i64 n_rows, bool accumulate) {
  // `if` clause: a parallel region on a tiny matrix costs more than the
  // work itself (and spins cgroup CPU quota in containers)
#pragma omp parallel for schedule(static) if (n_rows > 16384)
  for (i64 i = 0; i < n_rows; ++i) {
    T acc = accumulate ? y[i] : T(0);
    for (i64 jp = indptr[i]; jp < indptr[i + 1]; ++jp) {
      acc += vals[jp] * x[indices[jp]];
    }
    y[i] = acc;
  }
}

// Gustavson symbolic: row_nnz[i] = |union of B-row col sets over A's row i|.
template <typename I>
static void spgemm_symbolic_impl(const i64* A_indptr, const I* A_indices,
                                 i64 n_rowsA, const i64* B_indptr,
                                 const I* B_indices, i64 n_colsB,
                                 i64* row_nnz) {
#pragma omp parallel if (n_rowsA > 4096)
  {
    std::vector<i64> marker(n_colsB, -1);
#pragma omp for schedule(dynamic, 64)
    for (i64 i = 0; i < n_rowsA; ++i) {
      i64 count = 0;
      for (i64 jp = A_indptr[i]; jp < A_indptr[i + 1]; ++jp) {
        const i64 k = (i64)A_indices[jp];
        for (i64 kp = B_indptr[k]; kp < B_indptr[k + 1]; ++kp) {
          const i64 col = (i64)B_indices[kp];
          if (marker[col] != i) {
            marker[col] = i;
            ++count;
          }
        }
      }
      row_nnz[i] = count;
    }
  }
}

// Gustavson numeric: emits columns in sorted order (unlike the reference,
// which emits insertion order — spgemm_csr_csr_csr.cc:92-160; sorted output
// keeps scipy compatibility bit-clean).
template <typename T, typename I>
static void spgemm_numeric_impl(const i64* A_indptr, const I* A_indices,
                                const T* A_vals, i64 n_rowsA,
                                const i64* B_indptr, const I* B_indices,
                                const T* B_vals, i64 n_colsB,
                                const i64* C_indptr, I* C_indices,
                                T* C_vals) {
#pragma omp parallel if (n_rowsA > 4096)
  {
    std::vector<T> workspace(n_colsB, T(0));
    std::vector<i64> marker(n_colsB, -1);
    std::vector<i64> cols;
#pragma omp for schedule(dynamic, 64)
    for (i64 i = 0; i < n_rowsA; ++i) {
      cols.clear();
      for (i64 jp = A_indptr[i]; jp < A_indptr[i + 1]; ++jp) {
        const i64 k = (i64)A_indices[jp];
        const T a = A_vals[jp];
        for (i64 kp = B_indptr[k]; kp < B_indptr[k + 1]; ++kp) {
          const i64 col = (i64)B_indices[kp];
          if (marker[col] != i) {
            marker[col] = i;
            workspace[col] = a * B_vals[kp];
            cols.push_back(col);
          } else {
            workspace[col] += a * B_vals[kp];
          }
        }
      }
      std::sort(cols.begin(), cols.end());
      i64 out = C_indptr[i];
      for (i64 col : cols) {
        C_indices[out] = (I)col;
        C_vals[out] = workspace[col];
        ++out;
      }
    }
  }
}

enum DtypeCode { F32 = 0, F64 = 1, C64 = 2, C128 = 3 };

#define DISPATCH_VAL(code, CALL)                         \
  switch (code) {                                        \
    case F32: { using scalar_t = float;  CALL; break; }  \
    case F64: { using scalar_t = double; CALL; break; }  \
    case C64: { using scalar_t = std::complex<float>;  CALL; break; } \
    case C128:{ using scalar_t = std::complex<double>; CALL; break; } \
    default: throw std::runtime_error("bad dtype code");             \
  }

#define DISPATCH_IDX(code, CALL)                         \
  switch (code) {                                        \
    case 0: { using index_t = int32_t; CALL; break; }    \
    case 1: { using index_t = int64_t; CALL; break; }    \
    default: throw std::runtime_error("bad index dtype code"); \
  }

static void spmv(uintptr_t indptr, uintptr_t indices, uintptr_t vals,
                 uintptr_t x, uintptr_t y, i64 n_rows, int dtype,
                 int idx_dtype, bool accumulate) {
  DISPATCH_VAL(dtype, DISPATCH_IDX(idx_dtype, (spmv_impl<scalar_t, index_t>(
      reinterpret_cast<const i64*>(indptr),
      reinterpret_cast<const index_t*>(indices),
      reinterpret_cast<const scalar_t*>(vals),
      reinterpret_cast<const scalar_t*>(x),
      reinterpret_cast<scalar_t*>(y), n_rows, accumulate))));
}

static void spgemm_symbolic(uintptr_t A_indptr, uintptr_t A_indices,
                            i64 n_rowsA, uintptr_t B_indptr,
                            uintptr_t B_indices, i64 n_colsB,
                            uintptr_t row_nnz, int idx_dtype) {
  DISPATCH_IDX(idx_dtype, (spgemm_symbolic_impl<index_t>(
      reinterpret_cast<const i64*>(A_indptr),
      reinterpret_cast<const index_t*>(A_indices), n_rowsA,
      reinterpret_cast<const i64*>(B_indptr),
      reinterpret_cast<const index_t*>(B_indices), n_colsB,
      reinterpret_cast<i64*>(row_nnz))));
}

static void spgemm_numeric(uintptr_t A_indptr, uintptr_t A_indices,
                           uintptr_t A_vals, i64 n_rowsA, uintptr_t B_indptr,
                           uintptr_t B_indices, uintptr_t B_vals, i64 n_colsB,
                           uintptr_t C_indptr, uintptr_t C_indices,
                           uintptr_t C_vals, int dtype, int idx_dtype) {
  DISPATCH_VAL(dtype, DISPATCH_IDX(idx_dtype, (
      spgemm_numeric_impl<scalar_t, index_t>(
          reinterpret_cast<const i64*>(A_indptr),
          reinterpret_cast<const index_t*>(A_indices),
          reinterpret_cast<const scalar_t*>(A_vals), n_rowsA,
          reinterpret_cast<const i64*>(B_indptr),
          reinterpret_cast<const index_t*>(B_indices),
          reinterpret_cast<const scalar_t*>(B_vals), n_colsB,
          reinterpret_cast<const i64*>(C_indptr),
          reinterpret_cast<index_t*>(C_indices),
          reinterpret_cast<scalar_t*>(C_vals)))));
}

PYBIND11_MODULE(_cpu_kernels, m) {
  m.doc() = "legate_sparse CPU/OpenMP kernels";
  m.def("spmv", &spmv, py::arg("indptr"), py::arg("indices"), py::arg("vals"),
        py::arg("x"), py::arg("y"), py::arg("n_rows"), py::arg("dtype"),
        py::arg("idx_dtype"), py::arg("accumulate") = false);
  m.def("spgemm_symbolic", &spgemm_symbolic);
  m.def("spgemm_numeric", &spgemm_numeric);
#ifdef _OPENMP
  m.attr("has_openmp") = true;
#else
  m.attr("has_openmp") = false;
#endif
}
