# SPDX-License-Identifier: Apache-2.0
"""SpGEMM vs scipy oracle (reference test_spgemm.py semantics)."""
import numpy as np
import pytest
import scipy.sparse as sp

import legate_sparse as lsp
from testutils import sample_csr, banded_matrix, to_np


def _check(A_s, B_s, tol=1e-12):
    C = lsp.csr_array(A_s) @ lsp.csr_array(B_s)
    Cs = (A_s @ B_s).toarray()
    np.testing.assert_allclose(to_np(C.todense()), Cs, rtol=tol, atol=tol)


@pytest.mark.parametrize("shape", [(10, 10, 10), (20, 15, 12), (7, 30, 7)])
def test_spgemm_random(shape):
    m, k, n = shape
    _check(sample_csr(m, k, 0.3, seed=1), sample_csr(k, n, 0.3, seed=2))


def test_spgemm_banded():
    A = banded_matrix(150, nnz_per_row=5)
    _check(A, A.copy())


@pytest.mark.parametrize("dtype",
                         [np.float32, np.float64, np.complex64,
                          np.complex128])
def test_spgemm_dtypes(dtype):
    tol = 1e-4 if np.dtype(dtype).itemsize <= 8 else 1e-12
    A = sample_csr(12, 14, 0.4, seed=3, dtype=dtype)
    B = sample_csr(14, 9, 0.4, seed=4, dtype=dtype)
    C = lsp.csr_array(A) @ lsp.csr_array(B)
    np.testing.assert_allclose(to_np(C.todense()), (A @ B).toarray(),
                               rtol=tol, atol=tol)


def test_spgemm_empty_rows():
    A = sp.csr_array(np.array([[0, 0, 1.0], [0, 0, 0], [2.0, 0, 0]]))
    B = sp.csr_array(np.array([[0, 1.0, 0], [0, 0, 0], [0, 0, 3.0]]))
    _check(A, B)


def test_spgemm_dense_rows():
    # a fully dense row exercises the largest hash bin on GPU; on CPU it
    # covers the wide-scratch path
    rng = np.random.default_rng(5)
    D = rng.random((30, 30))
    D[1:, :] *= (rng.random((29, 30)) > 0.7)
    A = sp.csr_array(D)
    _check(A, A)


def test_spgemm_mismatched_shapes():
    A = lsp.csr_array((3, 4))
    B = lsp.csr_array((5, 3))
    with pytest.raises(ValueError):
        A @ B


def test_spgemm_indices_sorted():
    A = sample_csr(25, 25, 0.3, seed=6)
    C = lsp.csr_array(A) @ lsp.csr_array(A)
    indptr = to_np(C.indptr)
    indices = to_np(C.indices)
    for i in range(C.shape[0]):
        row = indices[indptr[i]:indptr[i + 1]]
        assert np.all(np.diff(row) > 0)
