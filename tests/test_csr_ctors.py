# SPDX-License-Identifier: Apache-2.0
"""csr_array constructor tests (reference csr.py:89-286 semantics)."""
import numpy as np
import pytest
import scipy.sparse as sp

import legate_sparse as lsp
from testutils import sample_csr, simple_system, to_np


def test_from_dense():
    D, _ = simple_system(13, 7)
    A = lsp.csr_array(D)
    S = sp.csr_array(D)
    assert A.shape == (13, 7)
    assert A.nnz == S.nnz
    np.testing.assert_allclose(to_np(A.todense()), D)
    np.testing.assert_array_equal(to_np(A.indptr), S.indptr)
    np.testing.assert_array_equal(to_np(A.indices), S.indices)
    np.testing.assert_allclose(to_np(A.data), S.data)


def test_from_scipy():
    S = sample_csr(20, 15, 0.25, seed=3)
    A = lsp.csr_array(S)
    np.testing.assert_allclose(to_np(A.todense()), S.toarray())


def test_from_coo():
    S = sample_csr(11, 9, 0.3, seed=4).tocoo()
    A = lsp.csr_array((S.data, (S.row, S.col)), shape=(11, 9))
    np.testing.assert_allclose(to_np(A.todense()), S.toarray())


def test_coo_duplicates_summed():
    rows = np.array([0, 0, 1, 2, 2, 2])
    cols = np.array([1, 1, 0, 2, 2, 1])
    vals = np.array([1.0, 2.0, 3.0, 4.0, 5.0, 6.0])
    A = lsp.csr_array((vals, (rows, cols)), shape=(3, 3))
    S = sp.csr_array((vals, (rows, cols)), shape=(3, 3))
    np.testing.assert_allclose(to_np(A.todense()), S.toarray())


def test_from_csr_triple():
    S = sample_csr(16, 16, 0.2, seed=5)
    A = lsp.csr_array((S.data, S.indices, S.indptr), shape=S.shape)
    np.testing.assert_allclose(to_np(A.todense()), S.toarray())


def test_empty():
    A = lsp.csr_array((6, 8), dtype=np.float64)
    assert A.shape == (6, 8)
    assert A.nnz == 0
    np.testing.assert_allclose(to_np(A.todense()), np.zeros((6, 8)))


def test_copy_ctor_and_copy():
    S = sample_csr(10, 10, 0.3, seed=6)
    A = lsp.csr_array(S)
    B = lsp.csr_array(A)
    C = A.copy()
    B.data.mul_(2.0)
    C.data.mul_(3.0)
    np.testing.assert_allclose(to_np(A.todense()), S.toarray())


def test_unsupported_dtype_raises():
    with pytest.raises(NotImplementedError):
        lsp.csr_array(np.ones((3, 3), dtype=np.int32))
    with pytest.raises(NotImplementedError):
        lsp.csr_array((3, 3), dtype=np.int64)


@pytest.mark.parametrize("dtype",
                         [np.float32, np.float64, np.complex64,
                          np.complex128])
def test_dtypes(dtype):
    S = sample_csr(9, 9, 0.4, seed=7, dtype=dtype)
    A = lsp.csr_array(S)
    assert A.dtype == np.dtype(dtype)
    np.testing.assert_allclose(to_np(A.todense()), S.toarray())


def test_astype():
    S = sample_csr(8, 8, 0.4, seed=8)
    A = lsp.csr_array(S).astype(np.float32)
    assert A.dtype == np.dtype(np.float32)
    np.testing.assert_allclose(to_np(A.todense()), S.toarray(), rtol=1e-6)


def test_data_setter():
    S = sample_csr(8, 8, 0.4, seed=9)
    A = lsp.csr_array(S)
    A.data = np.ones(S.nnz)
    S2 = S.copy()
    S2.data = np.ones(S.nnz)
    np.testing.assert_allclose(to_np(A.todense()), S2.toarray())


def test_issparse_helpers():
    A = lsp.csr_array((3, 3))
    assert lsp.issparse(A) and lsp.isspmatrix(A) and lsp.isspmatrix_csr(A)
    assert not lsp.isspmatrix_csr(np.ones(3))


def test_csr_triple_unsorted_canonicalized():
    """Raw CSR triples with unsorted columns or duplicates are
    canonicalized at construction (every kernel assumes sorted,
    duplicate-free rows; scipy defers to sort_indices/sum_duplicates)."""
    data = np.array([1.0, 2.0, 3.0])
    indices = np.array([2, 0, 1])
    indptr = np.array([0, 2, 3])
    A = lsp.csr_array((data, indices, indptr), shape=(2, 3))
    assert A.indices.tolist() == [0, 2, 1]
    S = sp.csr_array((data, indices, indptr), shape=(2, 3))
    assert np.allclose(to_np(A.todense()), S.toarray())

    d2 = np.array([1.0, 2.0, 5.0])
    i2 = np.array([0, 0, 1])
    p2 = np.array([0, 2, 3])
    B = lsp.csr_array((d2, i2, p2), shape=(2, 2))
    S2 = sp.csr_array((d2, i2, p2), shape=(2, 2))
    S2.sum_duplicates()
    assert B.nnz == S2.nnz
    assert np.allclose(to_np(B.todense()), S2.toarray())
