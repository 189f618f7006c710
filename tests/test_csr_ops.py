# SPDX-License-Identifier: Apache-2.0
"""Transpose, diagonal, ufuncs, sum, scalar multiply (reference
test_csr_transpose.py / test_diagonal.py / base.py:111-250 semantics)."""
import numpy as np
import pytest
import scipy.sparse as sp

import legate_sparse as lsp
from testutils import sample_csr, to_np


def test_transpose_roundtrip():
    S = sample_csr(17, 11, 0.3, seed=1)
    A = lsp.csr_array(S)
    np.testing.assert_allclose(to_np(A.T.todense()), S.T.toarray())
    np.testing.assert_allclose(to_np(A.T.T.todense()), S.toarray())


def test_transpose_complex_conj():
    S = sample_csr(9, 12, 0.4, seed=2, dtype=np.complex128)
    A = lsp.csr_array(S)
    AH = A.conj().transpose()
    np.testing.assert_allclose(to_np(AH.todense()), S.conj().T.toarray())


def test_diagonal():
    S = sample_csr(15, 15, 0.4, seed=3)
    A = lsp.csr_array(S)
    np.testing.assert_allclose(to_np(A.diagonal()), S.diagonal())


def test_diagonal_explicit_zeros():
    # explicit zero on the diagonal must be returned as stored
    D = np.array([[1.0, 2.0], [3.0, 0.0]])
    S = sp.csr_array(D)
    A = lsp.csr_array(D)
    np.testing.assert_allclose(to_np(A.diagonal()), S.diagonal())


def test_diagonal_k_not_supported():
    A = lsp.csr_array((4, 4))
    with pytest.raises(NotImplementedError):
        A.diagonal(k=1)


@pytest.mark.parametrize("ufunc", ["sin", "tanh", "sqrt", "expm1", "sign",
                                   "floor", "rint", "log1p"])
def test_unary_ufuncs(ufunc):
    S = sample_csr(10, 10, 0.4, seed=4)
    S.data = np.abs(S.data)  # keep sqrt/log1p in domain
    A = lsp.csr_array(S)
    got = getattr(A, ufunc)()
    want = S.copy()
    want.data = getattr(np, ufunc)(S.data)
    np.testing.assert_allclose(to_np(got.todense()), want.toarray(),
                               rtol=1e-12)


def test_numpy_ufunc_protocol():
    S = sample_csr(8, 8, 0.5, seed=5)
    A = lsp.csr_array(S)
    got = np.sin(A)
    want = S.copy()
    want.data = np.sin(S.data)
    np.testing.assert_allclose(to_np(got.todense()), want.toarray())


def test_scalar_multiply():
    S = sample_csr(9, 9, 0.4, seed=6)
    A = lsp.csr_array(S)
    np.testing.assert_allclose(to_np((A * 2.5).todense()),
                               (S * 2.5).toarray())
    np.testing.assert_allclose(to_np((2.5 * A).todense()),
                               (S * 2.5).toarray())
    with pytest.raises(NotImplementedError):
        A.multiply(np.ones(9))


def test_neg():
    S = sample_csr(7, 7, 0.4, seed=7)
    A = lsp.csr_array(S)
    np.testing.assert_allclose(to_np((-A).todense()), -S.toarray())


def test_sum():
    S = sample_csr(12, 8, 0.4, seed=8)
    A = lsp.csr_array(S)
    assert abs(float(A.sum()) - S.sum()) < 1e-12
    np.testing.assert_allclose(to_np(A.sum(axis=1)),
                               np.asarray(S.sum(axis=1)).ravel())
    np.testing.assert_allclose(to_np(A.sum(axis=0)),
                               np.asarray(S.sum(axis=0)).ravel())


def test_tocoo():
    S = sample_csr(6, 6, 0.5, seed=9)
    A = lsp.csr_array(S)
    r, c, v = A.tocoo()
    got = sp.coo_array((to_np(v), (to_np(r), to_np(c))), shape=(6, 6))
    np.testing.assert_allclose(got.toarray(), S.toarray())


def test_asformat():
    S = sample_csr(5, 5, 0.5, seed=10)
    A = lsp.csr_array(S)
    assert A.asformat("csr") is A
    assert A.asformat(None) is A


def test_getnnz_count_nonzero():
    S = sample_csr(9, 7, 0.4, seed=11)
    A = lsp.csr_array(S)
    assert A.getnnz() == S.nnz
    np.testing.assert_array_equal(to_np(A.getnnz(axis=1)),
                                  np.diff(S.indptr))
    np.testing.assert_array_equal(to_np(A.getnnz(axis=0)),
                                  S.getnnz(axis=0))
    assert A.count_nonzero() == S.count_nonzero()
    assert A.get_shape() == S.shape


def test_rmatmul_and_flags():
    S = sample_csr(10, 8, 0.4, seed=12)
    A = lsp.csr_array(S)
    x = np.random.default_rng(13).random(10)
    np.testing.assert_allclose(to_np(x @ A), x @ S, rtol=1e-12)
    assert A.has_sorted_indices
    assert A.has_canonical_format


def test_add_sub_sparse():
    A = sample_csr(11, 9, 0.3, seed=21)
    B = sample_csr(11, 9, 0.35, seed=22)
    La, Lb = lsp.csr_array(A), lsp.csr_array(B)
    np.testing.assert_allclose(to_np((La + Lb).todense()),
                               (A + B).toarray(), rtol=1e-12, atol=1e-14)
    np.testing.assert_allclose(to_np((La - Lb).todense()),
                               (A - B).toarray(), rtol=1e-12, atol=1e-14)
    with pytest.raises(ValueError):
        La + lsp.csr_array((3, 3))


def test_eliminate_zeros():
    D = np.array([[1.0, 0, 2], [0, 0, 0], [3, 4, 0]])
    A = lsp.csr_array(D)
    A.data[1] = 0.0  # make one stored entry explicitly zero
    A.eliminate_zeros()
    S = sp.csr_array(D)
    S.data[1] = 0.0
    S.eliminate_zeros()
    assert A.nnz == S.nnz
    np.testing.assert_allclose(to_np(A.todense()), S.toarray())
    A.sum_duplicates()  # no-op
