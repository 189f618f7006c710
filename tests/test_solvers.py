# SPDX-License-Identifier: Apache-2.0
"""CG/GMRES/axpby tests (reference test_cg_solve.py / test_gmres_solve.py /
test_cg_axpby.py semantics: residual oracle)."""
import numpy as np
import pytest
import scipy.sparse as sp
import torch

import legate_sparse as lsp
from legate_sparse.linalg import cg_axpby, LinearOperator, IdentityOperator
from testutils import spd_system, sample_csr, to_np


def test_cg_converges():
    n = 64
    S = spd_system(n, seed=1)
    b = np.random.default_rng(2).random(n)
    x, info = lsp.linalg.cg(lsp.csr_array(S), b, rtol=1e-10, maxiter=800)
    assert info == 0
    assert np.linalg.norm(S @ to_np(x) - b) <= 1e-8 * np.linalg.norm(b)


def test_cg_with_x0():
    n = 32
    S = spd_system(n, seed=3)
    b = np.random.default_rng(4).random(n)
    x0 = np.random.default_rng(5).random(n)
    x, info = lsp.linalg.cg(lsp.csr_array(S), b, x0=x0, rtol=1e-10,
                            maxiter=500)
    assert info == 0
    assert np.linalg.norm(S @ to_np(x) - b) <= 1e-8 * np.linalg.norm(b)


def test_cg_jacobi_preconditioner():
    n = 48
    S = spd_system(n, seed=6)
    A = lsp.csr_array(S)
    dinv = 1.0 / to_np(A.diagonal())
    dinv_t = torch.from_numpy(dinv)

    M = LinearOperator((n, n), matvec=lambda r: r * dinv_t,
                       dtype=np.float64)
    b = np.random.default_rng(7).random(n)
    x, info = lsp.linalg.cg(A, b, M=M, rtol=1e-10, maxiter=500)
    assert info == 0
    assert np.linalg.norm(S @ to_np(x) - b) <= 1e-8 * np.linalg.norm(b)


def test_cg_callback_counts():
    n = 32
    S = spd_system(n, seed=8)
    b = np.ones(n)
    count = []
    lsp.linalg.cg(lsp.csr_array(S), b, rtol=1e-10, maxiter=100,
                  callback=lambda x: count.append(1))
    assert len(count) > 0


def test_gmres_converges():
    n = 50
    S = (sample_csr(n, n, 0.3, seed=9) + n * sp.identity(n)).tocsr()
    b = np.random.default_rng(10).random(n)
    x, info = lsp.linalg.gmres(lsp.csr_array(S), b, rtol=1e-10,
                               restart=30, maxiter=600)
    assert info == 0
    assert np.linalg.norm(S @ to_np(x) - b) <= 1e-8 * np.linalg.norm(b)


def test_gmres_nonsymmetric():
    n = 40
    rng = np.random.default_rng(11)
    D = rng.random((n, n)) * (rng.random((n, n)) > 0.7) + n * np.eye(n)
    S = sp.csr_array(D)
    b = rng.random(n)
    x, info = lsp.linalg.gmres(lsp.csr_array(S), b, rtol=1e-12,
                               restart=20, maxiter=400)
    assert info == 0
    assert np.linalg.norm(D @ to_np(x) - b) <= 1e-9 * np.linalg.norm(b)


@pytest.mark.parametrize("isalpha", [True, False])
@pytest.mark.parametrize("negate", [True, False])
def test_cg_axpby(isalpha, negate):
    n = 100
    rng = np.random.default_rng(12)
    y0 = rng.random(n)
    x0 = rng.random(n)
    a, b = 3.0, 2.0
    y = torch.from_numpy(y0.copy())
    x = torch.from_numpy(x0.copy())
    at = torch.tensor([a], dtype=torch.float64)
    bt = torch.tensor([b], dtype=torch.float64)
    cg_axpby(y, x, at, bt, isalpha=isalpha, negate=negate)
    val = (-1 if negate else 1) * a / b
    want = (val * x0 + y0) if isalpha else (x0 + val * y0)
    np.testing.assert_allclose(to_np(y), want, rtol=1e-14)


def test_linear_operator_matvec():
    n = 16
    mat = np.random.default_rng(13).random((n, n))
    op = LinearOperator((n, n),
                        matvec=lambda v: torch.from_numpy(
                            mat @ to_np(v)), dtype=np.float64)
    v = np.ones(n)
    np.testing.assert_allclose(to_np(op.matvec(torch.from_numpy(v))),
                               mat @ v)


def test_identity_operator():
    op = IdentityOperator((5, 5), dtype=np.float64)
    v = torch.arange(5.0)
    np.testing.assert_allclose(to_np(op.matvec(v)), to_np(v))


def test_sparse_operator_rmatvec():
    S = sample_csr(10, 10, 0.4, seed=14, dtype=np.complex128)
    from legate_sparse.linalg import _SparseMatrixLinearOperator
    op = _SparseMatrixLinearOperator(lsp.csr_array(S))
    rng = np.random.default_rng(15)
    v = rng.random(10) + 1j * rng.random(10)
    np.testing.assert_allclose(to_np(op.rmatvec(torch.from_numpy(v))),
                               S.conj().T @ v)


def test_gmres_complex():
    n = 30
    rng = np.random.default_rng(16)
    D = (rng.random((n, n)) + 1j * rng.random((n, n)))
    D = D * (rng.random((n, n)) > 0.6) + n * np.eye(n)
    S = sp.csr_array(D)
    b = rng.random(n) + 1j * rng.random(n)
    x, info = lsp.linalg.gmres(lsp.csr_array(S), b, rtol=1e-10,
                               restart=20, maxiter=300)
    assert info == 0
    assert np.linalg.norm(D @ to_np(x) - b) <= 1e-8 * np.linalg.norm(b)


def test_sparse_norms():
    from legate_sparse.linalg import norm
    import scipy.sparse.linalg as spla
    S = sample_csr(14, 9, 0.4, seed=20)
    A = lsp.csr_array(S)
    assert abs(norm(A) - spla.norm(S)) < 1e-12
    assert abs(norm(A, 1) - spla.norm(S, 1)) < 1e-12
    assert abs(norm(A, np.inf) - spla.norm(S, np.inf)) < 1e-12


def test_cg_zero_rhs():
    """b = 0 returns x = 0 immediately (scipy semantics) instead of
    dividing 0/0 in the fused update."""
    import scipy.sparse as sp
    P = (50 * sp.identity(30) + sp.random(30, 30, 0.2,
                                          random_state=1)).tocsr()
    P = (0.5 * (P + P.T)).tocsr()
    x, info = lsp.linalg.cg(lsp.csr_array(P), np.zeros(30), rtol=1e-10)
    assert info == 0
    assert float(torch.linalg.norm(x)) == 0.0


def test_gmres_edge_restarts():
    """restart > n, restart = 1, maxiter < restart all converge or stop
    cleanly."""
    import scipy.sparse as sp
    rng = np.random.default_rng(0)
    n = 40
    S = (sp.random(n, n, 0.3, format="csr", random_state=1)
         + 10 * sp.identity(n)).tocsr()
    A = lsp.csr_array(S)
    b = rng.random(n)
    x, info = lsp.linalg.gmres(A, b, restart=1000, maxiter=2000,
                               rtol=1e-12)
    assert info == 0
    assert np.linalg.norm(S @ to_np(x) - b) < 1e-8
    x, info = lsp.linalg.gmres(A, b, restart=1, maxiter=500, rtol=1e-10)
    assert info == 0
    x, info = lsp.linalg.gmres(A, b, restart=30, maxiter=5, rtol=1e-14)
    assert info in (0, 5)
