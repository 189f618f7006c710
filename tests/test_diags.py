# SPDX-License-Identifier: Apache-2.0
"""diags/eye vs scipy (reference test_diags.py semantics)."""
import numpy as np
import pytest
import scipy.sparse as sp

import legate_sparse as lsp
from testutils import to_np


@pytest.mark.parametrize("offsets", [[0], [0, 1], [-1, 0, 1], [-2, 0, 3]])
def test_diags_square(offsets):
    n = 9
    diagonals = [np.random.default_rng(o + 5).random(n - abs(o))
                 for o in offsets]
    got = lsp.diags(diagonals, offsets, shape=(n, n), format="csr")
    want = sp.diags(diagonals, offsets, shape=(n, n), format="csr")
    np.testing.assert_allclose(to_np(got.todense()), want.toarray())


def test_diags_rect():
    got = lsp.diags([np.ones(4), np.full(4, 2.0)], [0, -1], shape=(5, 4),
                    format="csr")
    want = sp.diags([np.ones(4), np.full(4, 2.0)], [0, -1], shape=(5, 4),
                    format="csr")
    np.testing.assert_allclose(to_np(got.todense()), want.toarray())


def test_diags_scalar_broadcast():
    got = lsp.diags([2.0], [0], shape=(6, 6), format="csr")
    want = sp.diags([2.0], [0], shape=(6, 6), format="csr")
    np.testing.assert_allclose(to_np(got.todense()), want.toarray())


def test_diags_infer_shape():
    d = [np.arange(1.0, 5.0), np.arange(1.0, 4.0)]
    got = lsp.diags(d, [0, 1], format="csr")
    want = sp.diags(d, [0, 1], format="csr")
    assert got.shape == want.shape
    np.testing.assert_allclose(to_np(got.todense()), want.toarray())


def test_diags_dia_format():
    got = lsp.diags([np.ones(5)], [0])
    assert got.format == "dia"
    np.testing.assert_allclose(to_np(got.tocsr().todense()), np.eye(5))


def test_eye_identity():
    np.testing.assert_allclose(
        to_np(lsp.eye(7, format="csr").todense()), np.eye(7))
    np.testing.assert_allclose(
        to_np(lsp.eye(5, 8, k=2, format="csr").todense()),
        sp.eye(5, 8, k=2).toarray())
    np.testing.assert_allclose(
        to_np(lsp.identity(4, format="csr").todense()), np.eye(4))


def test_dia_array_tocsr():
    data = np.array([[1.0, 2, 3, 4], [5, 6, 7, 8]])
    offsets = np.array([0, -1])
    got = lsp.dia_array((data, offsets), shape=(4, 4)).tocsr()
    want = sp.dia_array((data, offsets), shape=(4, 4)).tocsr()
    np.testing.assert_allclose(to_np(got.todense()), want.toarray())


def test_dia_transpose():
    data = np.array([[1.0, 2, 3, 4], [5, 6, 7, 8]])
    offsets = np.array([0, 1])
    got = lsp.dia_array((data, offsets), shape=(4, 4)).T.tocsr()
    want = sp.dia_array((data, offsets), shape=(4, 4)).T.tocsr()
    np.testing.assert_allclose(to_np(got.todense()), want.toarray())


def test_dia_nnz():
    data = np.ones((2, 5))
    offsets = np.array([0, 2])
    got = lsp.dia_array((data, offsets), shape=(5, 5))
    want = sp.dia_array((data, offsets), shape=(5, 5))
    assert got.nnz == want.nnz


def test_dia_ufuncs_and_astype():
    import numpy as np
    data = np.abs(np.random.default_rng(3).standard_normal((2, 6))) + 0.1
    offsets = np.array([0, 1])
    A = lsp.dia_array((data, offsets), shape=(6, 6))
    got = A.sqrt().tocsr()
    want = sp.dia_array((np.sqrt(data), offsets), shape=(6, 6)).tocsr()
    np.testing.assert_allclose(to_np(got.todense()), want.toarray())
    assert A.astype(np.float32).dtype == np.dtype(np.float32)


def test_stencil_grid_matches_poisson():
    """5-pt stencil through stencil_grid == the direct poisson builder
    (transposed grid convention: stencil_grid is (rows, cols))."""
    from legate_sparse.gallery import stencil_grid, poisson_2d
    S5 = np.array([[0, -1, 0], [-1, 4, -1], [0, -1, 0]], dtype=float)
    A = stencil_grid(S5, (9, 7))
    B = poisson_2d(7, 9)
    assert np.allclose(to_np(A.todense()), to_np(B.todense()))


def test_stencil_grid_9pt_boundaries():
    """9-pt stencil: interior row has 9 entries, corner row loses the
    out-of-grid couplings; matches a dense construction."""
    from legate_sparse.gallery import stencil_grid
    rng = np.random.default_rng(3)
    S = rng.standard_normal((3, 3))
    nx, ny = 5, 6
    A = to_np(stencil_grid(S, (nx, ny)).todense())
    want = np.zeros((nx * ny, nx * ny))
    for i in range(nx):
        for j in range(ny):
            for di in (-1, 0, 1):
                for dj in (-1, 0, 1):
                    ii, jj = i + di, j + dj
                    if 0 <= ii < nx and 0 <= jj < ny:
                        want[i * ny + j, ii * ny + jj] = S[di + 1, dj + 1]
    assert np.allclose(A, want)


def test_diffusion_2d_spd():
    """Rotated anisotropic diffusion (reference examples/common.py:
    330-347 instance): symmetric positive definite, zero interior row
    sums (conservation)."""
    from legate_sparse.gallery import diffusion_2d
    D = to_np(diffusion_2d(8, 0.1, np.pi / 4).todense())
    assert np.allclose(D, D.T)
    assert np.linalg.eigvalsh(D).min() > 0
    assert abs(D[3 * 8 + 4].sum()) < 1e-12


def test_dia_duplicate_offsets_rejected():
    """scipy parity: duplicate offsets raise (silently keeping them
    would make non-canonical CSR rows on conversion)."""
    with pytest.raises(ValueError):
        lsp.dia_array((np.ones((2, 4)), np.array([0, 0])), shape=(4, 4))
