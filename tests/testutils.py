# SPDX-License-Identifier: Apache-2.0
"""Test fixtures mirroring the reference's tests/integration/utils
(sample.py:26-55, banded_matrix.py:20-101) — re-written, scipy-oracle."""
import numpy as np
import scipy.sparse as sp


def sample_csr(m, n, density=0.3, seed=0, dtype=np.float64):
    """Seeded scipy.sparse.random CSR with normal values."""
    rng = np.random.default_rng(seed)
    A = sp.random(m, n, density=density, random_state=np.random.RandomState(
        seed), format="csr", dtype=np.float64)
    A.data = rng.standard_normal(A.nnz)
    if np.dtype(dtype).kind == "c":
        A = A.astype(dtype)
        A.data += 1j * rng.standard_normal(A.nnz)
    else:
        A = A.astype(dtype)
    A.sort_indices()
    return A


def simple_system(m, n, seed=0, thresh=0.5):
    """Random dense thresholded to ~50% sparsity + random x."""
    rng = np.random.default_rng(seed)
    D = rng.random((m, n))
    D[D < thresh] = 0.0
    x = rng.random(n)
    return D, x


def spd_system(n, density=0.3, seed=0):
    """SPD via 0.5(A+A^T) + N*I (reference test_cg_solve.py:26-29)."""
    A = sample_csr(n, n, density, seed)
    A = 0.5 * (A + A.T) + n * sp.identity(n)
    return A.tocsr()


def banded_matrix(n, nnz_per_row=11, seed=0, dtype=np.float64):
    """Banded matrix like the reference microbenchmarks
    (examples/common.py:206-249 semantics, independent construction)."""
    half = nnz_per_row // 2
    offs = list(range(-half, half + 1))
    rng = np.random.default_rng(seed)
    diags = [rng.random(n - abs(o)) + 1.0 for o in offs]
    return sp.diags(diags, offs, shape=(n, n), format="csr").astype(dtype)


def to_np(t):
    import torch
    if isinstance(t, torch.Tensor):
        return t.detach().cpu().numpy()
    return np.asarray(t)
