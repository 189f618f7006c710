# SPDX-License-Identifier: Apache-2.0
"""Seeded randomized cross-check of the full op surface against scipy,
sweeping shapes/densities/dtypes (the CPU analogue of
benchmarks/gpu_stress.py; a 250-round sweep of the same generator ran
clean during development)."""
import numpy as np
import pytest
import scipy.sparse as sp

import legate_sparse as lsp


@pytest.mark.parametrize("seed", [42, 7])
def test_random_ops_vs_scipy(seed):
    rng = np.random.default_rng(seed)
    for rd in range(12):
        m = int(rng.integers(1, 100))
        k = int(rng.integers(1, 100))
        n = int(rng.integers(1, 100))
        d1, d2 = rng.uniform(0, 0.3), rng.uniform(0, 0.3)
        dt = [np.float64, np.float32, np.complex128][rd % 3]
        S = sp.random(m, k, d1, format="csr",
                      random_state=int(rng.integers(1e6))).astype(dt)
        T = sp.random(k, n, d2, format="csr",
                      random_state=int(rng.integers(1e6))).astype(dt)
        A, B = lsp.csr_array(S), lsp.csr_array(T)
        rtol, atol = ((2e-4, 1e-4) if dt == np.float32
                      else (1e-10, 1e-12))
        assert np.allclose((A @ B).todense().numpy(), (S @ T).toarray(),
                           rtol=rtol, atol=atol)
        x = rng.standard_normal(k).astype(dt)
        assert np.allclose((A @ x).numpy(), S @ x, rtol=rtol, atol=atol)
        assert np.allclose(A.T.todense().numpy(), S.T.toarray(),
                           rtol=rtol, atol=atol)
        S2 = sp.random(m, k, d2, format="csr",
                       random_state=int(rng.integers(1e6))).astype(dt)
        A2 = lsp.csr_array(S2)
        assert np.allclose((A + A2).todense().numpy(),
                           (S + S2).toarray(), rtol=rtol, atol=atol)
        assert np.allclose((A - A2).todense().numpy(),
                           (S - S2).toarray(), rtol=rtol, atol=atol)
        if m == k:
            assert np.allclose(A.diagonal().numpy(), S.diagonal(),
                               rtol=rtol, atol=atol)
        sc = 1e-4 if dt == np.float32 else 1e-9
        assert abs(complex(A.sum()) - S.sum()) <= sc * max(
            1, abs(S.sum()))
        assert np.allclose(A.sum(axis=0).numpy(),
                           np.asarray(S.sum(axis=0)).ravel(),
                           rtol=rtol, atol=atol)
        assert np.allclose(A.sum(axis=1).numpy(),
                           np.asarray(S.sum(axis=1)).ravel(),
                           rtol=rtol, atol=atol)
        assert np.allclose(
            lsp.csr_array(A.todense().numpy()).todense().numpy(),
            S.toarray(), rtol=rtol, atol=atol)
