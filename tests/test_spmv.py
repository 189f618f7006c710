# SPDX-License-Identifier: Apache-2.0
"""SpMV vs dense oracle (reference test_spmv.py semantics)."""
import numpy as np
import pytest
import torch

import legate_sparse as lsp
from testutils import sample_csr, simple_system, banded_matrix, to_np


@pytest.mark.parametrize("shape", [(10, 10), (31, 17), (1, 5), (64, 64)])
def test_spmv_random(shape):
    D, x = simple_system(*shape, seed=1)
    A = lsp.csr_array(D)
    np.testing.assert_allclose(to_np(A @ x), D @ x, rtol=1e-12)


@pytest.mark.parametrize("dtype",
                         [np.float32, np.float64, np.complex64,
                          np.complex128])
def test_spmv_dtypes(dtype):
    S = sample_csr(23, 19, 0.3, seed=2, dtype=dtype)
    rng = np.random.default_rng(3)
    x = rng.random(19).astype(dtype)
    if np.dtype(dtype).kind == "c":
        x = x + 1j * rng.random(19).astype(dtype)
    tol = 1e-5 if np.dtype(dtype).itemsize <= 8 else 1e-12
    np.testing.assert_allclose(to_np(lsp.csr_array(S) @ x), S @ x, rtol=tol)


def test_spmv_out_param():
    D, x = simple_system(12, 12, seed=4)
    A = lsp.csr_array(D)
    out = torch.zeros(12, dtype=torch.float64)
    y = A.dot(x, out=out)
    assert y.data_ptr() == out.data_ptr()
    np.testing.assert_allclose(to_np(out), D @ x)


def test_spmv_2d_column():
    D, x = simple_system(9, 9, seed=5)
    A = lsp.csr_array(D)
    y = A @ x.reshape(-1, 1)
    assert y.shape == (9, 1)
    np.testing.assert_allclose(to_np(y).ravel(), D @ x)


def test_spmv_banded():
    S = banded_matrix(200, nnz_per_row=11)
    x = np.random.default_rng(6).random(200)
    np.testing.assert_allclose(to_np(lsp.csr_array(S) @ x), S @ x)


def test_spmv_promotes_dtype():
    S = sample_csr(8, 8, 0.5, seed=7, dtype=np.float32)
    x = np.random.default_rng(8).random(8)  # float64
    y = lsp.csr_array(S) @ x
    assert y.dtype == torch.float64


def test_spmv_unsupported_dtype():
    S = sample_csr(6, 6, 0.5, seed=9)
    A = lsp.csr_array(S)
    with pytest.raises((NotImplementedError, KeyError, TypeError)):
        A @ np.ones(6, dtype=np.int64)


def test_module_level_spmv():
    D, x = simple_system(10, 10, seed=10)
    A = lsp.csr_array(D)
    y = torch.zeros(10, dtype=torch.float64)
    lsp.spmv(A, x, y)
    np.testing.assert_allclose(to_np(y), D @ x)
