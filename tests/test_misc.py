# SPDX-License-Identifier: Apache-2.0
"""Utility/edge coverage: dtype promotion, factor_int, error paths,
settings plumbing."""
import numpy as np
import pytest
import torch

import legate_sparse as lsp
from legate_sparse import utils as lsu
from legate_sparse.settings import settings
from testutils import sample_csr, to_np


def test_find_common_type():
    a = np.zeros(3, dtype=np.float32)
    b = np.zeros(3, dtype=np.float64)
    assert lsu.find_common_type(a, b) == np.dtype(np.float64)
    c = np.zeros(3, dtype=np.complex64)
    assert lsu.find_common_type(a, c) == np.dtype(np.complex64)


def test_factor_int():
    for n in (1, 4, 12, 36, 97):
        a, b = lsu.factor_int(n)
        assert a * b == n


def test_is_dtype_supported():
    assert lsu.is_dtype_supported(np.float64)
    assert lsu.is_dtype_supported(np.complex64)
    assert not lsu.is_dtype_supported(np.int32)
    assert not lsu.is_dtype_supported("not-a-dtype")


def test_spmv_dim_mismatch():
    A = lsp.csr_array((4, 6))
    with pytest.raises(ValueError):
        A @ np.ones(5)


def test_matmul_2d_dense_rejected():
    A = lsp.csr_array((4, 4))
    with pytest.raises(NotImplementedError):
        A @ np.ones((4, 2))


def test_diags_bad_format():
    with pytest.raises(NotImplementedError):
        lsp.diags([np.ones(3)], [0], shape=(3, 3), format="coo")


def test_diags_offset_out_of_bounds():
    with pytest.raises(ValueError):
        lsp.diags([np.ones(1)], [5], shape=(3, 3))


def test_settings_env(monkeypatch):
    monkeypatch.setenv("LS_FAST_SPGEMM", "1")
    assert settings.fast_spgemm
    monkeypatch.setenv("LS_FAST_SPGEMM", "0")
    assert not settings.fast_spgemm
    monkeypatch.setenv("LS_PRECISE_IMAGES", "0")
    assert not settings.precise_images


def test_force_fallback_spmv(monkeypatch):
    # LS_FORCE_FALLBACK routes through the torch path; results identical
    S = sample_csr(12, 12, 0.4, seed=3)
    x = np.random.default_rng(4).random(12)
    want = to_np(lsp.csr_array(S) @ x)
    monkeypatch.setenv("LS_FORCE_FALLBACK", "1")
    got = to_np(lsp.csr_array(S) @ x)
    np.testing.assert_allclose(got, want, rtol=1e-12)


def test_local_full_vector_roundtrip():
    v = np.arange(11.0)
    local = lsu.local_vector(v)
    full = lsu.full_vector(local, 11)
    np.testing.assert_allclose(to_np(full), v)


def test_partition_properties():
    from legate_sparse.runtime import Partition
    for n, ws in ((10, 3), (7, 8), (0, 4), (100, 1)):
        p = Partition(n, ws)
        total = sum(p.count(r) for r in range(ws))
        assert total == n
        for r in range(ws - 1):
            assert p.hi(r) == p.lo(r + 1)


def test_exclusive_scan():
    from legate_sparse.comm import exclusive_scan_offsets
    t = torch.tensor([3, 1, 4, 1])
    np.testing.assert_array_equal(to_np(exclusive_scan_offsets(t)),
                                  [0, 3, 4, 8])


def test_astype_nocopy():
    S = sample_csr(5, 5, 0.5, seed=6)
    A = lsp.csr_array(S)
    assert A.astype(np.float64, copy=False) is A
    B = A.astype(np.float32)
    assert B.dtype == np.dtype(np.float32)
