# SPDX-License-Identifier: Apache-2.0
"""Public type aliases.

Mirrors the reference's ``legate_sparse/types.py`` (reference types.py:20-25):
``coord_ty``/``nnz_ty`` are part of the public API surface (module.py:55 there)
and examples import them.  We keep numpy dtypes for API compatibility and map
them to torch dtypes internally.
"""
from __future__ import annotations

import numpy as np
import torch

# Public aliases (reference: types.py:20-25)
coord_ty = np.int64
nnz_ty = np.uint64
int32 = np.int32
int64 = np.int64
uint64 = np.uint64
float32 = np.float32
float64 = np.float64
complex64 = np.complex64
complex128 = np.complex128

# Value dtypes every kernel is instantiated for
# (reference: util/dispatch.h:56-77 and utils.py:28-33).
SUPPORTED_VALUE_DTYPES = (np.float32, np.float64, np.complex64, np.complex128)

_NP_TO_TORCH = {
    np.dtype(np.float32): torch.float32,
    np.dtype(np.float64): torch.float64,
    np.dtype(np.complex64): torch.complex64,
    np.dtype(np.complex128): torch.complex128,
    np.dtype(np.int32): torch.int32,
    np.dtype(np.int64): torch.int64,
    np.dtype(np.uint64): torch.uint64,
    np.dtype(np.uint8): torch.uint8,
    np.dtype(np.bool_): torch.bool,
}

_TORCH_TO_NP = {v: k for k, v in _NP_TO_TORCH.items()}


def to_torch_dtype(dt) -> torch.dtype:
    if isinstance(dt, torch.dtype):
        return dt
    return _NP_TO_TORCH[np.dtype(dt)]


def to_numpy_dtype(dt) -> np.dtype:
    if isinstance(dt, torch.dtype):
        return _TORCH_TO_NP[dt]
    return np.dtype(dt)


def is_supported_value_dtype(dt) -> bool:
    try:
        npdt = to_numpy_dtype(dt)
    except (KeyError, TypeError):
        return False
    return npdt in (np.dtype(d) for d in SUPPORTED_VALUE_DTYPES)
