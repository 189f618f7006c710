# SPDX-License-Identifier: Apache-2.0
"""Public namespace re-exports (reference legate_sparse/module.py:49-70)."""
from __future__ import annotations

from .csr import csr_array, csr_matrix, spmv, spgemm_csr_csr_csr  # noqa
from .dia import dia_array, dia_matrix  # noqa
from .gallery import diags, eye, identity  # noqa
from .io import mmread, mmwrite  # noqa
from .types import coord_ty, nnz_ty  # noqa


def is_sparse_matrix(o) -> bool:
    return isinstance(o, (csr_array, dia_array))


def issparse(o) -> bool:
    return is_sparse_matrix(o)


def isspmatrix(o) -> bool:
    return is_sparse_matrix(o)


def isspmatrix_csr(o) -> bool:
    return isinstance(o, csr_array)


def isspmatrix_dia(o) -> bool:
    return isinstance(o, dia_array)
