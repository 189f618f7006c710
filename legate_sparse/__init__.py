# SPDX-License-Identifier: Apache-2.0
"""legate_sparse — an MI355X-native distributed sparse-array library.

Drop-in scipy.sparse-style API (CSR + DIA, SpMV, SpGEMM, transpose, diags,
mmread, CG/GMRES) built from scratch for AMD MI355X (gfx950):
PyTorch-ROCm tensors, hand-written CDNA4 HIP kernels, RCCL collectives
over xGMI with explicit 1-D row partitioning (one process per GPU).

Same capabilities as nv-legate/legate-sparse (the reference blueprint —
see SURVEY.md), none of its Legion/CUDA machinery.
"""
from __future__ import annotations

import scipy.sparse as _sp

from .module import (  # noqa: F401
    csr_array, csr_matrix, dia_array, dia_matrix, diags, eye, identity,
    mmread, mmwrite, save_npz, load_npz, spmv, spgemm_csr_csr_csr,
    is_sparse_matrix, issparse,
    isspmatrix, isspmatrix_csr, isspmatrix_dia, coord_ty, nnz_ty,
)
from . import linalg  # noqa: F401
from . import types  # noqa: F401
from . import utils  # noqa: F401
from .runtime import runtime  # noqa: F401
from .coverage import clone_module

__version__ = "0.1.0"

# scipy fall-through for uncovered names (reference __init__.py:26)
clone_module(_sp, globals())
del _sp, clone_module
