# SPDX-License-Identifier: Apache-2.0
"""Native extension loading.

Two in-tree extensions (built by ``legate_sparse/_build.py``, driven from
``__graft_entry__.build()``):

- ``_cpu_kernels``  — C++/OpenMP kernels (SpMV, SpGEMM), the analogue of the
  reference's cpu/omp task variants (e.g. spmv_omp.cc:36-45).  Used on CPU
  and testable without a GPU.
- ``_hip_kernels``  — hand-written gfx950 HIP kernels (SpMV, SpGEMM, axpby,
  dot, conversions), the analogue of the reference's gpu variants
  (spmv.cu, spgemm_csr_csr_csr.cu, axpby.cu).  REQUIRED on GPU: ops raise
  if a CUDA tensor reaches an op and this module is missing — no silent
  eager fallback.
"""
from __future__ import annotations

import os
import sys

_pkg_dir = os.path.dirname(os.path.abspath(__file__))
if _pkg_dir not in sys.path:
    sys.path.insert(0, _pkg_dir)

try:
    import _cpu_kernels as cpu_kernels  # type: ignore
except ImportError:
    cpu_kernels = None

try:
    import _hip_kernels as hip_kernels  # type: ignore
except ImportError:
    hip_kernels = None


class MissingExtensionError(RuntimeError):
    pass


def require_hip():
    if hip_kernels is None:
        raise MissingExtensionError(
            "legate_sparse._hip_kernels is not built but a GPU tensor "
            "reached a sparse op. Build it with "
            "`python -m legate_sparse._build` (hipcc --offload-arch=gfx950) "
            "— GPU ops never fall back to eager PyTorch."
        )
    return hip_kernels


def require_cpu():
    if cpu_kernels is None:
        raise MissingExtensionError(
            "legate_sparse._cpu_kernels is not built. Build it with "
            "`python -m legate_sparse._build`."
        )
    return cpu_kernels


def has_hip() -> bool:
    return hip_kernels is not None


def has_cpu() -> bool:
    return cpu_kernels is not None
