# SPDX-License-Identifier: Apache-2.0
"""Shared helpers for the examples (timing + operators).

Fresh implementations of the roles played by the reference's
examples/common.py (timers, banded matrix builder) on torch/MI355X.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import legate_sparse as lsp
from legate_sparse.runtime import runtime

if not torch.cuda.is_available():
    torch.set_num_threads(1)


def sync():
    import torch.distributed as dist
    if runtime.world_size > 1 and dist.is_initialized():
        dist.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()


class Timer:
    def __enter__(self):
        sync()
        self.t0 = time.perf_counter()
        return self

    def __exit__(self, *a):
        sync()
        self.elapsed = time.perf_counter() - self.t0


def banded_matrix(n, nnz_per_row=11, dtype=np.float64, seed=0):
    """Banded n x n matrix with nnz_per_row diagonals (odd), random
    positive values — the reference microbenchmarks' operator shape
    (spmv_microbenchmark.py), built through the library's diags path."""
    half = nnz_per_row // 2
    offs = list(range(-half, half + 1))
    rng = np.random.default_rng(seed)
    diagonals = [rng.random(n - abs(o)) + 1.0 for o in offs]
    return lsp.diags(diagonals, offs, shape=(n, n), format="csr",
                     dtype=dtype)


def d2_mat_dirichlet_2d(nx, ny, dtype=np.float64):
    """2-D Dirichlet Laplacian (5-diagonal), the pde.py operator.
    Built via the scalable direct path."""
    from legate_sparse.gallery import poisson_2d
    return poisson_2d(nx, ny, dtype=dtype)


def print0(*args, **kw):
    if runtime.rank == 0:
        print(*args, **kw, flush=True)
