#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""Within-process interleaved A/B sweep of SpMV variants (guide §5.4
rule 24: interleave variants in ONE process, report median).

Sweeps {W} x {nt} x {index width} on the 5-pt Poisson operator and a
banded matrix.  Run on the GPU box:
    python benchmarks/spmv_sweep.py [--nx 4096] [--rounds 7]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from legate_sparse import ops
from legate_sparse.gallery import poisson_2d
from legate_sparse.runtime import runtime


def time_variant(A, x, y, w, pair, swz, iters=30):
    idp, idx, dat = A._indptr, A._indices, A._data
    for _ in range(3):
        ops.spmv(idp, idx, dat, x, y, w_override=w, pair=pair, swz=swz)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.spmv(idp, idx, dat, x, y, w_override=w, pair=pair, swz=swz)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nx", type=int, default=4096)
    ap.add_argument("--rounds", type=int, default=7)
    ap.add_argument("--iters", type=int, default=30)
    args = ap.parse_args()
    assert torch.cuda.is_available()

    A = poisson_2d(args.nx, args.nx)
    n = A.shape[0]
    nnz = A.nnz
    x = torch.rand(n, dtype=torch.float64, device="cuda")
    y = torch.empty_like(x)
    # int64-index clone (bypasses _init_local's int32 cast)
    import copy
    A64 = copy.copy(A)
    A64._indices = A._indices.to(torch.int64).contiguous()

    variants = []
    for w in (1, 2, 4):
        for pair in (0, 1):
            variants.append(("i32", A, w, pair, 0))
    variants.append(("i32", A, 2, 2, 0))   # stream PE=8
    variants.append(("i32", A, 2, 3, 0))   # stream PE=16

    results = {v[:1] + v[2:]: [] for v in variants}
    for r in range(args.rounds):
        for tag, mat, w, pair, swz in variants:
            dt = time_variant(mat, x, y, w, pair, swz, args.iters)
            results[(tag, w, pair, swz)].append(dt)
    print(f"# nx={args.nx} rows={n} nnz={nnz}")
    for (tag, w, pair, swz), ts in sorted(results.items()):
        med = np.median(ts) * 1e3
        mn = min(ts) * 1e3
        gf = 2 * nnz / (np.median(ts)) / 1e9
        print(f"idx={tag} W={w} pair={pair} swz={swz}: median {med:.4f} ms "
              f"(min {mn:.4f})  {gf:7.1f} GFLOP/s")


if __name__ == "__main__":
    main()
