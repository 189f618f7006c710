#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""SpMV microbenchmark — the reference's examples/spmv_microbenchmark.py
metric (ms/iter on banded NxN CSR, N swept x2 between --nmin/--nmax,
default 11 nnz/row, 100 iters after 5 warmups), on the MI355X framework.

    python examples/spmv_microbenchmark.py --nmin 1048576 --nmax 16777216
    torchrun --nproc-per-node 8 examples/spmv_microbenchmark.py ...
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

import legate_sparse as lsp
from legate_sparse.runtime import runtime
from common import Timer, banded_matrix, print0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nmin", type=int, default=2 ** 16)
    ap.add_argument("--nmax", type=int, default=2 ** 22)
    ap.add_argument("--nnz-per-row", type=int, default=11)
    ap.add_argument("-i", "--iters", type=int, default=100)
    ap.add_argument("-w", "--warmup", type=int, default=5)
    ap.add_argument("--repartition", action="store_true",
                    help="alternate A@x->y and A@y->x every iteration "
                         "(the reference's mode that forces Legion to "
                         "re-derive partitions each step, spmv_"
                         "microbenchmark.py:18-29; here halo plans are "
                         "cached per matrix, so the delta measures that "
                         "the SPMD design has no repartitioning cost)")
    args = ap.parse_args()

    n = args.nmin
    print0(f"# ranks={runtime.world_size} nnz/row={args.nnz_per_row}")
    print0(f"# {'N':>12} {'nnz':>14} {'ms/iter':>10} {'GFLOP/s':>10} "
           f"{'GB/s':>8}")
    while n <= args.nmax:
        A = banded_matrix(n, args.nnz_per_row)
        lo, hi = runtime.local_range(n)
        x = torch.rand(hi - lo, dtype=torch.float64, device=runtime.device)
        y = torch.empty_like(x)
        def step(i):
            if args.repartition and i % 2:
                lsp.spmv(A, y, x)
            else:
                lsp.spmv(A, x, y)

        for i in range(args.warmup):
            step(i)
        with Timer() as t:
            for i in range(args.iters):
                step(i)
        per = t.elapsed / args.iters
        nnz = A.nnz
        bpi = 4 if n < 2 ** 31 else 8  # index bytes
        bytes_iter = nnz * (8 + bpi) + n * 24
        print0(f"  {n:>12} {nnz:>14} {per*1e3:>10.4f} "
               f"{2*nnz/per/1e9:>10.1f} {bytes_iter/per/1e9:>8.0f}")
        n *= 2


if __name__ == "__main__":
    main()
