#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""SpGEMM microbenchmark — the reference's
examples/spgemm_microbenchmark.py metric (ms/iter for banded A @ A,
default 5 nnz/row; prints nnz(A) x nnz(B) -> nnz(C)), on the MI355X
framework.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import legate_sparse as lsp
from legate_sparse.runtime import runtime
from common import Timer, banded_matrix, print0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-n", type=int, default=2 ** 20)
    ap.add_argument("--nnz-per-row", type=int, default=5)
    ap.add_argument("-i", "--iters", type=int, default=10)
    ap.add_argument("-w", "--warmup", type=int, default=2)
    ap.add_argument("--fresh", action="store_true",
                    help="rebuild the operands every iteration (the "
                         "reference's non --stable mode: includes "
                         "construction and plan setup)")
    args = ap.parse_args()

    A = banded_matrix(args.n, args.nnz_per_row, seed=1)
    B = banded_matrix(args.n, args.nnz_per_row, seed=2)
    for _ in range(args.warmup):
        C = A @ B
    c_nnz = C.nnz
    del C
    with Timer() as t:
        for it in range(args.iters):
            if args.fresh:
                A = banded_matrix(args.n, args.nnz_per_row, seed=1)
                B = banded_matrix(args.n, args.nnz_per_row, seed=2)
            C = A @ B
            del C
    per = t.elapsed / args.iters
    print0(f"SpGEMM n={args.n}: nnz(A)={A.nnz} x nnz(B)={B.nnz} -> "
           f"nnz(C)={c_nnz}")
    print0(f"  {per*1e3:.3f} ms/iter, {c_nnz/per/1e9:.3f} Gnnz(C)/s, "
           f"ranks={runtime.world_size}")


if __name__ == "__main__":
    main()
