#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""R-MAT SpGEMM benchmark (R-MAT scale-24 on 8 GPUs).
    python benchmarks/spgemm_rmat.py --scale 20
    torchrun --nproc-per-node 8 benchmarks/spgemm_rmat.py --scale 24
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from legate_sparse.gallery import rmat
from legate_sparse.runtime import runtime


def sync():
    if runtime.world_size > 1:
        dist.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--scale", type=int, default=18)
    ap.add_argument("--edge-factor", type=int, default=16)
    ap.add_argument("-i", "--iters", type=int, default=3)
    args = ap.parse_args()
    t0 = time.perf_counter()
    A = rmat(args.scale, args.edge_factor, seed=1)
    sync()
    t_gen = time.perf_counter() - t0
    a_nnz = A.nnz  # collective: must run on every rank
    if runtime.rank == 0:
        print(f"rmat scale={args.scale}: n={A.shape[0]}, nnz={a_nnz}, "
              f"gen {t_gen*1e3:.0f} ms, ranks={runtime.world_size}",
              flush=True)
    C = A @ A  # warmup
    c_nnz = C.nnz
    del C
    sync()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        C = A @ A
        del C
    sync()
    per = (time.perf_counter() - t0) / args.iters
    if runtime.rank == 0:
        fast = os.environ.get("LS_FAST_SPGEMM", "0")
        print(f"SpGEMM: {per*1e3:.1f} ms/iter, nnz(C)={c_nnz}, "
              f"{c_nnz/per/1e9:.3f} Gnnz/s (LS_FAST_SPGEMM={fast})",
              flush=True)


if __name__ == "__main__":
    main()
