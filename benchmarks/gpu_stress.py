#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""Randomized GPU stress: HIP kernels vs the torch fallback on-device,
sweeping shapes/densities across every SpGEMM bin, SpMV width, and the
conversion/transpose paths.  Run on the MI355X box:
    python benchmarks/gpu_stress.py [--rounds 40] [--seed 0]
Exits nonzero on the first mismatch.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import legate_sparse as lsp
from legate_sparse import ops
from legate_sparse.settings import settings


def rand_csr(rng, m, n, mode):
    if mode == "uniform":
        density = rng.uniform(0.001, 0.02)
        nnz = max(1, int(m * n * density))
        r = rng.integers(0, m, nnz)
        c = rng.integers(0, n, nnz)
    elif mode == "banded":
        half = int(rng.integers(1, 12))
        offs = np.arange(-half, half + 1)
        r = np.repeat(np.arange(m), len(offs))
        c = r.reshape(m, -1) + offs
        keep = (c >= 0) & (c < n)
        r = np.repeat(np.arange(m), len(offs))[keep.ravel()]
        c = c.ravel()[keep.ravel()]
    else:  # powerlaw
        nnz = max(1, int(m * rng.uniform(2, 12)))
        r = (m * rng.power(0.3, nnz)).astype(np.int64) % m
        c = (n * rng.power(0.3, nnz)).astype(np.int64) % n
    v = rng.standard_normal(len(r))
    return lsp.csr_array((v, (r, c)), shape=(m, n))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rounds", type=int, default=40)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    rng = np.random.default_rng(args.seed)

    for rd in range(args.rounds):
        m = int(rng.integers(16, 900))
        k = int(rng.integers(16, 900))
        n = int(rng.integers(16, 900))
        mode = ["uniform", "banded", "powerlaw"][rd % 3]
        A = rand_csr(rng, m, k, mode)
        B = rand_csr(rng, k, n, mode)

        # SpGEMM: HIP vs ESC fallback on device
        C = A @ B
        os.environ["LS_FORCE_FALLBACK"] = "1"
        try:
            ip, ix, vv = ops.spgemm_local(
                A._indptr, A._indices.to(torch.int64), A._data,
                B._indptr, B._indices.to(torch.int64), B._data, n)
        finally:
            del os.environ["LS_FORCE_FALLBACK"]
        ok = (torch.equal(C._indptr, ip)
              and torch.equal(C._indices.to(torch.int64), ix)
              and torch.allclose(C._data, vv, rtol=1e-10, atol=1e-11))
        if not ok:
            print(f"ROUND {rd} MODE {mode} ({m}x{k}x{n}): SPGEMM MISMATCH",
                  flush=True)
            sys.exit(1)

        # SpMV
        x = torch.randn(k, dtype=torch.float64, device="cuda")
        y1 = A @ x
        os.environ["LS_FORCE_FALLBACK"] = "1"
        try:
            y2 = A @ x
        finally:
            del os.environ["LS_FORCE_FALLBACK"]
        if not torch.allclose(y1, y2, rtol=1e-10, atol=1e-11):
            print(f"ROUND {rd}: SPMV MISMATCH", flush=True)
            sys.exit(1)

        # transpose round trip
        if not torch.allclose(A.T.T.todense(), A.todense()):
            print(f"ROUND {rd}: TRANSPOSE MISMATCH", flush=True)
            sys.exit(1)
        if rd % 10 == 9:
            print(f"  {rd + 1}/{args.rounds} rounds ok", flush=True)
    print(f"STRESS OK ({args.rounds} rounds)", flush=True)


if __name__ == "__main__":
    main()
