#!/usr/bin/env python3
# SPDX-License-Identifier: Apache-2.0
"""Geometric multigrid preconditioned CG — the reference's examples/gmg.py
workload (GMG solve ms/iter, Poisson 2-D), re-implemented on
the MI355X framework.

The grid hierarchy uses interior-point grids nx = 2^k - 1 so vertex
coarsening is exact: coarse point I sits at fine index 2I+1.  The
prolongation P is bilinear interpolation (built directly as distributed
CSR), restriction R = P^T (Galerkin), coarse operators A_c = R @ A @ P via
the framework's SpGEMM, smoothing is weighted Jacobi — every building
block is a framework op (SpMV, SpGEMM, transpose, diagonal, axpby).

    python examples/gmg.py -k 9          # 511 x 511 grid
    torchrun --nproc-per-node 8 examples/gmg.py -k 12
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np
import torch

import legate_sparse as lsp
from legate_sparse.csr import csr_array
from legate_sparse.linalg import LinearOperator
from legate_sparse.runtime import runtime
from common import Timer, print0


def interp_1d_pairs(i, nxc):
    """1-D bilinear interpolation stencil for fine index i (torch,
    vectorized): returns (c0, w0, c1, w1) with c<0 meaning absent."""
    odd = (i % 2) == 1
    cI = torch.div(i - 1, 2, rounding_mode="floor")
    half = torch.div(i, 2, rounding_mode="floor")
    c0 = torch.where(odd, cI, half - 1)
    w0 = torch.where(odd, torch.ones_like(i, dtype=torch.float64),
                     torch.full_like(i, 0.5, dtype=torch.float64))
    c1 = torch.where(odd, torch.full_like(i, -1), half)
    w1 = torch.full_like(i, 0.5, dtype=torch.float64)
    c0 = torch.where(c0 >= 0, c0, torch.full_like(c0, -1))
    c1 = torch.where(c1 < nxc, c1, torch.full_like(c1, -1))
    return c0, w0, c1, w1


def build_prolongation(nxf, nxc):
    """Distributed bilinear P: (nxf^2, nxc^2), local fine rows."""
    dev = runtime.device
    nf = nxf * nxf
    lo, hi = runtime.local_range(nf)
    r = torch.arange(lo, hi, device=dev)
    fi = torch.div(r, nxf, rounding_mode="floor")
    fj = r % nxf
    ci0, wi0, ci1, wi1 = interp_1d_pairs(fi, nxc)
    cj0, wj0, cj1, wj1 = interp_1d_pairs(fj, nxc)
    cols = []
    wts = []
    for ci, wi in ((ci0, wi0), (ci1, wi1)):
        for cj, wj in ((cj0, wj0), (cj1, wj1)):
            valid = (ci >= 0) & (cj >= 0)
            col = torch.where(valid, ci * nxc + cj, torch.full_like(ci, -1))
            cols.append(col)
            wts.append(torch.where(valid, wi * wj,
                                   torch.zeros_like(wi)))
    cols = torch.stack(cols, dim=1)  # (lr, 4)
    wts = torch.stack(wts, dim=1)
    # sort entries within each row by column for canonical CSR
    sort_key = torch.where(cols >= 0, cols,
                           torch.full_like(cols, 2 ** 40))
    order = torch.argsort(sort_key, dim=1)
    cols = torch.gather(cols, 1, order)
    wts = torch.gather(wts, 1, order)
    mask = cols >= 0
    indices = cols[mask]
    data = wts[mask]
    counts = mask.sum(dim=1)
    indptr = torch.zeros(hi - lo + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, dim=0, out=indptr[1:])
    return csr_array(None, shape=(nf, nxc * nxc),
                     _local=(indptr, indices, data.contiguous()))


class GMG(LinearOperator):
    """V-cycle preconditioner as a LinearOperator for cg(M=...)."""

    def __init__(self, A, k, n_levels=None, nu=2, omega=0.8,
                 coarse_iters=30):
        self.nu = nu
        self.omega = omega
        self.coarse_iters = coarse_iters
        self.levels = []  # (A, dinv)
        self.P = []
        nx = 2 ** k - 1
        if n_levels is None:
            n_levels = max(k - 3, 1)
        Al = A
        for l in range(n_levels):
            dinv = 1.0 / Al.diagonal()
            self.levels.append((Al, dinv))
            if l == n_levels - 1:
                break
            nxc = 2 ** (k - l - 1) - 1
            P = build_prolongation(nx, nxc)
            self.P.append(P)
            R = P.T
            Al = (R @ Al) @ P
            nx = nxc
        super().__init__(A.shape, dtype=A.dtype)

    def _smooth(self, A, dinv, x, b, rounds):
        from legate_sparse import ops
        y = torch.empty_like(x)
        for _ in range(rounds):
            A.dot(x, out=y)
            ops.jacobi_update(x, b, y, dinv, self.omega)
        return x

    def _vcycle(self, l, b):
        A, dinv = self.levels[l]
        x = torch.zeros_like(b)
        if l == len(self.levels) - 1:
            return self._smooth(A, dinv, x, b, self.coarse_iters)
        x = self._smooth(A, dinv, x, b, self.nu)
        r = b - A @ x
        P = self.P[l]
        rc = self._restrict(l, r)
        ec = self._vcycle(l + 1, rc)
        x = x + (P @ ec)
        x = self._smooth(A, dinv, x, b, self.nu)
        return x

    def _restrict(self, l, r):
        # R = P^T cached per level
        if not hasattr(self, "_Rt"):
            self._Rt = {}
        if l not in self._Rt:
            self._Rt[l] = self.P[l].T
        return self._Rt[l] @ r

    def matvec(self, x, out=None):
        y = self._vcycle(0, x)
        if out is not None:
            out.copy_(y)
            return out
        return y


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-k", type=int, default=8, help="grid edge = 2^k - 1")
    ap.add_argument("--rtol", type=float, default=1e-10)
    ap.add_argument("--maxiter", type=int, default=200)
    ap.add_argument("--nu", type=int, default=2)
    args = ap.parse_args()

    nx = 2 ** args.k - 1
    from common import d2_mat_dirichlet_2d
    with Timer() as t_build:
        A = d2_mat_dirichlet_2d(nx, nx)
        M = GMG(A, args.k, nu=args.nu)
    n = A.shape[0]
    lo, hi = runtime.local_range(n)
    print0(f"build+hierarchy: {t_build.elapsed*1e3:.1f} ms  "
           f"(n={n}, levels={len(M.levels)}, ranks={runtime.world_size})")
    for l, (Al, _) in enumerate(M.levels):
        print0(f"  level {l}: n={Al.shape[0]}, nnz={Al.nnz}")

    b = torch.ones(hi - lo, dtype=torch.float64, device=runtime.device)
    iters = []
    with Timer() as t:
        x, info = lsp.linalg.cg(A, b, rtol=args.rtol, maxiter=args.maxiter,
                                M=M, conv_test_iters=5,
                                callback=lambda _: iters.append(1))
    r = b - A @ x
    num = torch.linalg.norm(r) ** 2
    den = torch.linalg.norm(b) ** 2
    if runtime.world_size > 1:
        from legate_sparse import comm
        comm.allreduce_(num)
        comm.allreduce_(den)
    rel = float(torch.sqrt(num / den))
    n_it = len(iters)
    print0(f"solve: {t.elapsed*1e3:.1f} ms in {n_it} iters "
           f"({t.elapsed/max(n_it,1)*1e3:.2f} ms/iter), info={info}, "
           f"relative residual {rel:.3e}")
    assert rel <= max(args.rtol * 100, 1e-8), "residual check FAILED"
    print0("PASS")


if __name__ == "__main__":
    main()
