# SPDX-License-Identifier: Apache-2.0
"""Iterative solvers: CG, GMRES, LinearOperator.

Counterpart of the reference's ``legate_sparse/linalg.py`` (linalg.py:85-668).
The key property preserved is the *async solver pipeline* (SURVEY §3.4):

- ``cg_axpby`` consumes its scalars as 1-element DEVICE tensors; the
  division a/b happens inside the fused HIP kernel (reference axpby.cu:25-47)
  so no ``.item()`` ever occurs in the iteration body.
- Scalar reductions (rho, pq) are local block-reduce + RCCL all-reduce of a
  single element, left on device.
- Convergence is tested only every ``conv_test_iters`` iterations
  (reference linalg.py:529-533) — the only host syncs in the loop.

All vectors are LOCAL shards of partition(n) in SPMD mode.
"""
from __future__ import annotations

from typing import Callable, Optional

import numpy as np
import torch

from . import comm, ops
from .base import as_torch_1d
from .runtime import runtime
from .types import to_torch_dtype


def _to_local_vec(v, n_global: int, dtype, device) -> torch.Tensor:
    """Accept a global (replicated) or local-shard vector; return the
    local shard."""
    t = as_torch_1d(v, device=device).to(to_torch_dtype(dtype))
    part = runtime.partition(n_global)
    lo, hi = part.lo(runtime.rank), part.hi(runtime.rank)
    if t.numel() == n_global and n_global != (hi - lo):
        return t[lo:hi].contiguous()
    if t.numel() == (hi - lo):
        return t.contiguous()
    if t.numel() == n_global:
        return t.contiguous()
    raise ValueError(
        f"vector length {t.numel()} matches neither global {n_global} nor "
        f"local shard {hi - lo}")


def _gdot(x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """Global <x, y> as a 1-element device tensor."""
    s = ops.vdot(x, y, conj=True)
    if runtime.world_size > 1:
        comm.allreduce_(s)
    return s


def _gnorm(x: torch.Tensor) -> torch.Tensor:
    s = _gdot(x, x)
    return torch.sqrt(torch.abs(s))


# ---------------------------------------------------------------------------
# LinearOperator family (reference linalg.py:85-414)
# ---------------------------------------------------------------------------
class LinearOperator:
    def __init__(self, shape, matvec=None, rmatvec=None, dtype=None):
        if matvec is not None:
            # scipy-style factory usage
            self.shape = tuple(int(s) for s in shape)
            self.dtype = np.dtype(dtype) if dtype is not None else None
            self._matvec_fn = matvec
            self._rmatvec_fn = rmatvec
        else:
            self.shape = tuple(int(s) for s in shape)
            self.dtype = np.dtype(dtype) if dtype is not None else None
            self._matvec_fn = None
            self._rmatvec_fn = None

    def matvec(self, x, out=None):
        if self._matvec_fn is None:
            raise NotImplementedError
        y = self._matvec_fn(x)
        if out is not None:
            out.copy_(y if isinstance(y, torch.Tensor) else torch.as_tensor(y))
            return out
        return y

    def rmatvec(self, x, out=None):
        if self._rmatvec_fn is None:
            raise NotImplementedError
        y = self._rmatvec_fn(x)
        if out is not None:
            out.copy_(y if isinstance(y, torch.Tensor) else torch.as_tensor(y))
            return out
        return y

    def __matmul__(self, x):
        return self.matvec(x)

    def dot(self, x):
        return self.matvec(x)


class IdentityOperator(LinearOperator):
    def __init__(self, shape, dtype=None):
        super().__init__(shape, dtype=dtype)

    def matvec(self, x, out=None):
        if out is not None:
            out.copy_(x)
            return out
        return x.clone() if isinstance(x, torch.Tensor) else x

    rmatvec = matvec


class _SparseMatrixLinearOperator(LinearOperator):
    """Wraps a csr_array; caches A.conj().T for rmatvec
    (reference linalg.py:375-387)."""

    def __init__(self, A):
        self.A = A
        self._AH = None
        super().__init__(A.shape, dtype=A.dtype)

    def matvec(self, x, out=None):
        return self.A.dot(x, out=out)

    def rmatvec(self, x, out=None):
        if self._AH is None:
            self._AH = self.A.conj().transpose()
        return self._AH.dot(x, out=out)


def aslinearoperator(A):
    if isinstance(A, LinearOperator):
        return A
    if hasattr(A, "format"):
        return _SparseMatrixLinearOperator(A if A.format == "csr"
                                           else A.tocsr())
    raise TypeError(f"cannot wrap {type(A)} as LinearOperator")


# reference linalg.py:417-431 exposes make_linear_operator; scipy calls
# the same thing aslinearoperator — provide both names.
make_linear_operator = aslinearoperator


def norm(A, ord="fro"):
    """Matrix norm of a sparse array: 'fro', 1 (max column abs sum), or
    inf (max row abs sum) — scipy.sparse.linalg.norm-compatible subset."""
    import math as _math

    if not hasattr(A, "indptr"):
        raise TypeError("norm expects a sparse array")
    data = A.data
    absd = data.abs()
    if ord in ("fro", None):
        s = (absd * absd).sum().reshape(1)
        if runtime.world_size > 1:
            comm.allreduce_(s)
        return float(torch.sqrt(s))
    if ord == 1:
        col = torch.zeros(A.shape[1], dtype=absd.dtype, device=absd.device)
        col.scatter_add_(0, A.indices.long(), absd)
        if runtime.world_size > 1:
            comm.allreduce_(col)
        return float(col.max()) if col.numel() else 0.0
    if ord in (np.inf, float("inf"), "inf"):
        row = torch.zeros(A.indptr.numel() - 1, dtype=absd.dtype,
                          device=absd.device)
        ids = torch.repeat_interleave(
            torch.arange(row.numel(), device=absd.device),
            A.indptr[1:] - A.indptr[:-1])
        row.scatter_add_(0, ids, absd)
        m = row.max().reshape(1) if row.numel() else torch.zeros(
            1, dtype=absd.dtype, device=absd.device)
        if runtime.world_size > 1:
            comm.allreduce_(m, op="max")
        return float(m)
    raise NotImplementedError(f"norm ord={ord!r} not supported")


# ---------------------------------------------------------------------------
# Fused CG update (reference linalg.py:433-451 + axpby.cu)
# ---------------------------------------------------------------------------
def cg_axpby(y: torch.Tensor, x: torch.Tensor, a: torch.Tensor,
             b: torch.Tensor, isalpha: bool, negate: bool = False):
    """y = (±a/b)·x + y  (isalpha) or y = x + (±a/b)·y  — a, b are
    1-element device tensors consumed inside the kernel."""
    return ops.axpby(y, x, a, b, isalpha, negate)


# ---------------------------------------------------------------------------
# CG (reference linalg.py:465-535, CuPy-derived)
# ---------------------------------------------------------------------------
from .coverage import track_provenance


@track_provenance
def cg(A, b, x0=None, *, rtol: float = 1e-5, atol: float = 0.0,
       maxiter: Optional[int] = None, M=None,
       callback: Optional[Callable] = None, conv_test_iters: int = 25):
    """Conjugate gradient. Returns (x, info): info=0 converged, >0 not."""
    Aop = aslinearoperator(A)
    n = Aop.shape[0]
    dtype = Aop.dtype if Aop.dtype is not None else np.float64
    device = runtime.device
    b = _to_local_vec(b, n, dtype, device)
    if maxiter is None:
        maxiter = n * 10
    Mop = aslinearoperator(M) if M is not None and not isinstance(
        M, LinearOperator) else (M or IdentityOperator(Aop.shape,
                                                       dtype=dtype))

    bnrm2 = _gnorm(b)
    # b = 0: the exact solution is x = 0; entering the loop would divide
    # 0/0 in the fused axpby (scipy returns immediately too).  One host
    # sync before the loop, not inside it.
    if float(bnrm2.item()) == 0.0:
        return torch.zeros_like(b), 0
    atol_t = torch.clamp(bnrm2 * rtol, min=atol)

    if x0 is None:
        x = torch.zeros_like(b)
        r = b.clone()
    else:
        x = _to_local_vec(x0, n, dtype, device).clone()
        r = b - Aop.matvec(x)

    # Unpreconditioned CG aliases z = r (saves two full vector passes
    # per iteration vs copying through an identity preconditioner).
    ident_M = isinstance(Mop, IdentityOperator)
    if ident_M:
        z = r
    else:
        z = Mop.matvec(r)
        if not isinstance(z, torch.Tensor):
            z = torch.as_tensor(z, device=device)
        if z.data_ptr() == r.data_ptr():
            z = z.clone()
    p = z.clone()
    q = torch.empty_like(b)
    rho = _gdot(r, z)
    iters = 0
    info = maxiter

    # ---- hipGraph-captured inner loop --------------------------------
    # The unpreconditioned single-GPU iteration is a fixed 6-kernel
    # sequence on stable buffers; capturing it in a hipGraph removes the
    # per-iteration launch/Python overhead (goal: graphs for launch-bound
    # inner loops).  Distributed / preconditioned / callback runs use the
    # eager loop below.
    import os as _os
    graph = None
    # hipGraph replay has a ~10-16 us floor: a measured win below ~2M
    # local rows (1.7x at 262K rows), ~2% loss at 16.7M — gate by size
    # (benchmarks/cg_graph_ab.py)
    _graph_env = _os.environ.get("LS_CG_GRAPH", "auto")
    _graph_on = (_graph_env == "1"
                 or (_graph_env == "auto" and b.numel() <= 2 ** 21))
    if (ident_M and callback is None and runtime.world_size == 1
            and device.type == "cuda" and maxiter > 8 and _graph_on):
        try:
            rho_buf = rho  # 1-element device tensors, stable addresses
            pq_buf = torch.zeros_like(rho_buf)
            rho1_buf = torch.zeros_like(rho_buf)

            def _one_iter():
                Aop.matvec(p, out=q)
                ops.vdot(p, q, out=pq_buf)
                cg_axpby(x, p, rho_buf, pq_buf, isalpha=True, negate=False)
                cg_axpby(r, q, rho_buf, pq_buf, isalpha=True, negate=True)
                rho1_buf.copy_(rho_buf)
                ops.vdot(r, z, out=rho_buf)
                cg_axpby(p, z, rho_buf, rho1_buf, isalpha=False,
                         negate=False)

            # warmup replays are REAL iterations (side-stream per torch
            # graph-capture protocol)
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    _one_iter()
                    iters += 1
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                _one_iter()  # recorded, NOT executed (stream capture)
        except Exception:
            graph = None  # fall back to the eager loop

    if graph is not None:
        step = conv_test_iters if conv_test_iters > 0 else maxiter
        while iters < maxiter:
            n_rep = min(step - (iters % step) if iters % step else step,
                        maxiter - iters)
            for _ in range(n_rep):
                graph.replay()
            iters += n_rep
            rnorm = _gnorm(r)
            if bool((rnorm <= atol_t).item()):
                info = 0
                break
        x = x  # state updated in place by replays
        if info != 0:
            rnorm = _gnorm(r)
            if bool((rnorm <= atol_t).item()):
                info = 0
        return x, info

    while iters < maxiter:
        Aop.matvec(p, out=q)
        pq = _gdot(p, q)
        # alpha = rho / pq ; x += alpha p ; r -= alpha q — all in-kernel
        cg_axpby(x, p, rho, pq, isalpha=True, negate=False)
        cg_axpby(r, q, rho, pq, isalpha=True, negate=True)
        iters += 1
        if callback is not None:
            callback(x)
        if conv_test_iters > 0 and (iters % conv_test_iters == 0
                                    or iters == maxiter):
            rnorm = _gnorm(r)  # host sync point (deliberate, amortized)
            if bool((rnorm <= atol_t).item()):
                info = 0
                break
        if not ident_M:
            z = Mop.matvec(r, out=z if isinstance(z, torch.Tensor)
                           else None)
            if not isinstance(z, torch.Tensor):
                z = torch.as_tensor(z, device=device)
        rho1 = rho
        rho = _gdot(r, z)
        # beta = rho / rho1 ; p = z + beta p
        cg_axpby(p, z, rho, rho1, isalpha=False, negate=False)

    if info != 0:
        rnorm = _gnorm(r)
        if bool((rnorm <= atol_t).item()):
            info = 0
    return x, info


# ---------------------------------------------------------------------------
# GMRES (reference linalg.py:540-668, restarted, host lstsq)
# ---------------------------------------------------------------------------
@track_provenance
def gmres(A, b, x0=None, *, rtol: float = 1e-5, atol: float = 0.0,
          restart: Optional[int] = None, maxiter: Optional[int] = None,
          M=None, callback: Optional[Callable] = None):
    """Restarted GMRES.  Arnoldi basis vectors are local shards; the small
    Hessenberg solve runs on host numpy (reference linalg.py:540-668)."""
    Aop = aslinearoperator(A)
    n = Aop.shape[0]
    dtype = Aop.dtype if Aop.dtype is not None else np.float64
    device = runtime.device
    b = _to_local_vec(b, n, dtype, device)
    if maxiter is None:
        maxiter = min(n, 1000)
    if restart is None:
        restart = min(20, n)
    restart = min(restart, maxiter)
    Mop = M if isinstance(M, LinearOperator) else (
        aslinearoperator(M) if M is not None
        else IdentityOperator(Aop.shape, dtype=dtype))

    bnrm2 = float(_gnorm(b).item())
    tol = max(rtol * bnrm2, atol)
    if bnrm2 == 0.0:
        return b.clone(), 0

    if x0 is None:
        x = torch.zeros_like(b)
    else:
        x = _to_local_vec(x0, n, dtype, device).clone()

    cdtype = to_torch_dtype(dtype)
    iters = 0
    info = 1
    while iters < maxiter:
        r = b - Aop.matvec(x)
        r = Mop.matvec(r)
        if not isinstance(r, torch.Tensor):
            r = torch.as_tensor(r, device=device)
        beta = float(_gnorm(r).item())
        if beta <= tol:
            info = 0
            break
        m = min(restart, maxiter - iters)
        V = torch.zeros(m + 1, r.numel(), dtype=cdtype, device=device)
        H = np.zeros((m + 1, m),
                     dtype=np.complex128 if np.dtype(dtype).kind == "c"
                     else np.float64)
        V[0] = r / beta
        j_done = m
        for j in range(m):
            w = Aop.matvec(V[j])
            w = Mop.matvec(w)
            if not isinstance(w, torch.Tensor):
                w = torch.as_tensor(w, device=device)
            # Gram-Schmidt (batched dot per column, allreduced)
            basis = V[:j + 1].conj() if w.is_complex() else V[:j + 1]
            hcol = basis @ w.reshape(-1, 1)
            if runtime.world_size > 1:
                comm.allreduce_(hcol)
            w = w - (hcol.reshape(1, -1) @ V[:j + 1]).reshape(-1)
            hnorm = float(_gnorm(w).item())
            H[: j + 1, j] = hcol.reshape(-1).cpu().numpy()
            H[j + 1, j] = hnorm
            iters += 1
            if hnorm <= 1e-14 * max(beta, 1.0):
                j_done = j + 1
                break
            V[j + 1] = w / hnorm
            j_done = j + 1
        e1 = np.zeros(j_done + 1, dtype=H.dtype)
        e1[0] = beta
        ym, *_ = np.linalg.lstsq(H[: j_done + 1, : j_done], e1, rcond=None)
        yt = torch.from_numpy(np.ascontiguousarray(ym)).to(
            device=device, dtype=cdtype)
        x = x + (yt.reshape(1, -1) @ V[:j_done]).reshape(-1)
        if callback is not None:
            callback(x)
        res = float(_gnorm(b - Aop.matvec(x)).item())
        if res <= tol:
            info = 0
            break
    else:
        pass
    if info != 0:
        res = float(_gnorm(b - Aop.matvec(x)).item())
        info = 0 if res <= tol else maxiter
    return x, info
