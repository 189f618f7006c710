# SPDX-License-Identifier: Apache-2.0
"""Matrix construction helpers: ``diags`` and ``eye``.

Counterpart of the reference's ``legate_sparse/gallery.py`` (diags,
gallery.py:77-195), scipy-compatible signature.
"""
from __future__ import annotations

import numpy as np
import torch

from .dia import dia_array
from .types import to_torch_dtype


def diags(diagonals, offsets=0, shape=None, format=None, dtype=None):
    """scipy.sparse.diags-compatible (reference gallery.py:77-195).

    ``diagonals``: scalar / 1-D / list of scalars / list of 1-D arrays.
    ``format``: "dia" (default) or "csr".
    """
    if format not in (None, "dia", "csr"):
        raise NotImplementedError(f"format {format} not supported")
    # normalize: list of (array, offset)
    if np.isscalar(diagonals) or (
            hasattr(diagonals, "ndim") and getattr(diagonals, "ndim") == 1
            and not isinstance(diagonals, (list, tuple))):
        diagonals = [np.atleast_1d(np.asarray(diagonals))]
        offsets = np.atleast_1d(np.asarray(offsets))
    else:
        diagonals = [np.atleast_1d(np.asarray(d)) for d in diagonals]
        offsets = np.atleast_1d(np.asarray(offsets))
    if len(diagonals) != len(offsets):
        raise ValueError("number of diagonals does not match offsets")

    if dtype is None:
        dtype = np.promote_types(
            np.result_type(*[d.dtype for d in diagonals]), np.float32)
        if dtype == np.float16:
            dtype = np.float32
    tdt = to_torch_dtype(dtype)

    # infer shape (scipy rule: square matrix fitting all diagonals)
    if shape is None:
        M = max(len(d) + abs(int(o)) for d, o in zip(diagonals, offsets))
        shape = (M, M)
    M, N = int(shape[0]), int(shape[1])

    n_diags = len(offsets)
    data = torch.zeros(n_diags, N, dtype=tdt)
    for k, (diag, off) in enumerate(zip(diagonals, offsets)):
        off = int(off)
        start = max(0, off)
        end = min(N, M + off)
        length = end - start
        if length < 0:
            raise ValueError(f"offset {off} out of bounds for shape {shape}")
        d = torch.from_numpy(np.ascontiguousarray(diag)).to(tdt).reshape(-1)
        if d.numel() == 1:
            data[k, start:end] = d.item() if not d.is_complex() else d[0]
        else:
            if d.numel() < length:
                raise ValueError(
                    f"diagonal {k} (offset {off}) too short: "
                    f"{d.numel()} < {length}")
            data[k, start:end] = d[:length]
    out = dia_array((data, torch.from_numpy(
        np.ascontiguousarray(offsets, dtype=np.int64))), shape=(M, N))
    if format == "csr":
        return out.tocsr()
    return out


def eye(m, n=None, k=0, dtype=np.float64, format=None):
    """scipy.sparse.eye-compatible identity/shifted-diagonal builder."""
    if n is None:
        n = m
    m, n = int(m), int(n)
    length = max(0, min(m + min(k, 0), n - max(k, 0)))
    return diags([np.ones(length, dtype=dtype)], [k], shape=(m, n),
                 format=format or "dia", dtype=dtype)


def identity(n, dtype=np.float64, format=None):
    return eye(n, dtype=dtype, format=format)


def poisson_2d(nx: int, ny: int, dtype=np.float64):
    """5-point 2-D Poisson (Dirichlet) operator on an nx*ny grid, built
    DIRECTLY as the distributed CSR (each rank constructs only its local
    row block — no replicated intermediate).  This is the benchmark
    operator of BASELINE.md (reference examples/pde.py builds it via
    diags; diags works too, this is the scalable path)."""
    import torch
    from .csr import csr_array
    from .runtime import runtime

    tdt = to_torch_dtype(dtype)
    M = nx * ny
    lo, hi = runtime.local_range(M)
    dev = runtime.device
    r = torch.arange(lo, hi, device=dev, dtype=torch.int64)
    j = r % nx
    lr = hi - lo
    cols = torch.stack([r - nx, r - 1, r, r + 1, r + nx], dim=1)
    mask = torch.stack([
        r >= nx, j > 0, torch.ones(lr, dtype=torch.bool, device=dev),
        j < nx - 1, r < M - nx,
    ], dim=1)
    vals = torch.tensor([-1.0, -1.0, 4.0, -1.0, -1.0], dtype=tdt,
                        device=dev).reshape(1, 5).expand(lr, 5)
    indices = cols[mask]
    data = vals[mask].contiguous()
    counts = mask.sum(dim=1)
    indptr = torch.zeros(lr + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, dim=0, out=indptr[1:])
    return csr_array(None, shape=(M, M),
                     _local=(indptr, indices, data))


def rmat(scale: int, edge_factor: int = 16, seed: int = 0,
         dtype=np.float64):
    """Synthetic R-MAT (Graph500-style) adjacency matrix, built
    DISTRIBUTED: each rank generates its share of edges on-device, routes
    them to row owners via all-to-allv, and assembles local CSR rows.
    Duplicate edges are summed (scipy COO semantics).  This is the
    BASELINE.md SpGEMM stress operator (R-MAT scale-24)."""
    import torch
    from . import comm
    from .csr import csr_array
    from .runtime import runtime

    N = 1 << scale
    ws = runtime.world_size
    rank = runtime.rank
    dev = runtime.device
    tdt = to_torch_dtype(dtype)
    E_total = N * edge_factor
    E = E_total // ws + (1 if rank < E_total % ws else 0)
    gen = torch.Generator(device=dev)
    gen.manual_seed(seed * 7919 + rank)
    src = torch.zeros(E, dtype=torch.int64, device=dev)
    dst = torch.zeros(E, dtype=torch.int64, device=dev)
    # quadrant probabilities (a, b, c, d) = (0.57, 0.19, 0.19, 0.05)
    for i in range(scale):
        r = torch.rand(E, generator=gen, device=dev)
        src_bit = r >= 0.76          # quadrants c + d
        dst_bit = ((r >= 0.57) & (r < 0.76)) | (r >= 0.95)  # b + d
        src |= src_bit.to(torch.int64) << i
        dst |= dst_bit.to(torch.int64) << i
    val = torch.ones(E, dtype=tdt, device=dev)
    part = runtime.partition(N)
    if ws > 1:
        owner = torch.clamp(
            torch.div(src, max(part.chunk, 1), rounding_mode="floor"),
            max=ws - 1)
        order = torch.argsort(owner)
        src, dst, val = src[order], dst[order], val[order]
        counts = torch.bincount(owner.cpu().long(), minlength=ws).tolist()
        src = torch.cat(comm.alltoallv(list(torch.split(src, counts))))
        dst = torch.cat(comm.alltoallv(list(torch.split(dst, counts))))
        val = torch.cat(comm.alltoallv(list(torch.split(val, counts))))
    lo, hi = part.lo(rank), part.hi(rank)
    from .csr import _assemble_local_rows
    indptr, dst, val = _assemble_local_rows(src - lo, dst, val, hi - lo, N,
                                            dedup=True)
    return csr_array(None, shape=(N, N),
                     _local=(indptr, dst, val.contiguous()))
