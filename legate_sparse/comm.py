# SPDX-License-Identifier: Apache-2.0
"""Explicit collectives replacing the reference's implicit Legion images.

The reference expresses all SpMV/SpGEMM data movement declaratively via
Legion image partitions (reference csr.py:588-591, 656-666) plus exactly one
NCCL allgather of per-rank nnz (spgemm_csr_csr_csr.cu:43-62).  Here each of
those patterns is an explicit RCCL collective over xGMI
(``torch.distributed`` backend "nccl" == RCCL on ROCm), with a gloo path so
multi-process CPU tests run without GPUs:

- ``allgatherv``      — SpMV x-vector halo (the MIN_MAX image of csr.py:591,
                        widened to the full vector in v1).
- ``alltoallv``       — SpGEMM B-row redistribution / transpose row shuffle.
- ``allreduce_``      — CG/GMRES scalar reductions, kept on-device as
                        1-element tensors (no host sync in the solver loop).
- ``allgather_scalars`` — per-rank nnz counts for global offsets.
"""
from __future__ import annotations

from typing import List, Sequence

import torch
import torch.distributed as dist

from .runtime import runtime


def _ws() -> int:
    return runtime.world_size


def allgatherv(local: torch.Tensor, counts: Sequence[int]) -> torch.Tensor:
    """Gather variable-size 1-D shards from every rank; returns concat.

    ``counts[r]`` is rank r's shard length (deterministic from the
    Partition, so no size exchange is needed).
    """
    if _ws() == 1:
        return local
    counts = list(counts)
    total = sum(counts)
    mx = max(counts)
    equal = all(c == mx for c in counts)
    use_into = dist.get_backend() != "gloo"
    # ncclAllGather needs equal counts: pad to the max (only the tail rank
    # differs under equal partition), gather, then pack.
    if equal and local.numel() == mx:
        send = local.contiguous()
    else:
        send = torch.zeros(mx, dtype=local.dtype, device=local.device)
        send[: local.numel()] = local
    if use_into:
        recv = torch.empty(mx * _ws(), dtype=local.dtype, device=local.device)
        dist.all_gather_into_tensor(recv, send)
    else:
        shards = [torch.empty(mx, dtype=local.dtype, device=local.device)
                  for _ in range(_ws())]
        dist.all_gather(shards, send)
        recv = torch.cat(shards)
    if equal:
        return recv[:total]
    out = torch.empty(total, dtype=local.dtype, device=local.device)
    off = 0
    for r, c in enumerate(counts):
        out[off:off + c] = recv[r * mx: r * mx + c]
        off += c
    return out


def allgather_scalars(value: int, device=None) -> torch.Tensor:
    """All-gather one int64 per rank (reference spgemm cu:43-62)."""
    if _ws() == 1:
        return torch.tensor([value], dtype=torch.int64)
    dev = device if device is not None else runtime.device
    t = torch.tensor([int(value)], dtype=torch.int64, device=dev)
    out = [torch.empty_like(t) for _ in range(_ws())]
    dist.all_gather(out, t)
    return torch.cat(out).cpu()


def allreduce_(t: torch.Tensor, op: str = "sum") -> torch.Tensor:
    """In-place all-reduce; returns ``t``.  Used for solver scalars —
    ``t`` stays a device tensor, no ``.item()`` (reference linalg.py keeps
    rho/pq as futures; we keep them as 1-element device tensors)."""
    if _ws() > 1:
        red = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX,
               "min": dist.ReduceOp.MIN}[op]
        dist.all_reduce(t, op=red)
    return t


def alltoallv(send_chunks: List[torch.Tensor]) -> List[torch.Tensor]:
    """Each rank sends ``send_chunks[r]`` (1-D, same dtype) to rank r and
    receives a list of chunks, one from each rank.

    nccl/RCCL: size exchange + ``all_to_all_single``.
    gloo (CPU tests): emulated with one all_gather per rank's concat —
    correct, not fast; tests only.
    """
    ws = _ws()
    if ws == 1:
        return [send_chunks[0]]
    dtype = send_chunks[0].dtype
    device = send_chunks[0].device
    send_counts = torch.tensor([c.numel() for c in send_chunks],
                               dtype=torch.int64, device=device)
    recv_counts = torch.empty_like(send_counts)
    dist.all_to_all_single(recv_counts, send_counts)
    backend_ok = dist.get_backend() != "gloo"
    if backend_ok:
        send_buf = torch.cat([c.contiguous() for c in send_chunks])
        recv_buf = torch.empty(int(recv_counts.sum().item()), dtype=dtype,
                               device=device)
        dist.all_to_all_single(
            recv_buf, send_buf,
            output_split_sizes=[int(x) for x in recv_counts],
            input_split_sizes=[int(x) for x in send_counts],
        )
        return list(torch.split(recv_buf, [int(x) for x in recv_counts]))
    # gloo emulation: every rank broadcasts its full concat; receivers slice.
    sizes = [torch.zeros(ws, dtype=torch.int64) for _ in range(ws)]
    dist.all_gather(sizes, send_counts.cpu())
    out = []
    me = runtime.rank
    for r in range(ws):
        concat_r = torch.empty(int(sizes[r].sum().item()), dtype=dtype)
        if r == me:
            concat_r = torch.cat([c.contiguous() for c in send_chunks])
        dist.broadcast(concat_r, src=r)
        offs = [0]
        for s in sizes[r].tolist():
            offs.append(offs[-1] + int(s))
        out.append(concat_r[offs[me]:offs[me + 1]].clone())
    return out


def alltoallv_single_async(send_buf: torch.Tensor, in_splits, out_splits):
    """Known-size all-to-all on a packed buffer; returns (recv, work).
    nccl: async (work.wait() inserts the stream dependency) — lets the
    caller overlap local compute with the exchange.  gloo: synchronous
    emulation (returns work=None)."""
    ws = _ws()
    if ws == 1:
        return send_buf, None
    device = send_buf.device
    recv = torch.empty(int(sum(out_splits)), dtype=send_buf.dtype,
                       device=device)
    if dist.get_backend() != "gloo":
        work = dist.all_to_all_single(
            recv, send_buf,
            output_split_sizes=[int(x) for x in out_splits],
            input_split_sizes=[int(x) for x in in_splits],
            async_op=True)
        return recv, work
    # gloo: emulate with per-rank broadcast of each rank's packed buffer
    sizes_all = [torch.zeros(ws, dtype=torch.int64) for _ in range(ws)]
    mine = torch.tensor([int(x) for x in in_splits], dtype=torch.int64)
    dist.all_gather(sizes_all, mine)
    me = runtime.rank
    off_out = 0
    for r in range(ws):
        buf = torch.empty(int(sizes_all[r].sum()), dtype=send_buf.dtype)
        if r == me:
            buf.copy_(send_buf)
        dist.broadcast(buf, src=r)
        offs = [0]
        for v in sizes_all[r].tolist():
            offs.append(offs[-1] + int(v))
        piece = buf[offs[me]:offs[me + 1]]
        recv[off_out:off_out + piece.numel()] = piece
        off_out += piece.numel()
    return recv, None


def exclusive_scan_offsets(per_rank: torch.Tensor) -> torch.Tensor:
    """Exclusive prefix sum of an all-gathered per-rank count vector
    (the thrust::exclusive_scan of reference spgemm cu:53-54)."""
    out = torch.zeros_like(per_rank)
    if per_rank.numel() > 1:
        out[1:] = torch.cumsum(per_rank[:-1], dim=0)
    return out
