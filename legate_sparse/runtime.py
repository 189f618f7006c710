# SPDX-License-Identifier: Apache-2.0
"""MI355X-native runtime: process/GPU topology, streams, row partitioning.

This replaces the reference's Legion/Legate runtime glue
(reference runtime.py:54-107): instead of a task runtime with a constraint
solver, we run SPMD — one process per GPU launched by ``torchrun`` — and the
"partition" is an explicit equal 1-D row decomposition (the same split
Legion's equal partition produces for the reference's align constraints,
csr.py:587).

Streams: a compute stream (torch default) plus a communication side stream so
RCCL halo exchange can overlap local SpMV work (reference gets this overlap
implicitly from Legion's dependence analysis; here it is explicit).
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist


def _ceil_div(a: int, b: int) -> int:
    return -(-a // b)


class Partition:
    """Equal 1-D block partition of ``n`` elements over ``world_size`` ranks.

    Rank ``r`` owns [lo(r), hi(r)).  Matches Legion's equal partition used by
    the reference for output rows (reference csr.py:587, base.py:66-90).
    """

    __slots__ = ("n", "world_size", "chunk")

    def __init__(self, n: int, world_size: int):
        self.n = int(n)
        self.world_size = int(world_size)
        self.chunk = _ceil_div(self.n, self.world_size) if self.n > 0 else 0

    def lo(self, rank: int) -> int:
        return min(rank * self.chunk, self.n)

    def hi(self, rank: int) -> int:
        return min((rank + 1) * self.chunk, self.n)

    def count(self, rank: int) -> int:
        return self.hi(rank) - self.lo(rank)

    def counts(self) -> list:
        return [self.count(r) for r in range(self.world_size)]

    def owner_of(self, idx: int) -> int:
        return min(idx // self.chunk, self.world_size - 1) if self.chunk else 0

    def __eq__(self, other):
        return (
            isinstance(other, Partition)
            and self.n == other.n
            and self.world_size == other.world_size
        )

    def __repr__(self):
        return f"Partition(n={self.n}, world={self.world_size})"


class Runtime:
    """Singleton holding process topology, device, and streams."""

    def __init__(self):
        self._initialized = False
        self._rank = 0
        self._world_size = 1
        self._device: Optional[torch.device] = None
        self._comm_stream: Optional[torch.cuda.Stream] = None
        self._comm_event: Optional[torch.cuda.Event] = None

    # -- initialization -------------------------------------------------
    def _lazy_init(self):
        if self._initialized:
            return
        self._initialized = True
        # torchrun sets RANK/WORLD_SIZE/LOCAL_RANK; single-process runs don't.
        env_world = int(os.environ.get("WORLD_SIZE", "1"))
        if env_world > 1 and not dist.is_initialized():
            backend = "nccl" if torch.cuda.is_available() else "gloo"
            dist.init_process_group(backend=backend)
        if dist.is_initialized():
            self._rank = dist.get_rank()
            self._world_size = dist.get_world_size()
        else:
            self._rank = 0
            self._world_size = 1
        if torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK", self._rank))
            local_rank %= max(torch.cuda.device_count(), 1)
            self._device = torch.device("cuda", local_rank)
            torch.cuda.set_device(self._device)
            self._comm_stream = torch.cuda.Stream(device=self._device)
            self._comm_event = torch.cuda.Event()
        else:
            self._device = torch.device("cpu")

    # -- topology -------------------------------------------------------
    @property
    def rank(self) -> int:
        self._lazy_init()
        return self._rank

    @property
    def world_size(self) -> int:
        self._lazy_init()
        return self._world_size

    @property
    def device(self) -> torch.device:
        self._lazy_init()
        return self._device

    @property
    def num_gpus(self) -> int:
        self._lazy_init()
        return self._world_size if self._device.type == "cuda" else 0

    @property
    def distributed(self) -> bool:
        return self.world_size > 1

    @property
    def comm_stream(self):
        self._lazy_init()
        return self._comm_stream

    @property
    def comm_event(self):
        self._lazy_init()
        return self._comm_event

    # -- partitioning ---------------------------------------------------
    def partition(self, n: int) -> Partition:
        return Partition(n, self.world_size)

    def local_range(self, n: int):
        p = self.partition(n)
        return p.lo(self.rank), p.hi(self.rank)

    # -- sync helpers ---------------------------------------------------
    def barrier(self):
        self._lazy_init()
        if dist.is_initialized():
            dist.barrier()

    def synchronize(self):
        self._lazy_init()
        if self._device.type == "cuda":
            torch.cuda.synchronize(self._device)


runtime = Runtime()
