"""In-process multi-rank simulation: thread-per-rank with loopback
collectives.

Lets the REAL sharded code paths (ops/sharded.py, ops/table.py sharded
branch) run at world sizes > 1 on a single device - W threads each drive
one shard's selector; the collectives rendezvous on a threading.Barrier
and combine tensors in rank order (deterministic, matching what a real
all-reduce delivers identically to every rank).

This is a testing/validation harness (tests/test_loopback.py): it
exercises exactly the multi-GPU RCCL call sites without needing multiple
GPUs.
"""
from __future__ import annotations

import threading
from typing import Callable, List

import torch

from .shard import Comm


class _Shared:
    def __init__(self, world: int):
        self.barrier = threading.Barrier(world)
        self.slots: List = [None] * world
        self.result: List = [None] * world


class LoopbackComm(Comm):
    def __init__(self, rank: int, world: int, shared: _Shared, device=None):
        super().__init__(rank=rank, world=world, device=device)
        self._sh = shared

    def _exchange(self, t: torch.Tensor, combine: Callable):
        sh = self._sh
        sh.slots[self.rank] = t
        sh.barrier.wait()
        if self.rank == 0:
            out = combine(sh.slots)
            for r in range(self.world):
                sh.result[r] = out
        sh.barrier.wait()
        res = sh.result[self.rank]
        sh.barrier.wait()  # all ranks read before slots are reused
        return res

    def all_reduce_(self, t: torch.Tensor) -> torch.Tensor:
        total = self._exchange(
            t, lambda slots: torch.stack([s.float() for s in slots]).sum(0))
        t.copy_(total.to(t.dtype))
        return t

    def broadcast_(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        val = self._exchange(t, lambda slots: slots[src].clone())
        t.copy_(val)
        return t

    def all_gather_cat(self, t: torch.Tensor, dim: int = 0,
                       sizes=None) -> torch.Tensor:
        return self._exchange(
            t, lambda slots: torch.cat([s.clone() for s in slots],
                                       dim=dim)).clone()

    def barrier(self):
        self._sh.barrier.wait()


def run_ranks(world: int, fn: Callable[[LoopbackComm], object],
              device=None):
    """Run fn(comm) on `world` threads; returns the per-rank results.

    Exceptions in any rank are re-raised (and the barrier broken so the
    other ranks do not deadlock).
    """
    shared = _Shared(world)
    results = [None] * world
    errors = [None] * world

    def worker(rank):
        comm = LoopbackComm(rank, world, shared, device=device)
        try:
            results[rank] = fn(comm)
        except BaseException as e:  # noqa: BLE001 - propagate to caller
            errors[rank] = e
            shared.barrier.abort()

    threads = [threading.Thread(target=worker, args=(r,))
               for r in range(world)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    for e in errors:
        if e is not None:
            raise e
    return results
