"""Model-axis sharding over torch.distributed (RCCL on ROCm, gloo on CPU).

The reference is single-device (main.py:110) with no collectives anywhere;
this module is the new first-class component that shards the model axis H
across the 8 GPUs of one MI355X node. Sharding is
strided - rank r owns models {h : h % world == r} - so slices stay balanced
for any H.

Collective sites (all small-to-medium, latency-critical in the EIG loop):
  - consensus sum over H            (N, C)   once at init
  - pi_hat sum over H               (N, C)   once per label
  - pbest log-cdf sum over H        (R, P)   inside the per-step EIG loop
  - pbest normalizer                (R,)     inside the per-step EIG loop
  - mixture entropy partials        (B, C)   per chunk
  - per-model scalars (losses, pbest marginals) via all_gather

xGMI is point-to-point (7 links per GPU), so these small frequent
all-reduces ride RCCL's latency-optimized algorithms; bucket tuning is not
needed because every message is a single contiguous tensor.
"""
from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


class Comm:
    """Thin communicator: no-op when world_size == 1."""

    def __init__(self, rank: int = 0, world: int = 1, device=None):
        self.rank = rank
        self.world = world
        self.device = device

    @property
    def is_distributed(self) -> bool:
        return self.world > 1

    def all_reduce_(self, t: torch.Tensor) -> torch.Tensor:
        if self.world > 1:
            dist.all_reduce(t)
        return t

    def broadcast_(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.world > 1:
            dist.broadcast(t, src=src)
        return t

    def shard_sizes(self, total: int):
        """Per-rank shard sizes under strided sharding of `total` items."""
        return [len(range(r, total, self.world))
                for r in range(self.world)]

    def all_gather_cat(self, t: torch.Tensor, dim: int = 0,
                       sizes=None) -> torch.Tensor:
        """Gather per-rank shards and concatenate along `dim`.

        Shards may have unequal sizes along `dim` (strided H sharding when
        H % world != 0). Pass `sizes` (per-rank sizes along `dim`, e.g.
        from shard_sizes) to skip the size-exchange collective - one
        round-trip saved per call, which matters for the per-chunk
        gathers inside the EIG loop.
        """
        if self.world == 1:
            return t
        if sizes is None:
            sz = [torch.zeros(1, dtype=torch.long, device=t.device)
                  for _ in range(self.world)]
            dist.all_gather(sz, torch.tensor([t.shape[dim]],
                                             device=t.device))
            sizes = [int(x.item()) for x in sz]
        maxd = max(sizes)
        padded_shape = list(t.shape)
        padded_shape[dim] = maxd
        send = t
        if t.shape[dim] < maxd:
            send = torch.zeros(padded_shape, dtype=t.dtype, device=t.device)
            send.narrow(dim, 0, t.shape[dim]).copy_(t)
        recv = [torch.empty(padded_shape, dtype=t.dtype, device=t.device)
                for _ in range(self.world)]
        dist.all_gather(recv, send.contiguous())
        return torch.cat([r.narrow(dim, 0, s) for r, s in zip(recv, sizes)],
                         dim=dim)

    def barrier(self):
        if self.world > 1:
            dist.barrier()

    def shard_of(self, total: int):
        """Global model indices owned by this rank under strided sharding."""
        return list(range(self.rank, total, self.world))

    def unshard_order(self, total: int) -> torch.Tensor:
        """Permutation mapping all_gather_cat order -> global model order.

        all_gather_cat over the H axis yields models in order
        [rank0's models, rank1's, ...] = [0, world, 2*world, ..., 1, ...];
        this returns idx such that gathered[idx] is in global order.
        """
        order = []
        for r in range(self.world):
            order.extend(range(r, total, self.world))
        perm = torch.empty(total, dtype=torch.long)
        perm[torch.tensor(order)] = torch.arange(total)
        return perm


_COMM: Optional[Comm] = None


def get_comm() -> Comm:
    global _COMM
    if _COMM is None:
        _COMM = Comm()
    return _COMM


def init_from_env(backend: Optional[str] = None, device=None) -> Comm:
    """Initialize torch.distributed from torchrun env vars (one proc/GPU).

    backend defaults to "nccl" (= RCCL on ROCm) when CUDA is available,
    else "gloo". Safe to call when WORLD_SIZE is unset (returns the no-op
    Comm).
    """
    global _COMM
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        _COMM = Comm(device=device)
        return _COMM
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if device is None:
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
            device = torch.device("cuda", local_rank)
        else:
            device = torch.device("cpu")
    if not dist.is_initialized():
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=300))
    _COMM = Comm(rank=rank, world=world, device=device)
    return _COMM
