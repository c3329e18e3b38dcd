"""Distributed context for the 8x MI355X node (the
reference's implicit parallelism — Kafka partitions / managed Flink —
mapped to explicit DP/TP/EP groups).

One process per GPU over torch.distributed: backend "nccl" IS RCCL on
ROCm, collectives ride xGMI (7 p2p links x ~153 GB/s per GPU).  CPU-only
test runs (this repo's CI) use gloo with the same call surface, so every
multi-rank path is covered by world_size=2 gloo tests without a GPU.

Group layout mirrors Megatron's grid: ranks = dp_size x tp_size, with TP
ranks contiguous (adjacent xGMI hops carry the latency-critical per-layer
all-reduces; DP/EP collectives are bandwidth-bound and overlap-friendly).
"""

from __future__ import annotations

import os
from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    backend: str = "none"
    device: str = "cpu"
    tp_size: int = 1
    tp_group: object = None
    dp_group: object = None

    @property
    def tp_rank(self) -> int:
        return self.rank % self.tp_size

    @property
    def dp_rank(self) -> int:
        return self.rank // self.tp_size

    @property
    def dp_size(self) -> int:
        return self.world_size // self.tp_size


def init_distributed(tp_size: int = 1, backend: str | None = None,
                     device: str | None = None) -> DistContext:
    """Initialize from torchrun env (RANK/WORLD_SIZE/LOCAL_RANK); no-op
    single-rank context when WORLD_SIZE is absent or 1."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = torch.cuda.is_available()
    if device is None:
        device = f"cuda:{local_rank}" if use_cuda else "cpu"
    if world <= 1:
        return DistContext(device=device)
    if backend is None:
        backend = "nccl" if use_cuda else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend)
    if use_cuda:
        torch.cuda.set_device(device)
    assert world % tp_size == 0, "world_size must divide by tp_size"
    tp_group = dp_group = None
    if tp_size > 1 or world > tp_size:
        # build ALL groups on every rank (collective contract)
        for start in range(0, world, tp_size):
            g = dist.new_group(list(range(start, start + tp_size)))
            if start <= rank < start + tp_size:
                tp_group = g
        for tr in range(tp_size):
            g = dist.new_group(list(range(tr, world, tp_size)))
            if rank % tp_size == tr:
                dp_group = g
    return DistContext(rank, world, local_rank, backend, device,
                       tp_size, tp_group, dp_group)


def barrier(ctx: DistContext) -> None:
    if ctx.world_size > 1:
        dist.barrier()


def max_over_ranks(ctx: DistContext, value: float) -> float:
    if ctx.world_size <= 1:
        return value
    t = torch.tensor([value], dtype=torch.float64)
    if ctx.backend == "nccl":
        t = t.to(ctx.device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def sum_over_ranks(ctx: DistContext, value: float,
                   group=None) -> float:
    if ctx.world_size <= 1:
        return value
    t = torch.tensor([value], dtype=torch.float64)
    if ctx.backend == "nccl":
        t = t.to(ctx.device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM, group=group)
    return float(t.item())
