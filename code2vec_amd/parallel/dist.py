"""Distributed context helpers — one process per GPU over RCCL/xGMI.

The reference has no distributed layer at all; this is the
green-field DP substrate: torch.distributed with backend "nccl" (RCCL on
ROCm) on GPU, "gloo" for CPU tests.  Rendezvous comes from the standard
torchrun env (RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


class DistContext:
    def __init__(self, rank: int, world_size: int, local_rank: int, device):
        self.rank = rank
        self.world_size = world_size
        self.local_rank = local_rank
        self.device = device

    @property
    def is_rank0(self) -> bool:
        return self.rank == 0

    @property
    def initialized(self) -> bool:
        return self.world_size > 1


def init_distributed(device_type: str = "auto", timeout_s: int = 600) -> DistContext:
    """Initialize from torchrun env; single-process if WORLD_SIZE unset/1."""
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if device_type == "auto":
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
    if device_type == "cuda":
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
        backend = "nccl"  # = RCCL on ROCm
    else:
        device = torch.device("cpu")
        backend = "gloo"
    if world_size > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world_size,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    return DistContext(rank, world_size, local_rank, device)


def all_reduce_sum_scalar(value: float, ctx: DistContext) -> float:
    if not ctx.initialized:
        return value
    t = torch.tensor([value], dtype=torch.float64, device=ctx.device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return float(t.item())


def all_reduce_sum_list(values, ctx: DistContext):
    if not ctx.initialized:
        return list(values)
    t = torch.tensor(list(values), dtype=torch.float64, device=ctx.device)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t.tolist()


def barrier(ctx: DistContext) -> None:
    if ctx.initialized:
        dist.barrier()
