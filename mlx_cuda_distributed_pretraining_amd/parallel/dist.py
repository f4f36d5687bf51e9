"""Process-group plumbing for single-node data parallelism.

Replaces the reference's thread-queue "distributed" layer
(/root/reference/distributed/hybrid_distributed.py — whose gradient averaging
is Python-mean over JSON-serialized numpy lists and whose remote compute is
mocked) with one process per GPU and RCCL collectives over
xGMI (torch.distributed backend "nccl" IS RCCL on ROCm). Tests use gloo on
CPU with world_size > 1.
"""
from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


def init_distributed(backend: Optional[str] = None, timeout_s: int = 600):
    """Initialize from torchrun env vars. Returns (rank, world_size, local_rank).

    Safe to call in single-process mode (no env vars): returns (0, 1, 0)
    without creating a process group.
    """
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size(), int(os.environ.get("LOCAL_RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return 0, 1, 0
    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        torch.cuda.set_device(local_rank)
    dist.init_process_group(
        backend=backend,
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    return rank, world_size, local_rank


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def barrier() -> None:
    if is_distributed():
        dist.barrier()


def all_reduce_scalar(value: torch.Tensor, op: str = "sum") -> torch.Tensor:
    """Scalar (loss / token-count) all-reduce — collective C2 in SURVEY.md §2.6."""
    if not is_distributed():
        return value
    value = value.clone()
    dist.all_reduce(value, op=dist.ReduceOp.SUM)
    if op == "mean":
        value /= get_world_size()
    return value


def broadcast_module(module: torch.nn.Module, src: int = 0) -> None:
    """Broadcast parameters and buffers from rank src (collective C3)."""
    if not is_distributed():
        return
    for t in list(module.parameters()) + list(module.buffers()):
        dist.broadcast(t.data, src=src)
