"""Process-group setup for multi-GPU execution.

One process per GPU, ``torch.distributed`` with backend "nccl" (RCCL on
ROCm) over xGMI; "gloo" for CPU tests.  Reads the torchrun env
(RANK/LOCAL_RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT).
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None, timeout_s: float = 300.0) -> int:
    """Initialize from the torchrun env; returns the local rank.  No-op when
    WORLD_SIZE is absent or 1."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
        )
    return local_rank


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def barrier() -> None:
    if is_distributed():
        dist.barrier()


def all_reduce_(t: torch.Tensor) -> torch.Tensor:
    if is_distributed():
        dist.all_reduce(t)
    return t
