"""Process-group bootstrap.

Reference behavior: ``dist.init_process_group(backend)`` with env:// rendezvous
from MASTER_ADDR / MASTER_PORT / RANK / WORLD_SIZE. On ROCm the "nccl" backend
IS RCCL, riding xGMI between GPUs of one node; "gloo" stays the CPU plumbing
path (BASELINE.json config 1).
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def pick_backend(explicit: str | None = None) -> str:
    if explicit:
        return explicit
    return "nccl" if torch.cuda.is_available() else "gloo"


def init_distributed(backend: str | None = None, rank: int | None = None,
                     world_size: int | None = None, timeout_s: int = 300) -> tuple[int, int]:
    """env:// rendezvous; returns (rank, world_size). Safe to call when the
    process group is already initialized (returns current values)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    rank = int(os.environ.get("RANK", 0)) if rank is None else rank
    world_size = int(os.environ.get("WORLD_SIZE", 1)) if world_size is None else world_size
    backend = pick_backend(backend)
    dist.init_process_group(
        backend=backend,
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    if backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))
    return rank, world_size
