"""Small distributed helpers (reference parity: core/distributed.py:43-59)."""

from __future__ import annotations

import os

import torch
import torch.distributed as dist

__all__ = [
    "get_rank",
    "get_world_size",
    "get_local_rank",
    "is_main_process",
    "setup_distributed",
    "cleanup_distributed",
    "barrier",
]


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def get_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", 0))


def is_main_process() -> bool:
    return get_rank() == 0


def setup_distributed(backend: str = None, timeout_s: float = 600.0) -> None:
    """Initialize torch.distributed from torchrun env vars.

    Backend defaults to nccl (=RCCL) when a GPU is visible, else gloo.
    """
    import datetime

    if dist.is_initialized():
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    if backend == "nccl":
        torch.cuda.set_device(get_local_rank())


def cleanup_distributed() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()
