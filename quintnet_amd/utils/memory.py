"""GPU memory stats (reference utils/memory.py:11-33)."""

from __future__ import annotations

import torch

__all__ = ["memory_stats", "print_memory_stats"]


def memory_stats(device=None) -> dict:
    if not torch.cuda.is_available():
        return {"allocated_mb": 0.0, "reserved_mb": 0.0, "max_allocated_mb": 0.0, "max_reserved_mb": 0.0}
    return {
        "allocated_mb": torch.cuda.memory_allocated(device) / 2**20,
        "reserved_mb": torch.cuda.memory_reserved(device) / 2**20,
        "max_allocated_mb": torch.cuda.max_memory_allocated(device) / 2**20,
        "max_reserved_mb": torch.cuda.max_memory_reserved(device) / 2**20,
    }


def print_memory_stats(tag: str = "", device=None) -> None:
    s = memory_stats(device)
    print(
        f"[mem{' ' + tag if tag else ''}] alloc {s['allocated_mb']:.0f}MB "
        f"reserved {s['reserved_mb']:.0f}MB peak {s['max_allocated_mb']:.0f}MB"
    )
