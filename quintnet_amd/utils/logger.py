"""Per-rank stdout/stderr tee logger (reference utils/logger.py:6-45)."""

from __future__ import annotations

import os
import sys

__all__ = ["setup_rank_logger", "print_rank_0"]


class _Tee:
    def __init__(self, stream, fh):
        self.stream = stream
        self.fh = fh

    def write(self, data):
        self.stream.write(data)
        self.fh.write(data)
        self.fh.flush()

    def flush(self):
        self.stream.flush()
        self.fh.flush()


def setup_rank_logger(log_dir: str = "logs") -> str:
    rank = int(os.environ.get("RANK", 0))
    os.makedirs(log_dir, exist_ok=True)
    path = os.path.join(log_dir, f"rank_{rank}.log")
    fh = open(path, "a", buffering=1)
    sys.stdout = _Tee(sys.__stdout__, fh)
    sys.stderr = _Tee(sys.__stderr__, fh)
    return path


def print_rank_0(*args, **kwargs):
    if int(os.environ.get("RANK", 0)) == 0:
        print(*args, **kwargs)
