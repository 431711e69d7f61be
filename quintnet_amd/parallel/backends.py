"""Communication backend seam for the DP engine.

Parity with reference parallelism/data_parallel/backends/ (base.py:30-91,
torch_backend.py:31-85, local_backend.py:9-22): a tiny ABC so the DDP
wrapper can run without any distributed init in unit tests.  On MI355X
the real backend is torch.distributed over RCCL.
"""

from __future__ import annotations

import abc
from typing import Optional

import torch
import torch.distributed as dist

__all__ = ["DistributedBackend", "TorchDistributedBackend", "LocalBackend"]


class DistributedBackend(abc.ABC):
    @abc.abstractmethod
    def is_initialized(self) -> bool: ...

    @abc.abstractmethod
    def get_world_size(self) -> int: ...

    @abc.abstractmethod
    def get_rank(self) -> int: ...

    @abc.abstractmethod
    def broadcast_tensor(self, tensor: torch.Tensor, src: int) -> None: ...

    @abc.abstractmethod
    def all_reduce_tensor(self, tensor: torch.Tensor, async_op: bool = False): ...

    def reduce_scatter_tensor(self, output: torch.Tensor, input: torch.Tensor,
                              async_op: bool = False):
        raise NotImplementedError

    def all_gather_into_tensor(self, output: torch.Tensor, input: torch.Tensor,
                               async_op: bool = False):
        raise NotImplementedError


class TorchDistributedBackend(DistributedBackend):
    """RCCL (or gloo) through torch.distributed on a given group."""

    def __init__(self, process_group: Optional[dist.ProcessGroup] = None):
        self.group = process_group

    def is_initialized(self) -> bool:
        return dist.is_initialized()

    def get_world_size(self) -> int:
        return dist.get_world_size(group=self.group) if dist.is_initialized() else 1

    def get_rank(self) -> int:
        return dist.get_rank(group=self.group) if dist.is_initialized() else 0

    def group_src_global_rank(self) -> int:
        if not dist.is_initialized():
            return 0
        if self.group is None:
            return 0
        return dist.get_global_rank(self.group, 0)

    def broadcast_tensor(self, tensor: torch.Tensor, src: Optional[int] = None) -> None:
        if self.get_world_size() <= 1:
            return
        if src is None:
            src = self.group_src_global_rank()
        dist.broadcast(tensor, src=src, group=self.group)

    def all_reduce_tensor(self, tensor: torch.Tensor, async_op: bool = False):
        if self.get_world_size() <= 1:
            return None
        return dist.all_reduce(tensor, op=dist.ReduceOp.SUM, group=self.group, async_op=async_op)

    def reduce_scatter_tensor(self, output: torch.Tensor, input: torch.Tensor,
                              async_op: bool = False):
        if self.get_world_size() <= 1:
            output.copy_(input[: output.numel()])
            return None
        return dist.reduce_scatter_tensor(
            output, input, op=dist.ReduceOp.SUM, group=self.group,
            async_op=async_op)

    def all_gather_into_tensor(self, output: torch.Tensor, input: torch.Tensor,
                               async_op: bool = False):
        if self.get_world_size() <= 1:
            output.copy_(input)
            return None
        return dist.all_gather_into_tensor(output, input, group=self.group,
                                           async_op=async_op)


class LocalBackend(DistributedBackend):
    """No-op backend for single-process unit tests.  ``world_size`` can
    be faked so bucket-LAYOUT logic (padding, chunking) is testable
    without spawning processes — the comm ops remain no-ops."""

    def __init__(self, world_size: int = 1, rank: int = 0):
        self._world, self._rank = world_size, rank

    def is_initialized(self) -> bool:
        return True

    def get_world_size(self) -> int:
        return self._world

    def get_rank(self) -> int:
        return self._rank

    def broadcast_tensor(self, tensor: torch.Tensor, src: int = 0) -> None:
        return None

    def all_reduce_tensor(self, tensor: torch.Tensor, async_op: bool = False):
        return None

    def reduce_scatter_tensor(self, output: torch.Tensor, input: torch.Tensor,
                              async_op: bool = False):
        output.copy_(input.view(self._world, -1)[self._rank])
        return None

    def all_gather_into_tensor(self, output: torch.Tensor, input: torch.Tensor,
                               async_op: bool = False):
        output.copy_(input)
        return None
