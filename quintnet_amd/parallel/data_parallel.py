"""Data parallelism: bucketed gradient all-reduce over RCCL.

Same capability as the reference DDP stack
(parallelism/data_parallel/core/ddp.py:49-179 + components/), but
implemented the *intended* way — the reference ships with gradient sync
silently disabled and a ready-counter that never resets (SURVEY.md §8.1,
§8.2).  MI355X-first design:

* one persistent flat gradient buffer per bucket; ``param.grad`` are
  views into it, so micro-batch accumulation lands in place and the
  bucket is all-reduced with zero pack/unpack copies;
* buckets fill in reverse parameter order (backward order), default
  capacity 25 MB — sized so a 2-rank xGMI all-reduce (one 153 GB/s
  link) stays bandwidth-bound, not latency-bound;
* reduction fires from post-accumulate-grad hooks on the LAST
  micro-batch only (``require_backward_grad_sync``), asynchronously —
  RCCL runs on its own stream, overlapping the remaining backward;
  ``finalize_gradients()`` waits and applies MEAN scaling;
* per-iteration ready counters reset in ``zero_grad``.
"""

from __future__ import annotations

import contextlib
import dataclasses
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from .backends import DistributedBackend, TorchDistributedBackend, LocalBackend

__all__ = ["DataParallel", "BucketConfig", "DistributedConfig", "GradientBucket"]


@dataclasses.dataclass
class BucketConfig:
    capacity_mb: float = 25.0
    # kept for API parity with the reference (core/config.py:47-84); flat
    # bucket views are always on here — that was the flag's intent.
    gradient_as_bucket_view: bool = True


@dataclasses.dataclass
class DistributedConfig:
    rank: int = 0
    world_size: int = 1
    process_group: Optional[object] = None


class GradientBucket:
    """A contiguous flat grad buffer covering a param group."""

    def __init__(self, params: List[nn.Parameter], dtype: torch.dtype, device: torch.device):
        self.params = params
        self.numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(self.numel, dtype=dtype, device=device)
        self.views: List[torch.Tensor] = []
        off = 0
        for p in params:
            self.views.append(self.flat[off : off + p.numel()].view_as(p))
            off += p.numel()
        self.ready_count = 0
        self.work = None  # in-flight async all-reduce

    def attach_grads(self) -> None:
        for p, v in zip(self.params, self.views):
            p.grad = v

    def reset(self) -> None:
        self.flat.zero_()
        self.ready_count = 0
        self.work = None


class DataParallel(nn.Module):
    def __init__(
        self,
        module: nn.Module,
        config: Optional[DistributedConfig] = None,
        bucket_config: Optional[BucketConfig] = None,
        backend: Optional[DistributedBackend] = None,
    ):
        super().__init__()
        self.module = module
        self.bucket_config = bucket_config or BucketConfig()
        if backend is not None:
            self.backend = backend
        elif config is not None and config.process_group is not None:
            self.backend = TorchDistributedBackend(config.process_group)
        elif dist.is_initialized():
            self.backend = TorchDistributedBackend(None)
        else:
            self.backend = LocalBackend()
        self.world_size = self.backend.get_world_size()
        self.require_backward_grad_sync = True
        self.buckets: List[GradientBucket] = []
        self._param_to_bucket: Dict[int, GradientBucket] = {}
        self._hooks = []
        self._setup()

    # ------------------------------------------------------------------
    def _setup(self) -> None:
        self._broadcast_parameters()
        self._create_buckets()
        self._register_hooks()

    def _broadcast_parameters(self) -> None:
        if self.world_size <= 1:
            return
        for p in self.module.parameters():
            self.backend.broadcast_tensor(p.data)
        for b in self.module.buffers():
            self.backend.broadcast_tensor(b.data)

    def _create_buckets(self) -> None:
        params = [p for p in self.module.parameters() if p.requires_grad]
        if not params:
            return
        # reverse order ≈ backward completion order
        params = list(reversed(params))
        cap = int(self.bucket_config.capacity_mb * 1024 * 1024)
        cur: List[nn.Parameter] = []
        cur_bytes = 0
        groups: List[List[nn.Parameter]] = []
        for p in params:
            sz = p.numel() * p.element_size()
            if cur and cur_bytes + sz > cap:
                groups.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += sz
        if cur:
            groups.append(cur)
        for g in groups:
            # one buffer per dtype within the group
            by_dtype: Dict[torch.dtype, List[nn.Parameter]] = {}
            for p in g:
                by_dtype.setdefault(p.dtype, []).append(p)
            for dt, ps in by_dtype.items():
                bucket = GradientBucket(ps, dt, ps[0].device)
                bucket.attach_grads()
                self.buckets.append(bucket)
                for p in ps:
                    self._param_to_bucket[id(p)] = bucket

    def _register_hooks(self) -> None:
        if self.world_size <= 1:
            return
        for bucket in self.buckets:
            for p in bucket.params:
                h = p.register_post_accumulate_grad_hook(self._make_hook(bucket))
                self._hooks.append(h)

    def _make_hook(self, bucket: GradientBucket):
        def hook(_param):
            if not self.require_backward_grad_sync:
                return
            bucket.ready_count += 1
            if bucket.ready_count == len(bucket.params) and bucket.work is None:
                # async: RCCL stream overlaps with the rest of backward
                bucket.work = self.backend.all_reduce_tensor(bucket.flat, async_op=True)

        return hook

    # ------------------------------------------------------------------
    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def backward(self, input_tensor, output_tensor, output_tensor_grad):
        """PP-style manual backward passthrough (reference ddp.py:137)."""
        if hasattr(self.module, "backward"):
            return self.module.backward(input_tensor, output_tensor, output_tensor_grad)
        torch.autograd.backward(output_tensor, grad_tensors=output_tensor_grad)
        return input_tensor.grad if input_tensor is not None else None

    @contextlib.contextmanager
    def no_sync(self):
        prev = self.require_backward_grad_sync
        self.require_backward_grad_sync = False
        try:
            yield
        finally:
            self.require_backward_grad_sync = prev

    def finalize_gradients(self) -> None:
        """Launch any pending reductions, wait all, apply MEAN scaling."""
        if self.world_size <= 1 or not self.require_backward_grad_sync:
            return
        for bucket in self.buckets:
            if bucket.work is None:
                bucket.work = self.backend.all_reduce_tensor(bucket.flat, async_op=True)
        for bucket in self.buckets:
            if bucket.work is not None:
                bucket.work.wait()
                bucket.work = None
            bucket.flat.div_(self.world_size)
            bucket.ready_count = 0

    def zero_grad(self, set_to_none: bool = False):  # noqa: ARG002 — views must persist
        for bucket in self.buckets:
            bucket.reset()
            bucket.attach_grads()
        # params outside buckets (requires_grad=False) have no grads
