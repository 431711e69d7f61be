"""ZeRO-3 parameter sharding (FSDP-style), beyond reference parity.

Each wrapped block's parameters are flattened into ONE flat tensor and
sharded 1/dp across the group; the block's own parameters are freed.
The forward runs inside a `torch.utils.checkpoint` region whose first
op all-gathers the shard — so the full parameters exist only while the
block computes, activations inside the block are recomputed in
backward (re-gathering the params then too), and the gather's backward
is a reduce-scatter that leaves each rank exactly its shard's gradient.
This is the production "FSDP + activation checkpointing" configuration:
at-rest and between-block memory is 1/dp for params, grads AND block
activations, at ~1/3 extra forward FLOPs.

Any plain optimizer stepping the shard parameters is automatically
ZeRO-sharded (each rank only ever owns 1/dp of the states).

MI355X note: with 288 GB HBM3E this is for the models that NEED it
(ZeRO-1 + full replicas is faster below ~70B-parameter scale); the
collectives are single `all_gather_into_tensor` / `reduce_scatter`
calls per block over RCCL/xGMI.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.utils.checkpoint import checkpoint

__all__ = ["ZeRO3Block", "apply_zero3"]


def _group_size(group) -> int:
    if group is None or not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)


class _GatherFlat(torch.autograd.Function):
    """all-gather the padded flat shard; backward reduce-scatters."""

    @staticmethod
    def forward(ctx, shard: torch.Tensor, group):
        ctx.group = group
        world = _group_size(group)
        ctx.world = world
        if world == 1:
            return shard
        full = torch.empty(
            world * shard.numel(), dtype=shard.dtype, device=shard.device
        )
        dist.all_gather_into_tensor(full, shard.contiguous(), group=group)
        return full

    @staticmethod
    def backward(ctx, grad):
        if ctx.world == 1:
            return grad, None
        grad = grad.contiguous()
        out = torch.empty(
            grad.numel() // ctx.world, dtype=grad.dtype, device=grad.device
        )
        backend = dist.get_backend(ctx.group)
        if backend == "nccl":  # RCCL on ROCm
            dist.reduce_scatter_tensor(out, grad, group=ctx.group)
        else:  # gloo has no reduce_scatter: all-reduce then slice
            dist.all_reduce(grad, group=ctx.group)
            rank = dist.get_rank(group=ctx.group)
            out.copy_(grad.view(ctx.world, -1)[rank])
        return out, None


class ZeRO3Block(nn.Module):
    """Wrap one module (typically a transformer block) with ZeRO-3
    parameter sharding + activation checkpointing."""

    def __init__(self, module: nn.Module, dp_group=None):
        super().__init__()
        # dropout IS recompute-stable here: FusedDropout draws its
        # per-call seed from the torch CPU RNG (ops/dropout.py), which
        # non-reentrant checkpoint stashes and restores — proven exact
        # in tests/test_activation_checkpointing.py dropout test.
        self.module = module
        self.dp_group = dp_group
        self.world = _group_size(dp_group)
        self.rank = dist.get_rank(group=dp_group) if self.world > 1 else 0

        names: List[str] = []
        metas: List[Tuple[torch.Size, int]] = []
        tensors: List[torch.Tensor] = []
        dtype = None
        for name, p in module.named_parameters():
            if dtype is None:
                dtype = p.dtype
            if p.dtype != dtype:
                raise ValueError("ZeRO3Block: params must share one dtype")
            names.append(name)
            metas.append((p.shape, p.numel()))
            tensors.append(p.detach().reshape(-1))
        if not tensors:
            raise ValueError("ZeRO3Block: module has no parameters to shard")
        self._names = names
        self._metas = metas
        total = sum(n for _, n in metas)
        pad = (-total) % self.world
        flat = torch.empty(total + pad, dtype=dtype, device=tensors[0].device)
        torch.cat(tensors + ([flat.new_zeros(pad)] if pad else []), out=flat)
        self.padded = total + pad

        # free the module's own parameter storage; shard is the only copy
        shard_len = self.padded // self.world
        shard = flat[self.rank * shard_len : (self.rank + 1) * shard_len].clone()
        self.shard = nn.Parameter(shard)
        for _, p in module.named_parameters():
            p.requires_grad_(False)
            p.data = torch.empty(0, dtype=p.dtype, device=p.device)

    # ------------------------------------------------------------------
    def _param_views(self, full: torch.Tensor) -> Dict[str, torch.Tensor]:
        out: Dict[str, torch.Tensor] = {}
        off = 0
        for name, (shape, numel) in zip(self._names, self._metas):
            out[name] = full[off : off + numel].view(shape)
            off += numel
        return out

    def _run(self, shard: torch.Tensor, *args, **kwargs):
        full = _GatherFlat.apply(shard, self.dp_group)
        return torch.func.functional_call(
            self.module, self._param_views(full), args, kwargs
        )

    def forward(self, *args, **kwargs):
        if torch.is_grad_enabled() and self.shard.requires_grad:
            return checkpoint(
                self._run, self.shard, *args, use_reentrant=False, **kwargs
            )
        return self._run(self.shard, *args, **kwargs)

    # residual-fused entry (GPT2Stage's block chain calls this instead
    # of forward — models/gpt2/stage.py); same gather/checkpoint shape
    def _run_fused(self, shard: torch.Tensor, x, pending):
        full = _GatherFlat.apply(shard, self.dp_group)
        return self._call_method(
            self._param_views(full), "forward_fused", x, pending
        )

    def _call_method(self, views, name, *args):
        """functional_call only invokes module.forward — reparametrize
        manually for other entry points (forward_fused)."""
        from torch.nn.utils.stateless import _reparametrize_module

        with _reparametrize_module(self.module, views):
            return getattr(self.module, name)(*args)

    def forward_fused(self, x, pending):
        if torch.is_grad_enabled() and self.shard.requires_grad:
            return checkpoint(
                self._run_fused, self.shard, x, pending, use_reentrant=False
            )
        return self._run_fused(self.shard, x, pending)

    # ------------------------------------------------------------------
    def full_state_dict_tensors(self) -> Dict[str, torch.Tensor]:
        """Re-gather the full parameters (e.g. for checkpoint saving)."""
        with torch.no_grad():
            full = _GatherFlat.apply(self.shard, self.dp_group)
            return {k: v.clone() for k, v in self._param_views(full).items()}


def apply_zero3(
    model: nn.Module,
    dp_group=None,
    attr: str = "blocks",
    extra_attrs: tuple = (),
) -> nn.Module:
    """Shard every element of ``model.<attr>`` (a ModuleList) in place;
    ``extra_attrs`` names additional single modules (e.g. "embedding",
    "classification_head") to shard the same way — with every trainable
    module wrapped, NO separate DP sync is needed (each shard's gradient
    is already reduce-scattered across the group)."""
    blocks = getattr(model, attr)
    wrapped = nn.ModuleList(ZeRO3Block(b, dp_group) for b in blocks)
    setattr(model, attr, wrapped)
    for name in extra_attrs:
        setattr(model, name, ZeRO3Block(getattr(model, name), dp_group))
    return model
