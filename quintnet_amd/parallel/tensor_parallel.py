"""Tensor parallelism: Column/Row-parallel linears + vocab-parallel embedding.

Megatron-style TP over RCCL xGMI subgroups.  Parity with reference
parallelism/tensor_parallel/layers.py:42-297 and model_wrapper.py:37-166,
backed by the hand-written MFMA GEMM (quintnet_amd.ops.linear) with
fused bias/activation epilogues.  For the [2,2,2] mesh each TP pair is
one xGMI link; the all-reduce of a RowParallelLinear output is a single
cross-write + local add on that link.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..core.comm import All_Gather, All_Reduce, ReduceScatter, copy_to_group
from ..ops import linear as fused_linear

__all__ = [
    "ColumnParallelLinear",
    "RowParallelLinear",
    "VocabParallelEmbedding",
    "apply_tensor_parallel",
    "ensure_divisibility",
]


def ensure_divisibility(numerator: int, denominator: int) -> None:
    if numerator % denominator != 0:
        raise ValueError(f"{numerator} is not divisible by {denominator}")


def _group_size(group) -> int:
    # None = NO tensor parallelism (size 1).  A PP/DP-only rank must
    # never fall back to the default world group — that silently turns
    # replicated layers into world-wide TP.  Pass dist.group.WORLD
    # explicitly to shard over the whole world.
    if group is None or not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)


def _group_rank(group) -> int:
    if group is None or not dist.is_initialized():
        return 0
    return dist.get_rank(group=group)


class ColumnParallelLinear(nn.Module):
    """Linear with output features sharded over the TP group.

    Y = X·Aᵀ with A = [A_1; …; A_p] sharded by rows of the [out, in]
    weight.  ``gather_output=True`` all-gathers shards along the last
    dim (backward slices); ``False`` leaves the output parallel (the
    Megatron pairing into a following RowParallelLinear).
    """

    def __init__(
        self,
        in_features: int,
        out_features: int,
        tp_group=None,
        bias: bool = True,
        gather_output: bool = True,
        activation: Optional[str] = None,
        sequence_parallel: bool = False,
        device=None,
        dtype=None,
    ):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.tp_group = tp_group
        self.sequence_parallel = sequence_parallel
        self.tp_size = _group_size(tp_group)
        self.tp_rank = _group_rank(tp_group)
        ensure_divisibility(out_features, self.tp_size)
        self.out_per_rank = out_features // self.tp_size
        self.gather_output = gather_output
        self.activation = activation
        kw = {"device": device, "dtype": dtype}
        self.weight = nn.Parameter(torch.empty(self.out_per_rank, in_features, **kw))
        self.bias = nn.Parameter(torch.empty(self.out_per_rank, **kw)) if bias else None
        if self.tp_size > 1:
            # grad-norm bookkeeping: these shards are DISTINCT per TP rank
            # (clip_grad_norm_global sums their squared norms over the TP
            # group; unmarked params are replicated and counted once)
            self.weight._tp_sharded = True
            if self.bias is not None:
                self.bias._tp_sharded = True
        self.reset_parameters()

    def reset_parameters(self) -> None:
        nn.init.normal_(self.weight, mean=0.0, std=0.02)
        if self.bias is not None:
            nn.init.zeros_(self.bias)

    @classmethod
    def from_linear(cls, lin: nn.Linear, tp_group, gather_output=True, device=None):
        m = cls(
            lin.in_features,
            lin.out_features,
            tp_group=tp_group,
            bias=lin.bias is not None,
            gather_output=gather_output,
            device=device or lin.weight.device,
            dtype=lin.weight.dtype,
        )
        r = m.tp_rank
        with torch.no_grad():
            sl = slice(r * m.out_per_rank, (r + 1) * m.out_per_rank)
            m.weight.copy_(lin.weight[sl].to(m.weight.device))
            if m.bias is not None:
                m.bias.copy_(lin.bias[sl].to(m.bias.device))
        return m

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.tp_size > 1:
            if self.sequence_parallel:
                # Megatron-SP g operator: all-gather the sequence shards
                # fwd, reduce-scatter grads bwd
                x = All_Gather.apply(x, self.tp_group, 1, "reduce_scatter")
            else:
                x = copy_to_group(x, self.tp_group)  # identity fwd, all-reduce bwd
        out = fused_linear(x, self.weight, self.bias, self.activation,
                           fp8=getattr(self, "fp8", False))
        if self.gather_output and self.tp_size > 1:
            out = All_Gather.apply(out, self.tp_group, -1, "slice")
        return out

    def extra_repr(self):
        return (
            f"in={self.in_features}, out={self.out_features}, tp={self.tp_size}, "
            f"gather_output={self.gather_output}"
        )


class RowParallelLinear(nn.Module):
    """Linear with input features sharded over the TP group.

    Y = Σ_p X_p·A_pᵀ — each rank computes a partial product on its
    [out, in/p] weight shard; partials are summed with one RCCL
    all-reduce (identity backward).  Bias is added after the reduction.
    """

    def __init__(
        self,
        in_features: int,
        out_features: int,
        tp_group=None,
        bias: bool = True,
        input_is_parallel: bool = True,
        sequence_parallel: bool = False,
        device=None,
        dtype=None,
    ):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.tp_group = tp_group
        self.sequence_parallel = sequence_parallel
        self.tp_size = _group_size(tp_group)
        self.tp_rank = _group_rank(tp_group)
        ensure_divisibility(in_features, self.tp_size)
        self.in_per_rank = in_features // self.tp_size
        self.input_is_parallel = input_is_parallel
        kw = {"device": device, "dtype": dtype}
        self.weight = nn.Parameter(torch.empty(out_features, self.in_per_rank, **kw))
        # NOTE: the row-parallel bias is REPLICATED (added after the
        # reduction) — only the weight is marked sharded
        self.bias = nn.Parameter(torch.empty(out_features, **kw)) if bias else None
        if self.tp_size > 1:
            self.weight._tp_sharded = True
        self.reset_parameters()

    def reset_parameters(self) -> None:
        nn.init.normal_(self.weight, mean=0.0, std=0.02)
        if self.bias is not None:
            nn.init.zeros_(self.bias)

    @classmethod
    def from_linear(cls, lin: nn.Linear, tp_group, input_is_parallel=False, device=None):
        m = cls(
            lin.in_features,
            lin.out_features,
            tp_group=tp_group,
            bias=lin.bias is not None,
            input_is_parallel=input_is_parallel,
            device=device or lin.weight.device,
            dtype=lin.weight.dtype,
        )
        r = m.tp_rank
        with torch.no_grad():
            sl = slice(r * m.in_per_rank, (r + 1) * m.in_per_rank)
            m.weight.copy_(lin.weight[:, sl].to(m.weight.device))
            if m.bias is not None:
                m.bias.copy_(lin.bias.to(m.bias.device))
        return m

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.tp_size == 1:
            # bias fuses into the GEMM epilogue when there is no reduction
            return fused_linear(x, self.weight, self.bias, None,
                                fp8=getattr(self, "fp8", False))
        if not self.input_is_parallel:
            x = x.chunk(self.tp_size, dim=-1)[self.tp_rank].contiguous()
        out = fused_linear(x, self.weight, None, None,
                           fp8=getattr(self, "fp8", False))
        if self.sequence_parallel:
            # Megatron-SP ḡ operator: reduce-scatter the partial sums
            # over the sequence dim (grads all-gather back)
            out = ReduceScatter.apply(out, self.tp_group, 1)
        else:
            out = All_Reduce.apply(out, self.tp_group)
        if self.bias is not None:
            out = out + self.bias
        return out

    def extra_repr(self):
        return f"in={self.in_features}, out={self.out_features}, tp={self.tp_size}"


class VocabParallelEmbedding(nn.Module):
    """Embedding with the vocab dim sharded over the TP group.

    Out-of-range ids are masked to 0, looked up locally, zeroed, then
    summed across the group (reference layers.py:266-297).
    """

    def __init__(
        self,
        num_embeddings: int,
        embedding_dim: int,
        tp_group=None,
        device=None,
        dtype=None,
    ):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.tp_group = tp_group
        self.tp_size = _group_size(tp_group)
        self.tp_rank = _group_rank(tp_group)
        ensure_divisibility(num_embeddings, self.tp_size)
        self.vocab_per_rank = num_embeddings // self.tp_size
        self.vocab_start = self.tp_rank * self.vocab_per_rank
        self.vocab_end = self.vocab_start + self.vocab_per_rank
        kw = {"device": device, "dtype": dtype}
        self.weight = nn.Parameter(torch.empty(self.vocab_per_rank, embedding_dim, **kw))
        if self.tp_size > 1:
            self.weight._tp_sharded = True
        nn.init.normal_(self.weight, mean=0.0, std=0.02)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        if self.tp_size == 1:
            return nn.functional.embedding(ids, self.weight)
        mask = (ids >= self.vocab_start) & (ids < self.vocab_end)
        local_ids = (ids - self.vocab_start).masked_fill(~mask, 0)
        out = nn.functional.embedding(local_ids, self.weight)
        out = out * mask.unsqueeze(-1).to(out.dtype)
        return All_Reduce.apply(out, self.tp_group)


def apply_tensor_parallel(
    model: nn.Module,
    tp_size: Optional[int] = None,
    tp_rank: Optional[int] = None,
    tp_group=None,
    device: Optional[torch.device] = None,
    mode: str = "column",
) -> nn.Module:
    """Recursively replace every shardable nn.Linear with a TP shard.

    Parity with reference model_wrapper.py:37-166.  ``mode='column'``
    replaces with all-gathered ColumnParallelLinear (generic models,
    e.g. the ViT); ``mode='row'`` with RowParallelLinear.  Layers whose
    dims don't divide by tp_size are left unsharded.
    """
    tp_size = tp_size if tp_size is not None else _group_size(tp_group)
    if tp_size <= 1:
        if device is not None:
            model.to(device)
        return model

    def replace(module: nn.Module) -> None:
        for name, child in list(module.named_children()):
            if isinstance(child, (ColumnParallelLinear, RowParallelLinear)):
                continue
            if isinstance(child, nn.Linear):
                if mode == "column" and child.out_features % tp_size == 0:
                    new = ColumnParallelLinear.from_linear(
                        child, tp_group, gather_output=True, device=device
                    )
                elif mode == "row" and child.in_features % tp_size == 0:
                    new = RowParallelLinear.from_linear(
                        child, tp_group, input_is_parallel=False, device=device
                    )
                else:
                    continue
                setattr(module, name, new)
            else:
                replace(child)

    replace(model)
    if device is not None:
        model.to(device)
    if dist.is_initialized():
        dist.barrier()
    return model
