"""Context parallelism (CP): the sequence axis sharded across GPUs for
attention.  Beyond reference parity (the reference has no CP).

MI355X-first design choice: with 288 GB of HBM3E per GPU, the all-gather-KV
formulation is the right first CP flavor — each rank holds its Q/K/V
sequence shard, all-gathers K and V over the CP group (one RCCL all-gather
each, backward = reduce-scatter of the KV grads via the autograd-aware
``All_Gather`` op), and runs the fused flash kernel on its Q shard against
the full K/V with the causal diagonal shifted by ``q_offset = cp_rank *
T_local`` (kernel support: csrc/attn.hip).  Activation memory for
everything OUTSIDE attention scales 1/cp; attention compute scales ~1/cp
on average (the causal upper ranks do more — ring/zigzag balancing is a
possible future refinement).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..core.comm import All_Gather
from ..ops.attention import attention

__all__ = ["context_parallel_attention", "scatter_to_context"]


def _ws(group) -> int:
    if group is None or not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)


def scatter_to_context(x: torch.Tensor, cp_group, dim: int = 1) -> torch.Tensor:
    """Keep this rank's sequence shard (data is assumed replicated)."""
    world = _ws(cp_group)
    if world == 1:
        return x
    if x.shape[dim] % world != 0:
        raise ValueError(
            f"context parallel: dim {dim} ({x.shape[dim]}) must divide by cp={world}"
        )
    rank = dist.get_rank(group=cp_group)
    return x.chunk(world, dim=dim)[rank].contiguous()


def context_parallel_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cp_group,
    causal: bool = True,
) -> torch.Tensor:
    """Attention over the full sequence from per-rank [B, H, T/cp, D] shards.

    Returns this rank's [B, H, T/cp, D] output shard.  Differentiable:
    dK/dV flow back through the all-gather's reduce-scatter, dQ stays
    local.
    """
    world = _ws(cp_group)
    if world == 1:
        return attention(q, k, v, causal=causal)
    rank = dist.get_rank(group=cp_group)
    t_local = q.shape[-2]
    k_full = All_Gather.apply(k, cp_group, -2, "reduce_scatter")
    v_full = All_Gather.apply(v, cp_group, -2, "reduce_scatter")
    return attention(q, k_full, v_full, causal=causal, q_offset=rank * t_local)
