"""Fused multi-tensor AdamW step (HIP kernel) + global grad-norm clip.

The kernel (csrc/adamw.hip) updates a flat fp32 master copy + m/v state
and writes the model's bf16 (or fp32) params in one pass — the ZeRO-1
optimizer shards these flat buffers across the DP group.  Replaces the
implicit eager Adam/AdamW loops at reference trainer.py:89 /
GPT2_Trainer.py:100-104 and optimizers/zero.py (a TODO stub there).
"""

from __future__ import annotations

from typing import Iterable, List, Optional

import torch

from . import _backend

__all__ = ["adamw_step_flat", "clip_grad_norm_local", "l2_norm"]


def adamw_step_flat(
    param_out: torch.Tensor,      # flat model-dtype params (written)
    master: torch.Tensor,         # flat fp32 master (updated)
    grad: torch.Tensor,           # flat grads (model dtype or fp32)
    exp_avg: torch.Tensor,        # flat fp32 m
    exp_avg_sq: torch.Tensor,     # flat fp32 v
    step: int,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step_dev: Optional[torch.Tensor] = None,  # int64 GPU scalar (graph-safe)
) -> None:
    """One fused AdamW step over a flat shard. Decoupled weight decay.

    When ``step_dev`` is given the kernel reads the step count from the
    device (bias correction stays correct under hipGraph replay)."""
    if _backend.use_native(param_out) and _backend.has_ext():
        _backend.ext().adamw_step(
            param_out, master, grad, exp_avg, exp_avg_sq,
            step, lr, beta1, beta2, eps, weight_decay, step_dev,
        )
        return
    g = grad.float()
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    master.mul_(1 - lr * weight_decay)
    master.addcdiv_(exp_avg, denom, value=-lr / bc1)
    param_out.copy_(master.to(param_out.dtype))


def l2_norm(tensors: Iterable[torch.Tensor]) -> torch.Tensor:
    """Sum-of-squares based L2 norm over a tensor list (fp32 accumulate)."""
    ts = [t for t in tensors if t is not None]
    if not ts:
        return torch.zeros(())
    if _backend.use_native(*ts) and _backend.has_ext():
        sq = _backend.ext().multi_tensor_sumsq(ts)
        return sq.sqrt()
    total = torch.zeros((), dtype=torch.float32, device=ts[0].device)
    for t in ts:
        total += t.float().pow(2).sum()
    return total.sqrt()


def clip_grad_norm_local(
    params: Iterable[torch.nn.Parameter], max_norm: float
) -> torch.Tensor:
    """Per-rank grad clipping (matches the reference's local-clip
    semantics — SURVEY.md §8.4; a reduced global-norm variant lives in
    the ZeRO optimizer)."""
    grads: List[torch.Tensor] = [p.grad for p in params if p.grad is not None]
    if not grads:
        return torch.zeros(())
    norm = l2_norm(grads)
    clip = max_norm / (norm + 1e-6)
    if clip < 1.0:
        for g in grads:
            g.mul_(clip.to(g.dtype))
    return norm
