"""Fused dropout (HIP kernel with counter-based PRNG on GPU; torch on CPU).

Replaces reference nn.Dropout uses (gpt2_attention.py:108-109,
gpt2_mlp.py:125, gpt2_embeddings.py:59).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from . import _backend

__all__ = ["fused_dropout", "FusedDropout"]


class _DropoutFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, seed):
        y, mask = _backend.ext().dropout_fwd(x, p, seed)
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        dx = _backend.ext().dropout_bwd(dy, mask, ctx.p)
        return dx, None, None


def fused_dropout(x: torch.Tensor, p: float, training: bool = True) -> torch.Tensor:
    if not training or p <= 0.0:
        return x
    if _backend.use_native(x) and _backend.has_ext():
        seed = int(torch.randint(0, 2**31 - 1, (1,)).item())
        return _DropoutFunction.apply(x, p, seed)
    return torch.nn.functional.dropout(x, p, training)


class FusedDropout(nn.Module):
    """Drop-in nn.Dropout replacement backed by the HIP kernel."""

    def __init__(self, p: float = 0.5):
        super().__init__()
        self.p = p

    def forward(self, x):
        return fused_dropout(x, self.p, self.training)

    def extra_repr(self):
        return f"p={self.p}"
