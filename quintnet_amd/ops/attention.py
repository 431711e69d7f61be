"""Attention for ViT (non-causal) and GPT-2 (causal) on MI355X.

Design (v1, memory-rich MI355X layout): QK^T and PV are plain GEMMs
(hipBLASLt batched), while the scale+causal-mask+softmax step — the
fusion opportunity — is one hand-written CDNA4 kernel (fwd + bwd,
csrc/softmax.hip): one pass over each score row, wave-per-row online
max/sum.  The probability tensor P is kept resident for backward
(288 GB HBM3E: at GPT-2 base B=8,H=12,T=1024 P is ~200 MB — re-reading
it beats recomputing on this part).

Replaces reference F.scaled_dot_product_attention
(utils/GPT2/gpt2_attention.py:156) and the explicit QK^T/softmax/V at
utils/model.py:98-105.
"""

from __future__ import annotations

import math

import torch

from . import _backend

__all__ = ["attention", "AttentionFunction", "causal_softmax", "softmax_bwd"]


def causal_softmax(scores: torch.Tensor, scale: float, causal: bool) -> torch.Tensor:
    """P = softmax(scale * scores [+ causal mask]) along the last dim."""
    if _backend.use_native(scores) and _backend.has_ext():
        return _backend.ext().softmax_fwd(scores.contiguous(), scale, causal)
    s = scores.float() * scale
    if causal:
        T, S = s.shape[-2], s.shape[-1]
        mask = torch.ones(T, S, dtype=torch.bool, device=s.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    return torch.softmax(s, dim=-1).to(scores.dtype)


def softmax_bwd(p: torch.Tensor, dp: torch.Tensor, scale: float) -> torch.Tensor:
    """dS = scale * P ⊙ (dP - rowsum(dP ⊙ P))  (softmax backward)."""
    if _backend.use_native(p) and _backend.has_ext():
        return _backend.ext().softmax_bwd(p.contiguous(), dp.contiguous(), scale)
    pf, dpf = p.float(), dp.float()
    row = (pf * dpf).sum(dim=-1, keepdim=True)
    return (scale * pf * (dpf - row)).to(p.dtype)


class AttentionFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        # q,k,v: [B, H, T, D]
        scale = 1.0 / math.sqrt(q.shape[-1])
        scores = torch.matmul(q, k.transpose(-2, -1))
        p = causal_softmax(scores, scale, causal)
        out = torch.matmul(p, v)
        ctx.save_for_backward(q, k, v, p)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, p = ctx.saved_tensors
        dout = dout.contiguous()
        dv = torch.matmul(p.transpose(-2, -1), dout)
        dp = torch.matmul(dout, v.transpose(-2, -1))
        ds = softmax_bwd(p, dp, ctx.scale)
        dq = torch.matmul(ds, k)
        dk = torch.matmul(ds.transpose(-2, -1), q)
        return dq, dk, dv, None


def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, causal: bool = False) -> torch.Tensor:
    """Multi-head attention core: softmax(QK^T/sqrt(d) [+mask]) V.

    q, k, v: [B, H, T, D] (same head count — TP shards heads upstream).
    """
    return AttentionFunction.apply(q, k, v, causal)
