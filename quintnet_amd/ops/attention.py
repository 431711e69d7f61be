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


def causal_softmax(
    scores: torch.Tensor, scale: float, causal: bool, q_offset: int = 0
) -> torch.Tensor:
    """P = softmax(scale * scores [+ causal mask]) along the last dim.

    ``q_offset`` shifts the causal diagonal: query row i is globally row
    ``i + q_offset`` (context parallelism — K covers the full sequence).
    """
    if (
        q_offset == 0
        and scores.shape[-1] == scores.shape[-2]
        and _backend.use_native(scores)
        and _backend.has_ext()
    ):
        return _backend.ext().softmax_fwd(scores.contiguous(), scale, causal)
    s = scores.float() * scale
    if causal:
        T, S = s.shape[-2], s.shape[-1]
        mask = torch.ones(T, S, dtype=torch.bool, device=s.device).tril(q_offset)
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
    def forward(ctx, q, k, v, causal, q_offset=0):
        # q: [B, H, Tq, D]; k, v: [B, H, Tk, D]
        scale = 1.0 / math.sqrt(q.shape[-1])
        scores = torch.matmul(q, k.transpose(-2, -1))
        p = causal_softmax(scores, scale, causal, q_offset)
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
        return dq, dk, dv, None, None


def _flash_ok(q: torch.Tensor) -> bool:
    import os

    if os.environ.get("QN_NO_FLASH") == "1":
        return False
    return (
        _backend.has_ext()
        and not _backend.force_eager()
        and q.is_cuda
        and q.dtype == torch.bfloat16
        and q.shape[-1] == 64
        and q.shape[-2] % 128 == 0
    )


class FlashAttentionFunction(torch.autograd.Function):
    """Fully-fused flash-style attention (csrc/attn.hip, D=64 bf16)."""

    @staticmethod
    def forward(ctx, q, k, v, causal, q_offset=0):
        scale = 1.0 / math.sqrt(q.shape[-1])
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        lse2 = _backend.ext().attn_fwd(q, k, v, out, scale, causal, q_offset)
        ctx.save_for_backward(q, k, v, out, lse2)
        ctx.scale = scale
        ctx.causal = causal
        ctx.q_offset = q_offset
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse2 = ctx.saved_tensors
        if dout.stride(-1) != 1 or dout.stride(-2) % 8 != 0:
            dout = dout.contiguous()
        dq = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        dk = torch.empty(k.shape, dtype=k.dtype, device=k.device)
        dv = torch.empty(v.shape, dtype=v.dtype, device=v.device)
        _backend.ext().attn_bwd(
            q, k, v, out, dout, lse2, dq, dk, dv, ctx.scale, ctx.causal,
            ctx.q_offset,
        )
        return dq, dk, dv, None, None


class FlashAttentionQKV(torch.autograd.Function):
    """Fused attention directly on the packed [B, T, 3*H*D] QKV tensor.

    Consumes the column-parallel c_attn output without any
    transpose/contiguous copies (strided fragment loads) and writes the
    [B, T, H*D] output layout the row-parallel c_proj wants; backward
    fills one packed dqkv the same way.
    """

    @staticmethod
    def forward(ctx, qkv, n_heads, causal):
        B, T, three_hl = qkv.shape
        hl = three_hl // 3
        D = hl // n_heads
        scale = 1.0 / math.sqrt(D)

        def split_view(t, off, width):
            return t[:, :, off : off + width].view(B, T, n_heads, D).permute(0, 2, 1, 3)

        qv = split_view(qkv, 0, hl)
        kv = split_view(qkv, hl, hl)
        vv = split_view(qkv, 2 * hl, hl)
        out = qkv.new_empty(B, T, hl)
        out_view = out.view(B, T, n_heads, D).permute(0, 2, 1, 3)
        lse2 = _backend.ext().attn_fwd(qv, kv, vv, out_view, scale, causal)
        ctx.save_for_backward(qkv, out, lse2)
        ctx.meta = (n_heads, D, hl, causal, scale)
        return out

    @staticmethod
    def backward(ctx, dout):
        qkv, out, lse2 = ctx.saved_tensors
        n_heads, D, hl, causal, scale = ctx.meta
        B, T, _ = qkv.shape
        dout = dout.contiguous()

        def split_view(t, off, width):
            return t[:, :, off : off + width].view(B, T, n_heads, D).permute(0, 2, 1, 3)

        qv = split_view(qkv, 0, hl)
        kv = split_view(qkv, hl, hl)
        vv = split_view(qkv, 2 * hl, hl)
        out_view = out.view(B, T, n_heads, D).permute(0, 2, 1, 3)
        dout_view = dout.view(B, T, n_heads, D).permute(0, 2, 1, 3)
        dqkv = torch.empty_like(qkv)
        dqv = split_view(dqkv, 0, hl)
        dkv = split_view(dqkv, hl, hl)
        dvv = split_view(dqkv, 2 * hl, hl)
        _backend.ext().attn_bwd(
            qv, kv, vv, out_view, dout_view, lse2, dqv, dkv, dvv, scale, causal
        )
        return dqkv, None, None


def attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = False,
    q_offset: int = 0,
) -> torch.Tensor:
    """Multi-head attention core: softmax(QK^T/sqrt(d) [+mask]) V.

    q: [B, H, Tq, D]; k, v: [B, H, Tk, D] (same head count — TP shards
    heads upstream).  ``q_offset`` places the query shard at global rows
    [q_offset, q_offset+Tq) of the K/V sequence (context parallelism).
    Dispatches to the fully-fused flash kernel on gfx950 (D=64,
    Tq/Tk/q_offset%128==0); otherwise the composed GEMM+softmax path.
    """
    if (
        _flash_ok(q)
        and k.shape == v.shape
        and q.shape[-1] == k.shape[-1]
        and k.shape[-2] % 128 == 0
        and q_offset % 128 == 0
        and q_offset + q.shape[-2] <= k.shape[-2]
    ):
        def ok(t):
            return t.stride(-1) == 1 and all(s % 8 == 0 for s in t.stride()[:-1])

        qc = q if ok(q) else q.contiguous()
        kc = k if ok(k) else k.contiguous()
        vc = v if ok(v) else v.contiguous()
        return FlashAttentionFunction.apply(qc, kc, vc, causal, q_offset)
    return AttentionFunction.apply(q, k, v, causal, q_offset)


def attention_qkv(qkv: torch.Tensor, n_heads: int, causal: bool = True) -> torch.Tensor:
    """Attention on the packed QKV projection output [B, T, 3*H*D] ->
    [B, T, H*D] (local layout ``[q | k | v]``, per-head sharded)."""
    B, T, three_hl = qkv.shape
    hl = three_hl // 3
    D = hl // n_heads
    import os

    if (
        os.environ.get("QN_NO_FLASH") != "1"
        and _backend.has_ext()
        and not _backend.force_eager()
        and qkv.is_cuda
        and qkv.dtype == torch.bfloat16
        and D == 64
        and T % 128 == 0
    ):
        return FlashAttentionQKV.apply(qkv, n_heads, causal)
    q, k, v = qkv.split(hl, dim=-1)

    def heads(t):
        return t.view(B, T, n_heads, D).transpose(1, 2)

    out = attention(heads(q), heads(k), heads(v), causal=causal)
    return out.transpose(1, 2).reshape(B, T, hl)
