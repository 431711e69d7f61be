"""Fused token+position embedding lookup (HIP gather; scatter-add bwd).

Replaces the wte(ids)+wpe(pos) eager gathers of reference
utils/GPT2/gpt2_embeddings.py:92-95.
"""

from __future__ import annotations

import torch

from . import _backend

__all__ = ["embedding_pair"]


class _EmbeddingPairFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, wte, wpe):
        out = _backend.ext().embedding_pair_fwd(ids, wte, wpe)
        ctx.save_for_backward(ids)
        ctx.shapes = (wte.shape[0], wpe.shape[0])
        return out

    @staticmethod
    def backward(ctx, dout):
        (ids,) = ctx.saved_tensors
        vocab, n_pos = ctx.shapes
        dwte, dwpe = _backend.ext().embedding_pair_bwd(ids, dout, vocab, n_pos)
        return None, dwte, dwpe


def embedding_pair(ids: torch.Tensor, wte: torch.Tensor, wpe: torch.Tensor) -> torch.Tensor:
    """out[b, t] = wte[ids[b, t]] + wpe[t]  (one fused gather pass)."""
    if _backend.use_native(wte) and _backend.has_ext() and ids.dtype == torch.int64:
        return _EmbeddingPairFunction.apply(ids, wte, wpe)
    T = ids.shape[-1]
    pos = torch.arange(T, device=ids.device)
    return torch.nn.functional.embedding(ids, wte) + torch.nn.functional.embedding(pos, wpe)
