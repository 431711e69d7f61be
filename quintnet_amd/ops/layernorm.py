"""Fused LayerNorm (fwd + bwd) — hand-written CDNA4 kernel on GPU.

One-pass Welford rowwise kernel, wave-per-row for hidden 64/768/3072
(csrc/layernorm.hip).  Replaces the implicit cuDNN/eager LayerNorm at
reference utils/model.py:214,216 and utils/GPT2/gpt2_block.py:116,139.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from . import _backend

__all__ = ["layer_norm", "FusedLayerNorm", "LayerNormFunction"]


class LayerNormFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ctx.eps = eps
        if _backend.use_native(x) and _backend.has_ext():
            y, mean, rstd = _backend.ext().layernorm_fwd(x.contiguous(), weight, bias, eps)
        else:
            xf = x.float()
            mean = xf.mean(dim=-1)
            var = xf.var(dim=-1, unbiased=False)
            rstd = torch.rsqrt(var + eps)
            y = ((xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1))
            y = (y * weight.float() + bias.float()).to(x.dtype)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        if _backend.use_native(x) and _backend.has_ext():
            dx, dw, db = _backend.ext().layernorm_bwd(
                dy.contiguous(), x.contiguous(), weight, mean, rstd
            )
        else:
            xf = x.float()
            dyf = dy.float()
            H = x.shape[-1]
            xhat = (xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
            wdy = dyf * weight.float()
            c1 = wdy.mean(dim=-1, keepdim=True)
            c2 = (wdy * xhat).mean(dim=-1, keepdim=True)
            dx = ((wdy - c1 - xhat * c2) * rstd.unsqueeze(-1)).to(x.dtype)
            dw = (dyf * xhat).reshape(-1, H).sum(0).to(weight.dtype)
            db = dyf.reshape(-1, H).sum(0).to(weight.dtype)
        return dx, dw, db, None


def layer_norm(x, weight, bias, eps: float = 1e-5):
    return LayerNormFunction.apply(x, weight, bias, eps)


class FusedLayerNorm(nn.Module):
    """Drop-in nn.LayerNorm replacement backed by the HIP kernel."""

    def __init__(self, hidden: int, eps: float = 1e-5, device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.normalized_shape = (hidden,)
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden, **kw))
        self.bias = nn.Parameter(torch.zeros(hidden, **kw))

    def forward(self, x):
        return layer_norm(x, self.weight, self.bias, self.eps)

    def extra_repr(self):
        return f"{self.normalized_shape}, eps={self.eps}"
