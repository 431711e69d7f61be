"""Fused LayerNorm (fwd + bwd) — hand-written CDNA4 kernel on GPU.

One-pass Welford rowwise kernel, wave-per-row for hidden 64/768/3072
(csrc/layernorm.hip).  Replaces the implicit cuDNN/eager LayerNorm at
reference utils/model.py:214,216 and utils/GPT2/gpt2_block.py:116,139.

``FusedLayerNorm.forward(x, residual=...)`` additionally fuses the
preceding residual add into the same kernel pass (SURVEY.md §2.4
"fused residual-add variant"), returning ``(normed, x + residual)`` so
the transformer residual stream never pays a standalone add kernel.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from . import _backend

__all__ = ["layer_norm", "layer_norm_residual", "FusedLayerNorm", "LayerNormFunction"]


class LayerNormFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ctx.eps = eps
        if _backend.use_native(x) and _backend.has_ext():
            y, mean, rstd = _backend.ext().layernorm_fwd(x.contiguous(), weight, bias, eps, None)
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
                dy.contiguous(), x.contiguous(), weight, mean, rstd, None
            )
        else:
            dx, dw, db = _ln_bwd_ref(dy, x, weight, mean, rstd)
        return dx, dw, db, None


def _ln_bwd_ref(dy, x, weight, mean, rstd):
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
    return dx, dw, db


class LayerNormResidualFunction(torch.autograd.Function):
    """(x, res) -> (LN(x+res), x+res); grads of both outputs fused into
    one backward pass (the residual stream's add never materializes as
    its own kernel)."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, eps):
        ctx.eps = eps
        if _backend.use_native(x) and _backend.has_ext():
            y, mean, rstd, s = _backend.ext().layernorm_fwd(
                x.contiguous(), weight, bias, eps, res.contiguous()
            )
        else:
            s = x + res
            sf = s.float()
            mean = sf.mean(dim=-1)
            var = sf.var(dim=-1, unbiased=False)
            rstd = torch.rsqrt(var + eps)
            y = ((sf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1))
            y = (y * weight.float() + bias.float()).to(x.dtype)
        ctx.save_for_backward(s, weight, mean, rstd)
        return y, s

    @staticmethod
    def backward(ctx, dy, ds):
        s, weight, mean, rstd = ctx.saved_tensors
        if _backend.use_native(s) and _backend.has_ext():
            dx, dw, db = _backend.ext().layernorm_bwd(
                dy.contiguous(), s.contiguous(), weight, mean, rstd,
                ds.contiguous() if ds is not None else None,
            )
        else:
            dx, dw, db = _ln_bwd_ref(dy, s, weight, mean, rstd)
            if ds is not None:
                dx = dx + ds
        return dx, dx, dw, db, None


def layer_norm(x, weight, bias, eps: float = 1e-5):
    return LayerNormFunction.apply(x, weight, bias, eps)


def layer_norm_residual(x, res, weight, bias, eps: float = 1e-5):
    return LayerNormResidualFunction.apply(x, res, weight, bias, eps)


class FusedLayerNorm(nn.Module):
    """Drop-in nn.LayerNorm replacement backed by the HIP kernel.

    ``forward(x, residual=r)`` returns ``(LN(x+r), x+r)``."""

    def __init__(self, hidden: int, eps: float = 1e-5, device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.normalized_shape = (hidden,)
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden, **kw))
        self.bias = nn.Parameter(torch.zeros(hidden, **kw))

    def forward(self, x, residual=None):
        if residual is None:
            return layer_norm(x, self.weight, self.bias, self.eps)
        return layer_norm_residual(x, residual, self.weight, self.bias, self.eps)

    def extra_repr(self):
        return f"{self.normalized_shape}, eps={self.eps}"
