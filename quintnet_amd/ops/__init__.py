"""Hand-written CDNA4 (gfx950) op library with CPU test fallbacks.

Every op dispatches to the in-tree HIP extension (quintnet_amd._C) on
GPU — failing loudly if it is missing — and to a plain PyTorch fp32
reference on CPU, which is also what the numerics tests compare the
kernels against.
"""

from ._backend import ext, has_ext, use_native, force_eager
from .linear import linear, LinearFunction, defer_wgrads, flush_deferred_wgrads
from .layernorm import layer_norm, layer_norm_residual, FusedLayerNorm, LayerNormFunction
from .attention import (
    attention,
    attention_qkv,
    AttentionFunction,
    FlashAttentionFunction,
    causal_softmax,
    softmax_bwd,
)
from .cross_entropy import cross_entropy, causal_lm_loss, shift_labels, CrossEntropyFunction
from .adamw import (
    adamw_step_flat,
    clip_grad_norm_global,
    clip_grad_norm_local,
    grad_sq_norm_contrib,
    l2_norm,
)
from .dropout import fused_dropout, FusedDropout
from .embedding import embedding_pair

__all__ = [
    "ext",
    "has_ext",
    "use_native",
    "force_eager",
    "linear",
    "LinearFunction",
    "defer_wgrads",
    "flush_deferred_wgrads",
    "layer_norm",
    "layer_norm_residual",
    "FusedLayerNorm",
    "LayerNormFunction",
    "attention",
    "attention_qkv",
    "AttentionFunction",
    "FlashAttentionFunction",
    "causal_softmax",
    "softmax_bwd",
    "cross_entropy",
    "causal_lm_loss",
    "shift_labels",
    "CrossEntropyFunction",
    "adamw_step_flat",
    "clip_grad_norm_local",
    "clip_grad_norm_global",
    "grad_sq_norm_contrib",
    "l2_norm",
    "fused_dropout",
    "FusedDropout",
    "embedding_pair",
]
