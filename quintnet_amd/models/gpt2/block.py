"""GPT-2 transformer block (reference utils/GPT2/gpt2_block.py:57-188).

Pre-norm residual block; LayerNorms are replicated across TP ranks and
run the fused HIP kernel.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import FusedLayerNorm
from .attention import GPT2Attention
from .config import GPT2Config
from .mlp import GPT2MLP

__all__ = ["GPT2Block"]


class GPT2Block(nn.Module):
    def __init__(self, config: GPT2Config, tp_group=None, device=None, dtype=None,
                 cp_group=None, ep_group=None):
        super().__init__()
        self.ln_1 = FusedLayerNorm(config.n_embd, eps=config.layer_norm_epsilon, device=device, dtype=dtype)
        self.attn = GPT2Attention(config, tp_group=tp_group, device=device, dtype=dtype,
                                  cp_group=cp_group)
        self.ln_2 = FusedLayerNorm(config.n_embd, eps=config.layer_norm_epsilon, device=device, dtype=dtype)
        if config.n_experts > 0:
            from ...parallel.expert_parallel import ExpertParallelMLP

            self.mlp = ExpertParallelMLP(
                config.n_embd, config.n_inner, config.n_experts,
                top_k=config.moe_top_k, ep_group=ep_group, tp_group=tp_group,
                device=device, dtype=dtype,
                capacity_factor=getattr(config, "moe_capacity_factor", 0.0),
            )
        else:
            self.mlp = GPT2MLP(config, tp_group=tp_group, device=device, dtype=dtype)
        # recompute-in-backward for the PLAIN forward path (interleaved
        # chunks call blocks directly); GPT2Stage handles its own fused
        # chain (models/gpt2/stage.py)
        self._ckpt = bool(
            getattr(config, "activation_checkpointing", False)
            and not config.n_experts
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._ckpt and self.training and torch.is_grad_enabled():
            from torch.utils.checkpoint import checkpoint

            m, s2 = checkpoint(self.forward_fused, x, None, use_reentrant=False)
        else:
            m, s2 = self.forward_fused(x, None)
        return s2 + m

    def forward_cached(self, x: torch.Tensor, kv_cache: dict) -> torch.Tensor:
        """Plain (non-residual-fused) form for incremental decoding."""
        n1 = self.ln_1(x)
        s1 = x + self.attn(n1, kv_cache=kv_cache)
        n2 = self.ln_2(s1)
        return s1 + self.mlp(n2)

    def forward_fused(self, x: torch.Tensor, pending):
        """Residual-fused form: the incoming pending residual (previous
        block's MLP output) is added INSIDE ln_1's kernel; this block's
        MLP output is returned as the next pending residual.

        Returns (mlp_out, post_attention_sum)."""
        if pending is None:
            n1, s1 = self.ln_1(x), x
        else:
            n1, s1 = self.ln_1(x, residual=pending)
        a = self.attn(n1)
        n2, s2 = self.ln_2(s1, residual=a)
        m = self.mlp(n2)
        return m, s2
