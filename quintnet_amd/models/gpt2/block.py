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
    def __init__(self, config: GPT2Config, tp_group=None, device=None, dtype=None):
        super().__init__()
        self.ln_1 = FusedLayerNorm(config.n_embd, eps=config.layer_norm_epsilon, device=device, dtype=dtype)
        self.attn = GPT2Attention(config, tp_group=tp_group, device=device, dtype=dtype)
        self.ln_2 = FusedLayerNorm(config.n_embd, eps=config.layer_norm_epsilon, device=device, dtype=dtype)
        self.mlp = GPT2MLP(config, tp_group=tp_group, device=device, dtype=dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x + self.attn(self.ln_1(x))
        x = x + self.mlp(self.ln_2(x))
        return x
