"""GPT-2 container for the interleaved (virtual-pipeline) schedule.

GPT2Stage fuses ln_f + the tied LM head into its own forward, so the
generic embedding/.blocks/head split the wrappers need cannot be
expressed with it.  This container exposes exactly that contract:
``.embedding`` (wte+wpe), ``.blocks`` (GPT2Block list, TP-aware) and
``.head`` = LayerNorm -> TiedLMHead sharing the wte Parameter.  Under
pp>1 the first and last global stages then hold separate copies of the
tied weight; InterleavedPipelineWrapper detects them (duck-typed `wte`
/ `tied_weight` attrs) and all-reduces their grads over the tied
first+last-stage subgroup."""

from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import FusedLayerNorm
from ...ops import linear as fused_linear
from .block import GPT2Block
from .config import GPT2Config
from .embeddings import GPT2Embedding

__all__ = ["TiedLMHead", "GPT2ForInterleaving"]


class TiedLMHead(nn.Module):
    """LM head projecting with the embedding matrix (weight tying)."""

    def __init__(self, wte: nn.Embedding, config: GPT2Config = None):
        super().__init__()
        self.wte = wte  # shared module -> shared Parameter
        self._config = config

    @property
    def tied_weight(self) -> nn.Parameter:
        return self.wte.weight

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        lim = self._config.vocab_size if self._config is not None else None
        return fused_linear(x, self.wte.weight, None, None,
                            prefer_library=True, logical_out=lim)


class GPT2ForInterleaving(nn.Module):
    def __init__(self, config: GPT2Config, tp_group=None, device=None, dtype=None,
                 ep_group=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.config = config
        self.embedding = GPT2Embedding(config, **kw)
        self.blocks = nn.ModuleList(
            GPT2Block(config, tp_group=tp_group, ep_group=ep_group, **kw)
            for _ in range(config.n_layer)
        )
        self.head = nn.Sequential(
            FusedLayerNorm(config.n_embd, eps=config.layer_norm_epsilon, **kw),
            TiedLMHead(self.embedding.wte, config),
        )
        self.seq_len = config.n_positions
        self.hidden_dim = config.n_embd

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        x = self.embedding(input_ids)
        for blk in self.blocks:
            x = blk(x)
        return self.head(x)
