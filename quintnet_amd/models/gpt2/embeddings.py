"""GPT-2 token + position embedding (reference utils/GPT2/gpt2_embeddings.py).

wte/wpe are replicated across TP ranks (the TP axis shards the block
GEMMs; the embedding gather is HBM-bandwidth-bound, not FLOP-bound).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import FusedDropout, embedding_pair
from .config import GPT2Config

__all__ = ["GPT2Embedding"]


class GPT2Embedding(nn.Module):
    def __init__(self, config: GPT2Config, device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.config = config
        # padded_vocab_size == vocab_size unless vocab_pad_to is set; the
        # pad rows are zero-init, masked out of the logits, and get zero
        # gradient — pure layout, identical math (config.py).
        self.wte = nn.Embedding(config.padded_vocab_size, config.n_embd, **kw)
        self.wpe = nn.Embedding(config.n_positions, config.n_embd, **kw)
        self.drop = FusedDropout(config.dropout)
        nn.init.normal_(self.wte.weight, std=config.initializer_range)
        if config.padded_vocab_size != config.vocab_size:
            with torch.no_grad():
                self.wte.weight[config.vocab_size:].zero_()
        nn.init.normal_(self.wpe.weight, std=config.initializer_range)

    def forward(self, input_ids: torch.Tensor, pos_offset: int = 0,
                positions: torch.Tensor = None) -> torch.Tensor:
        # pos_offset > 0: incremental decoding — positions start past the
        # KV cache (the fused kernel indexes positions 0..T-1 of whatever
        # wpe slice it is given, so a slice view is all that's needed).
        # positions: explicit NON-CONTIGUOUS position ids (zigzag CP).
        wpe = self.wpe.weight
        if positions is not None:
            wpe = wpe.index_select(0, positions)
        elif pos_offset:
            wpe = wpe[pos_offset : pos_offset + input_ids.shape[1]]
        x = embedding_pair(input_ids, self.wte.weight, wpe)
        return self.drop(x)
