"""GPT-2 config with base/medium/large/xl presets.

Reference parity: utils/GPT2/gpt2_config.py:22-168.
"""

from __future__ import annotations

import dataclasses
from typing import Optional

__all__ = ["GPT2Config"]


@dataclasses.dataclass
class GPT2Config:
    vocab_size: int = 50257
    n_positions: int = 1024
    n_embd: int = 768
    n_layer: int = 12
    n_head: int = 12
    n_inner: Optional[int] = None  # defaults to 4*n_embd
    dropout: float = 0.1
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02
    sequence_parallel: bool = False
    fp8: bool = False  # experimental: e4m3 forward GEMMs, bf16 backward
    cp_ring: bool = False  # context parallelism: ring KV exchange
    cp_zigzag: bool = False  # CP: zigzag load-balanced ring (implies ring)
    n_experts: int = 0  # >0: MoE MLP blocks (expert parallelism)
    moe_top_k: int = 2  # Megatron-SP over the TP group
    # >0: Switch-style expert capacity — each expert serves at most
    # ceil(cf * tokens * top_k / n_experts) assignments per rank;
    # overflow contributes zero (residual carries the token).  0 = the
    # default exact variable-size routing (nothing dropped).
    moe_capacity_factor: float = 0.0
    # Recompute each block's activations in backward instead of storing
    # them (torch.utils.checkpoint, non-reentrant): activation memory
    # drops from O(n_layer) to O(1) blocks at ~1.33x forward FLOPs.
    # Incompatible with MoE blocks (the aux load-balancing loss is read
    # outside the checkpointed region).
    activation_checkpointing: bool = False
    # Pad the EMBEDDING TABLE (and hence the logits width) up to a
    # multiple of this, keeping vocab_size as the logical width.  50257
    # gives every logits-sized tensor odd-element rows — unaligned
    # vectorized access in the LM-head GEMMs and CE kernels.  128-pad
    # (50304) measures 1.16 ms/step faster on the whole LM-head block
    # (tools/probe_vocab_pad.py).  Logits columns >= vocab_size are
    # masked to -inf before any loss/argmax, so the math is EXACTLY the
    # unpadded model's (pad rows get zero grad).  0 = off (checkpoint-
    # compatible with real HF shapes).
    vocab_pad_to: int = 0

    def __post_init__(self):
        if self.n_inner is None:
            self.n_inner = 4 * self.n_embd

    @property
    def head_dim(self) -> int:
        return self.n_embd // self.n_head

    @property
    def padded_vocab_size(self) -> int:
        p = self.vocab_pad_to
        if not p:
            return self.vocab_size
        return (self.vocab_size + p - 1) // p * p

    @classmethod
    def gpt2_base(cls, **kw) -> "GPT2Config":
        return cls(n_embd=768, n_layer=12, n_head=12, **kw)

    @classmethod
    def gpt2_medium(cls, **kw) -> "GPT2Config":
        return cls(n_embd=1024, n_layer=24, n_head=16, **kw)

    @classmethod
    def gpt2_large(cls, **kw) -> "GPT2Config":
        return cls(n_embd=1280, n_layer=36, n_head=20, **kw)

    @classmethod
    def gpt2_xl(cls, **kw) -> "GPT2Config":
        return cls(n_embd=1600, n_layer=48, n_head=25, **kw)

    @classmethod
    def from_name(cls, name: str, **kw) -> "GPT2Config":
        table = {
            "base": cls.gpt2_base,
            "gpt2": cls.gpt2_base,
            "small": cls.gpt2_base,
            "medium": cls.gpt2_medium,
            "large": cls.gpt2_large,
            "xl": cls.gpt2_xl,
        }
        return table[name](**kw)
