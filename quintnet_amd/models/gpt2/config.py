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

    def __post_init__(self):
        if self.n_inner is None:
            self.n_inner = 4 * self.n_embd

    @property
    def head_dim(self) -> int:
        return self.n_embd // self.n_head

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
