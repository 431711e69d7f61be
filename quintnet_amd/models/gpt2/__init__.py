from .config import GPT2Config
from .embeddings import GPT2Embedding
from .attention import GPT2Attention
from .mlp import GPT2MLP
from .block import GPT2Block
from .stage import GPT2Stage
from .interleaved import GPT2ForInterleaving, TiedLMHead
from .decode import StaticKVDecoder
from .speculative import speculative_generate
from .beam import beam_search

__all__ = [
    "StaticKVDecoder",
    "speculative_generate",
    "beam_search",
    "GPT2Config",
    "GPT2Embedding",
    "GPT2Attention",
    "GPT2MLP",
    "GPT2Block",
    "GPT2Stage",
    "GPT2ForInterleaving",
    "TiedLMHead",
]
