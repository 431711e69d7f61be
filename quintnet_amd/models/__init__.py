from .vit import (
    Model,
    VisionTransformer,
    PatchEmbedding,
    ViTEmbedding,
    Attention,
    MLP,
    TransformerBlock,
    ClassificationHead,
)
from .gpt2 import (
    StaticKVDecoder,
    GPT2Config,
    GPT2Embedding,
    GPT2Attention,
    GPT2MLP,
    GPT2Block,
    GPT2ForInterleaving, GPT2Stage,
)

__all__ = [
    "Model",
    "VisionTransformer",
    "PatchEmbedding",
    "ViTEmbedding",
    "Attention",
    "MLP",
    "TransformerBlock",
    "ClassificationHead",
    "GPT2Config",
    "GPT2Embedding",
    "GPT2Attention",
    "GPT2MLP",
    "GPT2Block",
    "GPT2ForInterleaving",
    "GPT2Stage",
    "StaticKVDecoder",
]
