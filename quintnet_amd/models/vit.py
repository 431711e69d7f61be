"""ViT for MNIST-scale classification (reference parity: utils/model.py).

Structure contract (consumed by the PP wrapper and the TP rewriter):
``model.embedding`` (patch + CLS + pos), ``model.blocks`` (ModuleList of
pre-norm transformer blocks), ``model.classification_head``.

MI355X-native notes: the patch-embedding Conv2d (kernel=stride=patch) is
implemented as the reshape+GEMM it actually is (SURVEY.md §2.4 — im2col
is free at stride==kernel), attention runs the shared fused-softmax
attention op, LayerNorms are the fused HIP kernel.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import attention as fused_attention
from ..ops import FusedLayerNorm

__all__ = [
    "PatchEmbedding",
    "ViTEmbedding",
    "Attention",
    "MLP",
    "TransformerBlock",
    "ClassificationHead",
    "Model",
    "VisionTransformer",
]


class PatchEmbedding(nn.Module):
    """[B,C,H,W] -> [B, N_patches, hidden] via reshape + linear.

    Equivalent to Conv2d(k=stride=patch) (reference utils/model.py:150-195)
    but expressed as the GEMM it is on MI355X.
    """

    def __init__(self, img_size=28, patch_size=4, in_channels=1, hidden_dim=64):
        super().__init__()
        assert img_size % patch_size == 0
        self.img_size = img_size
        self.patch_size = patch_size
        self.n_patches = (img_size // patch_size) ** 2
        self.proj = nn.Linear(in_channels * patch_size * patch_size, hidden_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, C, H, W = x.shape
        p = self.patch_size
        # [B,C,H/p,p,W/p,p] -> [B, (H/p)(W/p), C*p*p]
        x = x.reshape(B, C, H // p, p, W // p, p)
        x = x.permute(0, 2, 4, 1, 3, 5).reshape(B, self.n_patches, C * p * p)
        return self.proj(x)


class ViTEmbedding(nn.Module):
    """Patch embedding + CLS token + learned positional embedding."""

    def __init__(self, img_size=28, patch_size=4, in_channels=1, hidden_dim=64):
        super().__init__()
        self.patch_embed = PatchEmbedding(img_size, patch_size, in_channels, hidden_dim)
        self.cls_token = nn.Parameter(torch.zeros(1, 1, hidden_dim))
        self.pos_embed = nn.Parameter(
            torch.zeros(1, self.patch_embed.n_patches + 1, hidden_dim)
        )
        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)

    @property
    def seq_len(self) -> int:
        return self.patch_embed.n_patches + 1

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.patch_embed(x)
        cls = self.cls_token.expand(x.shape[0], -1, -1).to(x.dtype)
        x = torch.cat([cls, x], dim=1)
        return x + self.pos_embed.to(x.dtype)


class Attention(nn.Module):
    """Multi-head self-attention (non-causal) on the fused attention op."""

    def __init__(self, hidden_dim=64, n_heads=4):
        super().__init__()
        assert hidden_dim % n_heads == 0
        self.n_heads = n_heads
        self.head_dim = hidden_dim // n_heads
        self.q_proj = nn.Linear(hidden_dim, hidden_dim)
        self.k_proj = nn.Linear(hidden_dim, hidden_dim)
        self.v_proj = nn.Linear(hidden_dim, hidden_dim)
        self.out_proj = nn.Linear(hidden_dim, hidden_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, T, _ = x.shape
        # per-shard head count: TP replaces the projections with column
        # shards (gather_output=True keeps full hidden here)
        def split(t):
            return t.view(B, T, self.n_heads, -1).transpose(1, 2)

        q, k, v = split(self.q_proj(x)), split(self.k_proj(x)), split(self.v_proj(x))
        out = fused_attention(q, k, v, causal=False)
        out = out.transpose(1, 2).reshape(B, T, -1)
        return self.out_proj(out)


class MLP(nn.Module):
    def __init__(self, hidden_dim=64, expansion=4):
        super().__init__()
        self.fc1 = nn.Linear(hidden_dim, hidden_dim * expansion)
        self.relu = nn.ReLU()
        self.fc2 = nn.Linear(hidden_dim * expansion, hidden_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc2(self.relu(self.fc1(x)))


class TransformerBlock(nn.Module):
    """Pre-norm block (reference utils/model.py:197-233)."""

    def __init__(self, hidden_dim=64, n_heads=4):
        super().__init__()
        self.norm1 = FusedLayerNorm(hidden_dim)
        self.attn = Attention(hidden_dim, n_heads)
        self.norm2 = FusedLayerNorm(hidden_dim)
        self.mlp = MLP(hidden_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x + self.attn(self.norm1(x))
        x = x + self.mlp(self.norm2(x))
        return x


class ClassificationHead(nn.Module):
    """CLS token -> logits."""

    def __init__(self, hidden_dim=64, n_classes=10):
        super().__init__()
        self.norm = FusedLayerNorm(hidden_dim)
        self.fc = nn.Linear(hidden_dim, n_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc(self.norm(x[:, 0]))


class Model(nn.Module):
    """The ViT (reference utils/model.py:325-399): embedding + blocks + head."""

    def __init__(
        self,
        img_size=28,
        patch_size=4,
        in_channels=1,
        hidden_dim=64,
        n_heads=4,
        depth=8,
        n_classes=10,
        activation_checkpointing=False,
    ):
        super().__init__()
        self.embedding = ViTEmbedding(img_size, patch_size, in_channels, hidden_dim)
        self.blocks = nn.ModuleList(
            TransformerBlock(hidden_dim, n_heads) for _ in range(depth)
        )
        self.classification_head = ClassificationHead(hidden_dim, n_classes)
        self.hidden_dim = hidden_dim
        # recompute blocks in backward (same semantics as the GPT-2
        # path, models/gpt2/stage.py)
        self.activation_checkpointing = activation_checkpointing

    @property
    def seq_len(self) -> int:
        return self.embedding.seq_len

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.embedding(x)
        ckpt = (
            self.activation_checkpointing
            and self.training
            and torch.is_grad_enabled()
        )
        for blk in self.blocks:
            if ckpt:
                from torch.utils.checkpoint import checkpoint

                x = checkpoint(blk, x, use_reentrant=False)
            else:
                x = blk(x)
        return self.classification_head(x)


VisionTransformer = Model
