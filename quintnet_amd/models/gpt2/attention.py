"""GPT-2 causal self-attention with Megatron TP sharding.

Reference parity: utils/GPT2/gpt2_attention.py:24-181.  c_attn is a
column-parallel fused QKV projection (gather_output=False) and c_proj a
row-parallel projection whose all-reduce is the block's only TP
collective on the attention path.  Heads are sharded: each TP rank runs
n_head/tp heads through the fused attention op.

Local QKV layout note: the rank-local c_attn weight is
``[q_local; k_local; v_local]`` (each ``n_embd/tp`` rows) — i.e. Q/K/V
are sharded per-head *separately*, not a naive contiguous slice of the
[3*n_embd] output (which would put whole Q on rank 0 at tp=2).  The
staged checkpoint loader and the merge CLI both honor this layout.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import FusedDropout, attention_qkv
from ...parallel.tensor_parallel import ColumnParallelLinear, RowParallelLinear
from .config import GPT2Config

__all__ = ["GPT2Attention"]


class GPT2Attention(nn.Module):
    def __init__(self, config: GPT2Config, tp_group=None, device=None, dtype=None,
                 cp_group=None):
        super().__init__()
        self.config = config
        self.cp_group = cp_group
        sp = config.sequence_parallel
        self.c_attn = ColumnParallelLinear(
            config.n_embd,
            3 * config.n_embd,
            tp_group=tp_group,
            gather_output=False,
            sequence_parallel=sp,
            device=device,
            dtype=dtype,
        )
        self.c_proj = RowParallelLinear(
            config.n_embd,
            config.n_embd,
            tp_group=tp_group,
            input_is_parallel=True,
            sequence_parallel=sp,
            device=device,
            dtype=dtype,
        )
        self.c_attn.fp8 = self.c_proj.fp8 = config.fp8
        self.tp_size = self.c_attn.tp_size
        assert config.n_head % self.tp_size == 0, "n_head must divide by tp"
        self.n_head_local = config.n_head // self.tp_size
        self.hidden_local = config.n_embd // self.tp_size
        self.head_dim = config.head_dim
        self.resid_dropout = FusedDropout(config.dropout)

    def forward(self, x: torch.Tensor, kv_cache=None) -> torch.Tensor:
        qkv = self.c_attn(x)  # [B, T, 3*n_embd/tp] = [q_loc | k_loc | v_loc]
        if kv_cache is not None:
            out = self._forward_cached(qkv, kv_cache)
        elif self.cp_group is not None:
            out = self._forward_cp(qkv)
        else:
            out = attention_qkv(qkv, self.n_head_local, causal=True)
        return self.resid_dropout(self.c_proj(out))

    def _forward_cp(self, qkv: torch.Tensor) -> torch.Tensor:
        """Context parallelism: x is this rank's sequence shard; attention
        runs against the full sequence — all-gathered K/V by default, or
        the ring exchange (QN_CP_RING=1 / config.cp_ring) when the
        gathered KV itself is the memory bound."""
        import os

        from ...parallel.context_parallel import (
            context_parallel_attention,
            ring_attention,
            zigzag_ring_attention,
        )

        B, T, _ = qkv.shape
        H, D = self.n_head_local, self.head_dim
        hl = self.hidden_local

        def heads(t):
            return t.reshape(B, T, H, D).permute(0, 2, 1, 3)

        q = heads(qkv[:, :, :hl])
        k = heads(qkv[:, :, hl : 2 * hl])
        v = heads(qkv[:, :, 2 * hl :])
        if getattr(self.config, "cp_zigzag", False):
            fn = zigzag_ring_attention
        else:
            use_ring = getattr(self.config, "cp_ring", False) or os.environ.get(
                "QN_CP_RING"
            ) == "1"
            fn = ring_attention if use_ring else context_parallel_attention
        out = fn(q, k, v, self.cp_group, causal=True)
        return out.permute(0, 2, 1, 3).reshape(B, T, hl)

    def _forward_cached(self, qkv: torch.Tensor, kv_cache: dict) -> torch.Tensor:
        """Incremental decode: append this step's K/V to the cache and
        attend the new queries against the whole cache (causal diagonal
        shifted by the cache length — the same q_offset machinery the
        context-parallel path uses).  ``kv_cache['quant'] == 'int8'``
        stores the cache as per-token-per-head symmetric int8 (+fp scale),
        halving KV memory for long-context serving."""
        from ...ops.attention import attention

        B, T, _ = qkv.shape
        H, D = self.n_head_local, self.head_dim
        hl = self.hidden_local

        def heads(t):
            return t.view(B, T, H, D).permute(0, 2, 1, 3)

        q = heads(qkv[:, :, :hl])
        k = heads(qkv[:, :, hl : 2 * hl])
        v = heads(qkv[:, :, 2 * hl :])
        if kv_cache.get("quant") == "int8":
            kf, vf = self._append_int8(kv_cache, k, v)
        else:
            if kv_cache.get("k") is None:
                kv_cache["k"], kv_cache["v"] = k, v
            else:
                kv_cache["k"] = torch.cat([kv_cache["k"], k], dim=2)
                kv_cache["v"] = torch.cat([kv_cache["v"], v], dim=2)
            kf, vf = kv_cache["k"], kv_cache["v"]
        past = kf.shape[2] - T
        out = attention(q, kf, vf, causal=True, q_offset=past)
        return out.permute(0, 2, 1, 3).reshape(B, T, hl)

    @staticmethod
    def _append_int8(kv_cache: dict, k: torch.Tensor, v: torch.Tensor):
        def quant(t):
            s = t.abs().amax(dim=-1, keepdim=True).float().clamp(min=1e-8) / 127.0
            return (t.float() / s).round().clamp(-127, 127).to(torch.int8), s

        k8, ks = quant(k)
        v8, vs = quant(v)
        if kv_cache.get("k8") is None:
            kv_cache.update(k8=k8, ks=ks, v8=v8, vs=vs)
        else:
            kv_cache["k8"] = torch.cat([kv_cache["k8"], k8], dim=2)
            kv_cache["ks"] = torch.cat([kv_cache["ks"], ks], dim=2)
            kv_cache["v8"] = torch.cat([kv_cache["v8"], v8], dim=2)
            kv_cache["vs"] = torch.cat([kv_cache["vs"], vs], dim=2)
        kf = (kv_cache["k8"].float() * kv_cache["ks"]).to(k.dtype)
        vf = (kv_cache["v8"].float() * kv_cache["vs"]).to(v.dtype)
        return kf, vf
