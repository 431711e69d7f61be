"""hipGraph-captured single-token decode with a STATIC KV cache.

The eager KV-cached decode path (stage.generate) launches ~100 kernels
per token at GPT-2-124M — at batch 1 that is ~4.4 ms/token of pure
launch overhead (measured r2, bench_decode).  MI355X-native fix: make
every per-token tensor static — preallocated [B, H, Tmax, D] caches, a
device-side position counter, masking by comparison against it — and
capture the whole token step (embed → 12 blocks → logits → argmax →
self-append) into ONE hipGraph; decoding then replays the graph once
per token.

Greedy only (sampling needs host RNG), pp == 1, tp == 1.
"""

from __future__ import annotations

from typing import Optional

import torch

from .config import GPT2Config
from .stage import GPT2Stage

__all__ = ["StaticKVDecoder"]


class StaticKVDecoder:
    def __init__(self, stage: GPT2Stage, batch: int, max_len: Optional[int] = None):
        assert stage.is_first_stage and stage.is_last_stage, "pp==1 only"
        assert stage.tp_group is None, "graph decode: tp==1 only"
        self.stage = stage
        cfg: GPT2Config = stage.config
        self.cfg = cfg
        self.B = batch
        self.Tmax = int(max_len or cfg.n_positions)
        p = next(stage.parameters())
        dev, dt = p.device, p.dtype
        H, D = cfg.n_head, cfg.head_dim
        self.H, self.D = H, D
        self.kc = [
            torch.zeros(batch, H, self.Tmax, D, device=dev, dtype=dt)
            for _ in stage.blocks
        ]
        self.vc = [torch.zeros_like(k) for k in self.kc]
        self.past = torch.zeros((), dtype=torch.long, device=dev)
        self.tok = torch.zeros(batch, 1, dtype=torch.long, device=dev)
        self._pos = torch.arange(self.Tmax, device=dev)
        self._graph = None
        self._dev = dev
        self._dt = dt

    # ------------------------------------------------------------------
    def _block_step(self, i: int, x: torch.Tensor, t0: torch.Tensor,
                    T: int) -> torch.Tensor:
        """One transformer block over x[B, T, E], cache write at device
        positions [t0, t0+T) (t0 is a 0-dim device tensor)."""
        blk = self.stage.blocks[i]
        B, H, D = self.B, self.H, self.D
        h = blk.ln_1(x)
        qkv = blk.attn.c_attn(h)
        hl = H * D

        def heads(t):
            return t.view(B, T, H, D).permute(0, 2, 1, 3)

        q = heads(qkv[:, :, :hl])
        k = heads(qkv[:, :, hl : 2 * hl])
        v = heads(qkv[:, :, 2 * hl :])
        idx = t0 + torch.arange(T, device=x.device)
        self.kc[i].index_copy_(2, idx, k)
        self.vc[i].index_copy_(2, idx, v)
        scale = 1.0 / (D ** 0.5)
        scores = torch.matmul(q, self.kc[i].transpose(-2, -1)).float() * scale
        # causal mask vs the device position: key j visible to query row r
        # iff j <= t0 + r
        qpos = (t0 + torch.arange(T, device=x.device)).view(1, 1, T, 1)
        visible = self._pos.view(1, 1, 1, -1) <= qpos
        scores = scores.masked_fill(~visible, float("-inf"))
        p = torch.softmax(scores, dim=-1).to(x.dtype)
        out = torch.matmul(p, self.vc[i])  # [B, H, T, D]
        out = out.permute(0, 2, 1, 3).reshape(B, T, H * D)
        x = x + blk.attn.resid_dropout(blk.attn.c_proj(out))
        x = x + blk.mlp(blk.ln_2(x))
        return x

    def _forward_tokens(self, ids: torch.Tensor, t0: torch.Tensor) -> torch.Tensor:
        """ids [B, T] at device position t0 → logits [B, vocab]."""
        st = self.stage
        B, T = ids.shape
        pos = t0 + torch.arange(T, device=ids.device)
        x = st.embedding.wte(ids) + st.embedding.wpe.weight.index_select(0, pos)
        for i in range(len(st.blocks)):
            x = self._block_step(i, x, t0, T)
        x = st.ln_f(x[:, -1:])
        w = st.embedding.wte.weight if st.lm_head is None else st.lm_head
        logits = torch.nn.functional.linear(x, w)[:, -1]
        from .stage import mask_pad_logits

        return mask_pad_logits(logits, st.config)

    # ------------------------------------------------------------------
    def _token_step(self) -> None:
        """The captured body: one token in (self.tok) → cache append →
        next token back into self.tok; device position advances."""
        logits = self._forward_tokens(self.tok, self.past)
        self.tok.copy_(logits.argmax(dim=-1, keepdim=True))
        self.past.add_(1)

    @torch.no_grad()
    def capture(self) -> "StaticKVDecoder":
        # no_grad: decode never needs autograd, and the padded-vocab
        # logits mask requires grad-free in-place (models/gpt2/stage.py)
        # warmup + capture advance past and clobber tok; restore both.
        # (the cache rows the warmup writes sit beyond `past` and are
        # never read until a real step rewrites them)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        saved_past = int(self.past)
        saved_tok = self.tok.clone()
        with torch.cuda.stream(side):
            for _ in range(2):
                self._token_step()
        torch.cuda.current_stream().wait_stream(side)
        self.past.fill_(saved_past)
        self.tok.copy_(saved_tok)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._token_step()
        self.past.fill_(saved_past)
        self.tok.copy_(saved_tok)
        return self

    # ------------------------------------------------------------------
    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32) -> torch.Tensor:
        """Greedy generation: eager prefill fills the static caches, then
        one graph replay per new token."""
        assert input_ids.shape[0] == self.B
        P = input_ids.shape[1]
        assert P + max_new_tokens <= self.Tmax
        was_training = self.stage.training
        self.stage.eval()
        try:
            for k in self.kc:
                k.zero_()
            for v in self.vc:
                v.zero_()
            self.past.zero_()
            logits = self._forward_tokens(input_ids.to(self._dev), self.past)
            self.past.fill_(P)
            self.tok.copy_(logits.argmax(dim=-1, keepdim=True))
            outs = [input_ids.to(self._dev), self.tok.clone()]
            if self._graph is None and torch.cuda.is_available():
                self.capture()
            for _ in range(max_new_tokens - 1):
                if self._graph is not None:
                    self._graph.replay()
                else:
                    self._token_step()
                outs.append(self.tok.clone())
            return torch.cat(outs, dim=1)
        finally:
            if was_training:
                self.stage.train()
