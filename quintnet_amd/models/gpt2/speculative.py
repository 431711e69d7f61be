"""Greedy speculative decoding: a small DRAFT model proposes ``k``
tokens autoregressively, the TARGET model scores all of them in ONE
cached forward, and the longest prefix where the target's own greedy
choice agrees is accepted (plus the target's next token — so every
verify step emits at least one token).  With greedy acceptance the
output is EXACTLY the target model's greedy decode
(tests/test_speculative.py), while the target runs ~(accepted+1)x
fewer forwards.  Serving extension beyond the reference's feature set;
pairs naturally with the graph decoder (models/gpt2/decode.py) for the
draft loop.

Cache discipline: each model's KV cache always covers exactly the
tokens it has forwarded; every forward feeds ``out[:, len(cache):]``
(catch-up + new input in one call), and rejected proposal entries are
sliced off afterwards.  ``past`` therefore always equals the cache
length, which keeps positions correct through any accept length
including full accepts.
"""

from __future__ import annotations

from typing import List

import torch

from ...ops import linear as fused_linear
from .stage import GPT2Stage, mask_pad_logits

__all__ = ["speculative_generate"]


def _new_caches(stage: GPT2Stage) -> List[dict]:
    return [{"k": None, "v": None, "k8": None, "quant": None}
            for _ in stage.blocks]


def _clen(caches) -> int:
    c = caches[0]
    return 0 if c["k"] is None else c["k"].shape[2]


def _cached_logits(stage: GPT2Stage, caches, ids) -> torch.Tensor:
    """ids [1, t] continuing the cache -> logits [1, t, V]."""
    past = _clen(caches)
    x = stage.embedding(ids, pos_offset=past)
    for blk, c in zip(stage.blocks, caches):
        x = blk.forward_cached(x, c)
    x = stage.ln_f(x)
    w = stage.embedding.wte.weight if stage.lm_head is None else stage.lm_head
    logits = fused_linear(x, w, None, None, prefer_library=True)
    return mask_pad_logits(logits, stage.config)


def _truncate(caches, T: int) -> None:
    for c in caches:
        if c["k"] is not None and c["k"].shape[2] > T:
            c["k"] = c["k"][:, :, :T]
            c["v"] = c["v"][:, :, :T]


@torch.no_grad()
def speculative_generate(
    target: GPT2Stage,
    draft: GPT2Stage,
    input_ids: torch.Tensor,
    max_new_tokens: int = 32,
    draft_k: int = 4,
    eos_token_id: int = None,
) -> torch.Tensor:
    """Greedy speculative decode; returns [1, T0 + <=max_new_tokens].

    ``target`` and ``draft`` must share the tokenizer (vocab); the draft
    is typically a much smaller model.  Output is token-identical to
    ``target.generate(..., temperature=0)``.
    """
    assert input_ids.shape[0] == 1, "speculative decoding is per-sequence"
    assert target.is_first_stage and target.is_last_stage, "pp==1 only"
    was_t, was_d = target.training, draft.training
    target.eval(), draft.eval()
    try:
        out = input_ids
        tc, dc = _new_caches(target), _new_caches(draft)
        limit = min(target.config.n_positions, draft.config.n_positions)
        # prefill: target's greedy next token seeds the loop
        nxt = _cached_logits(target, tc, out)[:, -1].argmax(dim=-1, keepdim=True)
        out = torch.cat([out, nxt], dim=1)
        while (
            out.shape[1] - input_ids.shape[1] < max_new_tokens
            and out.shape[1] < limit
            and not (eos_token_id is not None and int(nxt) == eos_token_id)
        ):
            budget = max_new_tokens - (out.shape[1] - input_ids.shape[1])
            k = min(draft_k, budget, limit - out.shape[1])
            # draft proposes k tokens greedily (first call also catches
            # its cache up on every token it has not forwarded yet)
            proposal = []
            d_in = out[:, _clen(dc):]
            for _ in range(k):
                d_in = _cached_logits(draft, dc, d_in)[:, -1].argmax(
                    dim=-1, keepdim=True
                )
                proposal.append(d_in)
            prop = torch.cat(proposal, dim=1)  # [1, k]
            # ONE target forward over its own catch-up + the proposal;
            # the last k+1 logits are the target's choices after the
            # last accepted token and after each proposed token
            t_in = torch.cat([out[:, _clen(tc):], prop], dim=1)
            tl = _cached_logits(target, tc, t_in)
            choice = tl[:, -(k + 1):].argmax(dim=-1)  # [1, k+1]
            n_acc = 0
            while n_acc < k and int(choice[0, n_acc]) == int(prop[0, n_acc]):
                n_acc += 1
            nxt = choice[:, n_acc : n_acc + 1]  # target's own next token
            out = torch.cat([out, prop[:, :n_acc], nxt], dim=1)
            # drop rejected proposal entries; the caches then cover at
            # most the emitted sequence minus the trailing token
            _truncate(tc, out.shape[1] - 1)
            _truncate(dc, out.shape[1] - 1)
            if eos_token_id is not None:
                new = out[0, -(n_acc + 1):]
                hit = (new == eos_token_id).nonzero()
                if hit.numel():
                    out = out[:, : out.shape[1] - (n_acc + 1) + int(hit[0]) + 1]
                    break
        return out[:, : input_ids.shape[1] + max_new_tokens]
    finally:
        if was_t:
            target.train()
        if was_d:
            draft.train()
