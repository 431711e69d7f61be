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


def _dist(logits_row: torch.Tensor, temperature: float, top_k: int,
          top_p: float) -> torch.Tensor:
    """logits [V] -> probability vector under the sampling transform."""
    x = logits_row.float() / max(temperature, 1e-5)
    if top_k and top_k > 0:
        kth = torch.topk(x, min(int(top_k), x.shape[-1])).values[-1]
        x = x.masked_fill(x < kth, float("-inf"))
    if top_p and 0.0 < top_p < 1.0:
        sl, si = torch.sort(x, descending=True)
        probs = torch.softmax(sl, dim=-1)
        drop_sorted = probs.cumsum(-1) - probs >= top_p
        drop_sorted[..., 0] = False  # fp32-underflow guard: argmax kept
        drop = torch.zeros_like(drop_sorted).scatter(-1, si, drop_sorted)
        x = x.masked_fill(drop, float("-inf"))
    return torch.softmax(x, dim=-1)


def _spec_accept(p: torch.Tensor, q: torch.Tensor, tok: int,
                 gen: torch.Generator):
    """Speculative-sampling acceptance (Leviathan et al.): given target
    dist p, draft dist q and a token sampled from q, accept with prob
    min(1, p/q); on rejection return a token from norm(max(0, p - q)).
    The emitted token is then distributed EXACTLY as p
    (tests/test_speculative.py::test_accept_resample_lemma)."""
    pq = float(p[tok]) / max(float(q[tok]), 1e-20)
    u = float(torch.rand((), generator=gen))
    if u < pq:
        return True, tok
    resid = (p - q).clamp_min(0)
    tot = float(resid.sum())
    if tot <= 0:  # p <= q everywhere can only happen with p == q
        return True, tok
    new = int(torch.multinomial(resid / tot, 1, generator=gen))
    return False, new


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
    temperature: float = 0.0,
    top_k: int = 0,
    top_p: float = 0.0,
    seed: int = None,
) -> torch.Tensor:
    """Speculative decode; returns [1, T0 + <=max_new_tokens].

    ``target`` and ``draft`` must share the tokenizer (vocab); the draft
    is typically a much smaller model.  temperature==0: greedy
    acceptance, token-identical to ``target.generate(temperature=0)``.
    temperature>0: stochastic speculative sampling — each emitted token
    is distributed exactly as a sample from the TARGET's (temperature/
    top-k/top-p transformed) distribution (_spec_accept lemma).
    """
    assert input_ids.shape[0] == 1, "speculative decoding is per-sequence"
    assert target.is_first_stage and target.is_last_stage, "pp==1 only"
    was_t, was_d = target.training, draft.training
    target.eval(), draft.eval()
    try:
        out = input_ids
        tc, dc = _new_caches(target), _new_caches(draft)
        limit = min(target.config.n_positions, draft.config.n_positions)
        gen = torch.Generator()
        gen.manual_seed(seed if seed is not None else 0x5eed)
        # prefill: the target's next token seeds the loop
        pre = _cached_logits(target, tc, out)[:, -1]
        if temperature and temperature > 0:
            p0 = _dist(pre[0], temperature, top_k, top_p)
            nxt = out.new_tensor([[int(torch.multinomial(p0, 1, generator=gen))]])
        else:
            nxt = pre.argmax(dim=-1, keepdim=True)
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
            q_dists = []
            d_in = out[:, _clen(dc):]
            for _ in range(k):
                dl = _cached_logits(draft, dc, d_in)[:, -1]
                if temperature and temperature > 0:
                    q = _dist(dl[0], temperature, top_k, top_p)
                    q_dists.append(q)
                    d_in = out.new_tensor(
                        [[int(torch.multinomial(q, 1, generator=gen))]]
                    )
                else:
                    d_in = dl.argmax(dim=-1, keepdim=True)
                proposal.append(d_in)
            prop = torch.cat(proposal, dim=1)  # [1, k]
            # ONE target forward over its own catch-up + the proposal;
            # the last k+1 logits are the target's choices after the
            # last accepted token and after each proposed token
            t_in = torch.cat([out[:, _clen(tc):], prop], dim=1)
            tl = _cached_logits(target, tc, t_in)
            if temperature and temperature > 0:
                # stochastic acceptance; q dists were stashed by the
                # proposal loop above
                n_acc, forced = 0, None
                for i in range(k):
                    p_i = _dist(tl[0, -(k + 1) + i], temperature, top_k, top_p)
                    ok, tok2 = _spec_accept(p_i, q_dists[i],
                                            int(prop[0, i]), gen)
                    if ok:
                        n_acc += 1
                    else:
                        forced = tok2
                        break
                if forced is None:
                    p_last = _dist(tl[0, -1], temperature, top_k, top_p)
                    forced = int(torch.multinomial(p_last, 1, generator=gen))
                nxt = out.new_tensor([[forced]])
            else:
                choice = tl[:, -(k + 1):].argmax(dim=-1)  # [1, k+1]
                n_acc = 0
                while n_acc < k and int(choice[0, n_acc]) == int(prop[0, n_acc]):
                    n_acc += 1
                nxt = choice[:, n_acc : n_acc + 1]  # target's next token
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
