"""Beam-search decoding over the KV-cached stage (serving extension).

The beams ride the CACHE's batch dimension: one prefill, then the
per-step forward runs all beams as a batch and cache rows are reordered
with the surviving beams (`index_select` on dim 0) — no re-prefill, no
per-beam python loops in the hot path.  `num_beams=1` reduces exactly
to greedy decoding (tests/test_beam.py)."""

from __future__ import annotations

from typing import List, Optional

import torch

from ...ops import linear as fused_linear
from .stage import GPT2Stage, mask_pad_logits

__all__ = ["beam_search"]


def _step_logits(stage: GPT2Stage, caches, ids: torch.Tensor, past: int):
    x = stage.embedding(ids, pos_offset=past)
    for blk, c in zip(stage.blocks, caches):
        x = blk.forward_cached(x, c)
    x = stage.ln_f(x[:, -1:])
    w = stage.embedding.wte.weight if stage.lm_head is None else stage.lm_head
    logits = fused_linear(x, w, None, None, prefer_library=True)[:, -1]
    return mask_pad_logits(logits, stage.config)


def _reorder(caches, idx: torch.Tensor) -> None:
    for c in caches:
        if c["k"] is not None:
            c["k"] = c["k"].index_select(0, idx)
            c["v"] = c["v"].index_select(0, idx)


@torch.no_grad()
def beam_search(
    stage: GPT2Stage,
    input_ids: torch.Tensor,
    max_new_tokens: int = 32,
    num_beams: int = 4,
    length_penalty: float = 1.0,
    eos_token_id: Optional[int] = None,
) -> torch.Tensor:
    """Highest-scoring sequence under sum-logprob / len**length_penalty.

    Single-prompt interface ([1, T] in, [1, T+new] out), beams batched
    internally.  Finished hypotheses (eos) leave the active set; search
    stops when no active beam can beat the best finished score.
    """
    assert input_ids.shape[0] == 1, "beam_search is per-prompt"
    assert stage.is_first_stage and stage.is_last_stage, "pp==1 only"
    was_training = stage.training
    stage.eval()
    try:
        dev = input_ids.device
        T0 = input_ids.shape[1]
        limit = min(stage.config.n_positions - T0, max_new_tokens)
        caches = [{"k": None, "v": None, "k8": None, "quant": None}
                  for _ in stage.blocks]
        logits = _step_logits(stage, caches, input_ids, 0)  # [1, V]
        logprobs = torch.log_softmax(logits.float(), dim=-1)[0]
        # the active set starts at min(beams, |finite vocab|) — only V
        # distinct one-token prefixes exist — and grows back toward
        # num_beams as candidates multiply (found by the exhaustive-
        # optimum property test: topk(num_beams) with num_beams > V
        # both crashed and silently capped the width)
        k0 = min(num_beams, int(torch.isfinite(logprobs).sum()))
        scores, toks = logprobs.topk(k0)  # [k0]
        for c in caches:
            c["k"] = c["k"].expand(k0, -1, -1, -1).contiguous()
            c["v"] = c["v"].expand(k0, -1, -1, -1).contiguous()
        seqs = torch.cat(
            [input_ids.expand(k0, -1), toks.unsqueeze(1)], dim=1
        )
        finished: List[tuple] = []  # (score/penalty, tensor)

        def fin_score(s, length):
            return float(s) / max(length, 1) ** length_penalty

        if eos_token_id is not None:
            for b in range(k0):
                if int(toks[b]) == eos_token_id:
                    finished.append((fin_score(scores[b], 1), seqs[b]))
                    scores[b] = float("-inf")

        for step in range(1, limit):
            if bool(torch.isinf(scores).all()):
                break
            logits = _step_logits(stage, caches, seqs[:, -1:], seqs.shape[1] - 1)
            lp = torch.log_softmax(logits.float(), dim=-1)  # [beams, V]
            V = lp.shape[-1]
            total = scores.unsqueeze(1) + lp  # [-inf rows drop out]
            # 2*beams candidates so eos hits don't starve the active set
            k_cand = min(2 * num_beams, total.numel())
            cand_scores, flat = total.reshape(-1).topk(k_cand)
            beam_idx = flat // V
            tok_idx = flat % V
            new_scores, new_beams, new_toks = [], [], []
            for cs, bi, ti in zip(cand_scores, beam_idx, tok_idx):
                if eos_token_id is not None and int(ti) == eos_token_id:
                    finished.append((
                        fin_score(cs, step + 1),
                        torch.cat([seqs[bi], ti.view(1)]),
                    ))
                    continue
                new_scores.append(cs)
                new_beams.append(bi)
                new_toks.append(ti)
                if len(new_scores) == num_beams:
                    break
            if not new_scores:
                break
            while len(new_scores) < num_beams:  # all-eos corner: pad dead beams
                new_scores.append(torch.tensor(float("-inf"), device=dev))
                new_beams.append(new_beams[0])
                new_toks.append(new_toks[0])
            scores = torch.stack(new_scores)
            bsel = torch.stack(new_beams)
            seqs = torch.cat(
                [seqs.index_select(0, bsel),
                 torch.stack(new_toks).unsqueeze(1)], dim=1
            )
            _reorder(caches, bsel)
            if finished:
                best_fin = max(f[0] for f in finished)
                # best possible for an active beam: its raw score with the
                # most favorable remaining length under the penalty
                best_active = max(
                    fin_score(s, step + 1) for s in scores.tolist()
                )
                if len(finished) >= num_beams and best_fin >= best_active:
                    break

        for b in range(scores.shape[0]):
            if not bool(torch.isinf(scores[b])):
                finished.append(
                    (fin_score(scores[b], seqs.shape[1] - T0), seqs[b])
                )
        best = max(finished, key=lambda f: f[0])
        return best[1].unsqueeze(0)
    finally:
        if was_training:
            stage.train()
