"""Generation metrics: ROUGE-1/2/L + BLEU (pure-Python, no extra deps)
and greedy generation (reference utils/metrics.py:12-206).
"""

from __future__ import annotations

import collections
import math
from typing import Dict, List

import torch

__all__ = ["rouge_n", "rouge_l", "bleu", "compute_generation_metrics", "generate_greedy", "count_parameters"]


def _ngrams(tokens: List[str], n: int):
    return collections.Counter(tuple(tokens[i : i + n]) for i in range(len(tokens) - n + 1))


def rouge_n(candidate: str, reference: str, n: int = 1) -> float:
    c, r = candidate.split(), reference.split()
    if len(c) < n or len(r) < n:
        return 0.0
    cg, rg = _ngrams(c, n), _ngrams(r, n)
    overlap = sum((cg & rg).values())
    rec = overlap / max(sum(rg.values()), 1)
    prec = overlap / max(sum(cg.values()), 1)
    return 0.0 if rec + prec == 0 else 2 * rec * prec / (rec + prec)


def _lcs(a: List[str], b: List[str]) -> int:
    dp = [0] * (len(b) + 1)
    for x in a:
        prev = 0
        for j, y in enumerate(b, 1):
            cur = dp[j]
            dp[j] = prev + 1 if x == y else max(dp[j], dp[j - 1])
            prev = cur
    return dp[-1]


def rouge_l(candidate: str, reference: str) -> float:
    c, r = candidate.split(), reference.split()
    if not c or not r:
        return 0.0
    l = _lcs(c, r)
    rec, prec = l / len(r), l / len(c)
    return 0.0 if rec + prec == 0 else 2 * rec * prec / (rec + prec)


def bleu(candidate: str, reference: str, max_n: int = 4) -> float:
    c, r = candidate.split(), reference.split()
    if not c:
        return 0.0
    logs = []
    for n in range(1, max_n + 1):
        cg, rg = _ngrams(c, n), _ngrams(r, n)
        overlap = sum((cg & rg).values())
        total = max(sum(cg.values()), 1)
        logs.append(math.log(max(overlap, 0.5) / total) if total else -9.0)
    bp = 1.0 if len(c) > len(r) else math.exp(1 - len(r) / max(len(c), 1))
    return bp * math.exp(sum(logs) / max_n)


def compute_generation_metrics(candidates: List[str], references: List[str]) -> Dict[str, float]:
    n = max(len(candidates), 1)
    return {
        "rouge1": sum(rouge_n(c, r, 1) for c, r in zip(candidates, references)) / n,
        "rouge2": sum(rouge_n(c, r, 2) for c, r in zip(candidates, references)) / n,
        "rougeL": sum(rouge_l(c, r) for c, r in zip(candidates, references)) / n,
        "bleu": sum(bleu(c, r) for c, r in zip(candidates, references)) / n,
    }


@torch.no_grad()
def generate_greedy(model, tokenizer, prompt: str, max_new_tokens: int, device) -> str:
    ids = tokenizer(prompt, return_tensors="pt")["input_ids"].to(device)
    if hasattr(model, "generate"):  # KV-cached serving path (GPT2Stage)
        out = model.generate(ids, max_new_tokens=max_new_tokens, temperature=0.0)
        if tokenizer.eos_token_id is not None:
            new = out[0, ids.shape[1]:]
            eos = (new == tokenizer.eos_token_id).nonzero()
            if eos.numel():
                out = out[:, : ids.shape[1] + int(eos[0]) + 1]
        return tokenizer.decode(out[0], skip_special_tokens=True)
    for _ in range(max_new_tokens):
        logits = model(ids)
        nxt = logits[:, -1, :].argmax(dim=-1, keepdim=True)
        ids = torch.cat([ids, nxt], dim=1)
        if tokenizer.eos_token_id is not None and int(nxt) == tokenizer.eos_token_id:
            break
    return tokenizer.decode(ids[0], skip_special_tokens=True)


def count_parameters(model) -> int:
    return sum(p.numel() for p in model.parameters() if p.requires_grad)
