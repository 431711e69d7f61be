"""Greedy speculative decoding must be TOKEN-IDENTICAL to the target
model's plain greedy decode (models/gpt2/speculative.py)."""

import pytest
import torch

from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.models.gpt2.speculative import speculative_generate


def _models(vocab=96, pad=0):
    torch.manual_seed(11)
    target = GPT2Stage(GPT2Config(
        n_embd=64, n_layer=3, n_head=2, vocab_size=vocab, n_positions=128,
        dropout=0.0, vocab_pad_to=pad,
    )).eval()
    torch.manual_seed(99)
    draft = GPT2Stage(GPT2Config(
        n_embd=32, n_layer=1, n_head=2, vocab_size=vocab, n_positions=128,
        dropout=0.0, vocab_pad_to=pad,
    )).eval()
    return target, draft


@pytest.mark.parametrize("draft_k", [1, 3, 4, 8])
def test_speculative_matches_greedy(draft_k):
    target, draft = _models()
    ids = torch.randint(0, 96, (1, 10))
    want = target.generate(ids, max_new_tokens=16, temperature=0.0)
    have = speculative_generate(target, draft, ids, max_new_tokens=16,
                                draft_k=draft_k)
    assert torch.equal(have, want), (have, want)


def test_speculative_draft_equals_target_full_accepts():
    """Draft == target -> every proposal accepted; still exact."""
    target, _ = _models()
    ids = torch.randint(0, 96, (1, 8))
    want = target.generate(ids, max_new_tokens=12, temperature=0.0)
    have = speculative_generate(target, target, ids, max_new_tokens=12,
                                draft_k=4)
    assert torch.equal(have, want)


def test_speculative_padded_vocab():
    target, draft = _models(vocab=100, pad=64)
    ids = torch.randint(0, 100, (1, 6))
    want = target.generate(ids, max_new_tokens=10, temperature=0.0)
    have = speculative_generate(target, draft, ids, max_new_tokens=10,
                                draft_k=3)
    assert torch.equal(have, want)
    assert int(have.max()) < 100


def test_speculative_eos_stop():
    target, draft = _models()
    ids = torch.randint(0, 96, (1, 6))
    ref = target.generate(ids, max_new_tokens=20, temperature=0.0)
    eos = int(ref[0, ids.shape[1] + 4])  # force a stop 5 tokens in
    want = target.generate(ids, max_new_tokens=20, temperature=0.0,
                           eos_token_id=eos)
    have = speculative_generate(target, draft, ids, max_new_tokens=20,
                                draft_k=3, eos_token_id=eos)
    assert torch.equal(have, want), (have, want)


def test_accept_resample_lemma():
    """The speculative acceptance rule must emit tokens distributed
    EXACTLY as the target dist p, regardless of the draft dist q
    (Leviathan et al. correctness lemma) — checked empirically."""
    from quintnet_amd.models.gpt2.speculative import _spec_accept

    p = torch.tensor([0.5, 0.3, 0.15, 0.05])
    q = torch.tensor([0.1, 0.2, 0.3, 0.4])  # deliberately mismatched
    gen = torch.Generator().manual_seed(123)
    n = 40000
    toks = torch.multinomial(q.expand(n, -1), 1, generator=gen).squeeze(1)
    counts = torch.zeros(4)
    for i in range(n):
        _, tok = _spec_accept(p, q, int(toks[i]), gen)
        counts[tok] += 1
    emp = counts / n
    assert (emp - p).abs().max() < 0.015, emp


def test_stochastic_speculative_runs_and_is_deterministic():
    target, draft = _models()
    ids = torch.randint(0, 96, (1, 8))
    a = speculative_generate(target, draft, ids, max_new_tokens=12,
                             draft_k=3, temperature=0.8, top_p=0.9, seed=7)
    b = speculative_generate(target, draft, ids, max_new_tokens=12,
                             draft_k=3, temperature=0.8, top_p=0.9, seed=7)
    assert torch.equal(a, b)
    assert a.shape[1] == 20 and int(a.max()) < 96
