"""Beam search (models/gpt2/beam.py): beam=1 reduces exactly to greedy;
wider beams never score worse under the model; eos and padded-vocab
behavior."""

import torch

from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.models.gpt2.beam import beam_search


def _mk(vocab=96, pad=0):
    torch.manual_seed(9)
    return GPT2Stage(GPT2Config(
        n_embd=64, n_layer=2, n_head=2, vocab_size=vocab, n_positions=96,
        dropout=0.0, vocab_pad_to=pad,
    )).eval()


def _seq_logprob(stage, seq, t0):
    """Sum log p(token | prefix) for tokens t0..end under the model."""
    with torch.no_grad():
        logits = stage(seq)[0, :-1].float()
    lp = torch.log_softmax(logits, dim=-1)
    tgt = seq[0, 1:]
    total = 0.0
    for i in range(t0 - 1, seq.shape[1] - 1):
        total += float(lp[i, tgt[i]])
    return total


def test_beam1_equals_greedy():
    m = _mk()
    ids = torch.randint(0, 96, (1, 8))
    want = m.generate(ids, max_new_tokens=10, temperature=0.0)
    have = beam_search(m, ids, max_new_tokens=10, num_beams=1)
    assert torch.equal(have, want), (have, want)


def test_wider_beam_never_scores_worse():
    m = _mk()
    ids = torch.randint(0, 96, (1, 6))
    n = 8
    greedy = m.generate(ids, max_new_tokens=n, temperature=0.0)
    beam = beam_search(m, ids, max_new_tokens=n, num_beams=4,
                       length_penalty=1.0)
    # both full-length (no eos): beam's sum-logprob must be >= greedy's
    assert beam.shape == greedy.shape
    sg = _seq_logprob(m, greedy, ids.shape[1])
    sb = _seq_logprob(m, beam, ids.shape[1])
    assert sb >= sg - 1e-4, (sb, sg)


def test_beam_eos_stops_hypothesis():
    m = _mk()
    ids = torch.randint(0, 96, (1, 6))
    ref = beam_search(m, ids, max_new_tokens=12, num_beams=3)
    eos = int(ref[0, ids.shape[1] + 2])
    out = beam_search(m, ids, max_new_tokens=12, num_beams=3,
                      eos_token_id=eos)
    new = out[0, ids.shape[1]:]
    hits = (new == eos).nonzero()
    if hits.numel():  # if the winner contains eos it must be terminal
        assert int(hits[0]) == new.shape[0] - 1


def test_beam_padded_vocab_never_emits_pad():
    m = _mk(vocab=100, pad=64)
    ids = torch.randint(0, 100, (1, 6))
    out = beam_search(m, ids, max_new_tokens=8, num_beams=4)
    assert int(out.max()) < 100


def test_wide_beam_equals_exhaustive_optimum():
    """With num_beams >= V^(L-1) nothing can be pruned, so beam search
    must return the EXACT argmax over all V^L continuations."""
    import itertools

    torch.manual_seed(31)
    V, L = 5, 3
    m = GPT2Stage(GPT2Config(
        n_embd=32, n_layer=1, n_head=2, vocab_size=V, n_positions=32,
        dropout=0.0,
    )).eval()
    ids = torch.randint(0, V, (1, 4))

    def seq_score(cont):
        seq = torch.cat([ids, torch.tensor([list(cont)])], dim=1)
        with torch.no_grad():
            lp = torch.log_softmax(m(seq)[0, :-1].float(), dim=-1)
        s = 0.0
        for i, t in enumerate(cont):
            s += float(lp[ids.shape[1] - 1 + i, t])
        return s

    best = max(itertools.product(range(V), repeat=L), key=seq_score)
    out = beam_search(m, ids, max_new_tokens=L, num_beams=V * V)
    assert tuple(out[0, 4:].tolist()) == best, (out, best)


def _tp2_beam(rank, world):
    import torch.distributed as dist

    from quintnet_amd import init_process_groups

    pg = init_process_groups("cpu", [world], ["tp"])
    torch.manual_seed(15)
    cfg = GPT2Config(vocab_size=96, n_positions=64, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=pg.get_group("tp"))
    for p in stage.parameters():
        dist.broadcast(p.data, src=0)
    stage.eval()
    ids = torch.randint(0, 96, (1, 6))
    dist.broadcast(ids, src=0)
    out = beam_search(stage, ids, max_new_tokens=8, num_beams=3)
    ref = out.clone()
    dist.broadcast(ref, src=0)
    assert torch.equal(out, ref), "TP ranks diverged in beam search"


def test_tp2_beam_rank_consistent():
    from conftest import run_distributed

    run_distributed(_tp2_beam, 2)
