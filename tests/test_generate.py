"""KV-cached generation: equality with full re-forward greedy decoding."""

import torch


def _full_greedy(stage, input_ids, n):
    ids = input_ids
    for _ in range(n):
        logits = stage(ids)
        nxt = logits[:, -1].argmax(dim=-1, keepdim=True)
        ids = torch.cat([ids, nxt], dim=1)
    return ids


def test_kv_cache_greedy_matches_full_forward():
    from quintnet_amd.models import GPT2Config, GPT2Stage

    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=96, n_positions=64, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    stage.eval()
    ids = torch.randint(0, 96, (2, 7))
    with torch.no_grad():
        ref = _full_greedy(stage, ids, 12)
    out = stage.generate(ids, max_new_tokens=12, temperature=0.0)
    assert torch.equal(out, ref), (out, ref)


def test_generate_sampling_shapes_and_determinism():
    from quintnet_amd.models import GPT2Config, GPT2Stage

    torch.manual_seed(1)
    cfg = GPT2Config(vocab_size=64, n_positions=48, n_embd=16, n_layer=1,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    ids = torch.randint(0, 64, (1, 5))
    a = stage.generate(ids, max_new_tokens=8, temperature=0.8, top_k=10)
    b = stage.generate(ids, max_new_tokens=8, temperature=0.8, top_k=10)
    assert a.shape == (1, 13)
    assert torch.equal(a, b)  # tokens-seeded sampling is deterministic
    # respects the context window
    long_ids = torch.randint(0, 64, (1, 40))
    c = stage.generate(long_ids, max_new_tokens=30)
    assert c.shape[1] <= 48 + 1


def test_int8_kv_cache_close_to_fp():
    """int8 KV cache: halves cache memory; next-token logits stay close
    and a full generation completes."""
    from quintnet_amd.models import GPT2Config, GPT2Stage

    torch.manual_seed(3)
    cfg = GPT2Config(vocab_size=96, n_positions=64, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    stage.eval()
    ids = torch.randint(0, 96, (2, 12))
    a = stage.generate(ids, max_new_tokens=10)
    b = stage.generate(ids, max_new_tokens=10, cache_dtype="int8")
    assert b.shape == a.shape
    agree = (a == b).float().mean().item()
    assert agree > 0.8, (agree, a, b)  # int8 rounding may flip near-ties


def test_generate_top_p_masks_tail():
    """top_p keeps the argmax and excludes the improbable tail: with a
    tight nucleus the sample must come from the head of the dist."""
    import torch

    from quintnet_amd.models import GPT2Config, GPT2Stage

    torch.manual_seed(2)
    m = GPT2Stage(GPT2Config(n_embd=32, n_layer=1, n_head=2, vocab_size=64,
                             n_positions=32, dropout=0.0)).eval()
    ids = torch.randint(0, 64, (1, 8))
    out = m.generate(ids, max_new_tokens=6, temperature=0.7, top_p=0.05)
    assert out.shape[1] == 14
    # tiny nucleus ~= greedy: the same call with temperature 0
    want = m.generate(ids, max_new_tokens=6, temperature=0.0)
    # p=0.05 typically keeps only the top token; allow equality check on
    # the first generated token at least
    assert int(out[0, 8]) == int(want[0, 8])


def test_repetition_penalty_reduces_repeats():
    """A strong penalty must yield fewer immediate repeats than greedy
    (random-init GPT-2 collapses to one token greedily)."""
    import torch

    from quintnet_amd.models import GPT2Config, GPT2Stage

    torch.manual_seed(4)
    m = GPT2Stage(GPT2Config(n_embd=32, n_layer=1, n_head=2, vocab_size=64,
                             n_positions=64, dropout=0.0)).eval()
    ids = torch.randint(0, 64, (1, 6))

    def repeats(seq):
        new = seq[0, 6:]
        return int((new[1:] == new[:-1]).sum())

    greedy = m.generate(ids, max_new_tokens=12, temperature=0.0)
    pen = m.generate(ids, max_new_tokens=12, temperature=1e-4,
                     repetition_penalty=5.0)
    assert repeats(pen) < max(repeats(greedy), 1)


def _tp2_generate_rank_consistent(rank, world):
    """TP-sharded generate: every TP rank must emit the SAME tokens,
    including under temperature sampling (rank-consistent RNG) and the
    new top-p / repetition-penalty transforms."""
    import torch
    import torch.distributed as dist

    from quintnet_amd import init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage

    pg = init_process_groups("cpu", [world], ["tp"])
    torch.manual_seed(14)
    cfg = GPT2Config(vocab_size=96, n_positions=64, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=pg.get_group("tp"))
    for p in stage.parameters():
        dist.broadcast(p.data, src=0)
    stage.eval()
    ids = torch.randint(0, 96, (1, 6))
    dist.broadcast(ids, src=0)
    out = stage.generate(ids, max_new_tokens=10, temperature=0.8,
                         top_k=20, top_p=0.9, repetition_penalty=1.3)
    ref = out.clone()
    dist.broadcast(ref, src=0)
    assert torch.equal(out, ref), "TP ranks diverged during sampling"


def test_tp2_generate_rank_consistent():
    from conftest import run_distributed

    run_distributed(_tp2_generate_rank_consistent, 2)
