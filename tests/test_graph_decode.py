"""StaticKVDecoder (graph-decode path): eager form must reproduce the
stage's own greedy KV-cached generation (CPU; the hipGraph capture is
exercised on GPU in test_e2e_gpu / bench_decode)."""

import torch


def test_static_decoder_matches_generate():
    from quintnet_amd.models import GPT2Config, GPT2Stage, StaticKVDecoder

    torch.manual_seed(0)
    cfg = GPT2Config(n_embd=64, n_layer=3, n_head=2, vocab_size=128,
                     n_positions=96, dropout=0.0)
    stage = GPT2Stage(cfg).eval()
    ids = torch.randint(0, cfg.vocab_size, (2, 12))
    want = stage.generate(ids, max_new_tokens=10, temperature=0.0)
    dec = StaticKVDecoder(stage, batch=2, max_len=96)
    have = dec.generate(ids, max_new_tokens=10)
    assert torch.equal(have, want), (have, want)


def test_static_decoder_two_calls_reset():
    """A second generate call must reset the caches/position."""
    from quintnet_amd.models import GPT2Config, GPT2Stage, StaticKVDecoder

    torch.manual_seed(1)
    cfg = GPT2Config(n_embd=64, n_layer=2, n_head=2, vocab_size=64,
                     n_positions=64, dropout=0.0)
    stage = GPT2Stage(cfg).eval()
    dec = StaticKVDecoder(stage, batch=1, max_len=64)
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    a = dec.generate(ids, max_new_tokens=6)
    b = dec.generate(ids, max_new_tokens=6)
    assert torch.equal(a, b)


def test_static_decoder_padded_vocab_matches_unpadded():
    """Padded-vocab config decodes the same tokens as the unpadded one
    (pad logits masked to -inf before every argmax)."""
    from quintnet_amd.models import GPT2Config, GPT2Stage, StaticKVDecoder

    base = dict(n_embd=64, n_layer=2, n_head=2, vocab_size=100,
                n_positions=64, dropout=0.0)
    torch.manual_seed(5)
    s0 = GPT2Stage(GPT2Config(**base)).eval()
    torch.manual_seed(5)
    s1 = GPT2Stage(GPT2Config(**base, vocab_pad_to=64)).eval()
    with torch.no_grad():
        sd0 = s0.state_dict()
        sd1 = s1.state_dict()
        for k, v in sd0.items():
            if sd1[k].shape != v.shape:
                sd1[k].zero_()
                sd1[k][: v.shape[0]] = v
            else:
                sd1[k] = v
        s1.load_state_dict(sd1)
    ids = torch.randint(0, 100, (2, 8))
    want = StaticKVDecoder(s0, batch=2, max_len=64).generate(ids, max_new_tokens=8)
    have = StaticKVDecoder(s1, batch=2, max_len=64).generate(ids, max_new_tokens=8)
    assert torch.equal(have, want)
