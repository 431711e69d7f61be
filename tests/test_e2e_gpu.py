"""GPU end-to-end: training steps on the native kernel path reduce loss."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

DEV = torch.device("cuda", 0)


def test_gpt2_tiny_training_loss_decreases():
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import cross_entropy
    from quintnet_amd.optim import ZeroRedundancyAdamW

    torch.manual_seed(0)
    cfg = GPT2Config(n_embd=128, n_layer=2, n_head=2, vocab_size=512,
                     n_positions=128, dropout=0.0)
    stage = GPT2Stage(cfg, device=DEV, dtype=torch.bfloat16)
    opt = ZeroRedundancyAdamW(stage.parameters(), lr=3e-3)
    ids = torch.randint(0, 512, (4, 64), device=DEV)
    losses = []
    for _ in range(20):
        logits = stage(ids)
        loss = cross_entropy(logits[:, :-1], ids[:, 1:])
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.7, losses[:3] + losses[-3:]


def test_vit_training_loss_decreases():
    from quintnet_amd.models import Model

    torch.manual_seed(0)
    m = Model(hidden_dim=64, n_heads=4, depth=2).to(DEV, torch.bfloat16)
    opt = torch.optim.Adam(m.parameters(), lr=1e-3)
    x = torch.randn(16, 1, 28, 28, device=DEV, dtype=torch.bfloat16)
    y = torch.randint(0, 10, (16,), device=DEV)
    crit = torch.nn.CrossEntropyLoss()
    losses = []
    for _ in range(30):
        loss = crit(m(x).float(), y)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.7, losses[:3] + losses[-3:]


def test_gpt2_base_one_step_bf16():
    """One full GPT-2 124M step in bf16 — the bench configuration."""
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import cross_entropy
    from quintnet_amd.optim import ZeroRedundancyAdamW

    cfg = GPT2Config(dropout=0.0)
    stage = GPT2Stage(cfg, device=DEV, dtype=torch.bfloat16)
    opt = ZeroRedundancyAdamW(stage.parameters(), lr=1e-4)
    ids = torch.randint(0, cfg.vocab_size, (2, 1024), device=DEV)
    logits = stage(ids)
    loss = cross_entropy(logits[:, :-1], ids[:, 1:])
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
    assert 8.0 < float(loss) < 13.0  # ~ln(50257) at random init


@pytest.mark.gpu
def test_generate_kv_cache_gpu():
    """KV-cached generation on the GPU bf16 path matches full re-forward."""
    from quintnet_amd.models import GPT2Config, GPT2Stage

    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=256, n_positions=128, n_embd=128, n_layer=2,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      device=torch.device("cuda"), dtype=torch.bfloat16)
    stage.eval()
    ids = torch.randint(0, 256, (2, 9), device="cuda")
    with torch.no_grad():
        ref = ids
        for _ in range(8):
            logits = stage(ref)
            ref = torch.cat([ref, logits[:, -1].argmax(-1, keepdim=True)], dim=1)
    out = stage.generate(ids, max_new_tokens=8)
    # bf16 rounding can flip near-tie argmaxes between the two paths;
    # require the vast majority of tokens to agree
    agree = (out == ref).float().mean().item()
    assert agree > 0.9, (agree, out, ref)


@pytest.mark.gpu
def test_graph_decode_gpu():
    """StaticKVDecoder graph capture: token-exact vs its eager form."""
    import torch

    from quintnet_amd.models import GPT2Config, GPT2Stage, StaticKVDecoder

    torch.manual_seed(0)
    cfg = GPT2Config(n_embd=256, n_layer=3, n_head=4, vocab_size=512,
                     n_positions=128, dropout=0.0)
    stage = GPT2Stage(cfg, device=torch.device("cuda"),
                      dtype=torch.bfloat16).eval()
    ids = torch.randint(0, cfg.vocab_size, (2, 16), device="cuda")

    eager = StaticKVDecoder(stage, batch=2, max_len=128)
    cap = StaticKVDecoder.capture
    StaticKVDecoder.capture = lambda self: self  # keep _graph None
    try:
        want = eager.generate(ids, max_new_tokens=12)
    finally:
        StaticKVDecoder.capture = cap
    dec = StaticKVDecoder(stage, batch=2, max_len=128)
    have = dec.generate(ids, max_new_tokens=12)
    assert dec._graph is not None
    assert torch.equal(have, want)


def test_beam_and_speculative_on_gpu():
    """Serving extensions on real bf16 kernels: beam=1 == greedy, and
    greedy speculative == greedy (exactness survives the fused
    attention + library GEMM path)."""
    import torch

    from quintnet_amd.models import GPT2Config, GPT2Stage, beam_search
    from quintnet_amd.models.gpt2.speculative import speculative_generate

    torch.manual_seed(11)
    base = dict(vocab_size=128, n_positions=96, dropout=0.0)
    target = GPT2Stage(GPT2Config(n_embd=128, n_layer=2, n_head=2, **base),
                       device="cuda", dtype=torch.bfloat16).eval()
    draft = GPT2Stage(GPT2Config(n_embd=64, n_layer=1, n_head=2, **base),
                      device="cuda", dtype=torch.bfloat16).eval()
    ids = torch.randint(0, 128, (1, 8), device="cuda")
    want = target.generate(ids, max_new_tokens=8, temperature=0.0)
    assert torch.equal(
        beam_search(target, ids, max_new_tokens=8, num_beams=1), want
    )
    # speculative verify batches k+1 tokens through one GEMM while
    # generate steps one at a time — bf16 reduction order differs, so an
    # argmax NEAR-TIE may legitimately diverge; require agreement up to
    # the first divergence and a valid continuation after it
    spec = speculative_generate(target, draft, ids, max_new_tokens=8,
                                draft_k=3)
    assert spec.shape == want.shape and int(spec.max()) < 128
    n_match = 0
    for a, b in zip(spec[0, 8:].tolist(), want[0, 8:].tolist()):
        if a != b:
            break
        n_match += 1
    assert n_match >= 1, (spec, want)
    wide = beam_search(target, ids, max_new_tokens=8, num_beams=4)
    assert wide.shape == want.shape and int(wide.max()) < 128
