"""Padded-vocab layout (GPT2Config.vocab_pad_to) is EXACTLY the
unpadded model: pad logits columns are masked to -inf, pad embedding
rows stay zero and get zero gradient (models/gpt2/config.py,
models/gpt2/stage.py mask_pad_logits)."""

import torch

from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.ops import causal_lm_loss


def _mk(pad):
    torch.manual_seed(7)
    cfg = GPT2Config(
        n_embd=64, n_layer=2, n_head=2, vocab_size=300, n_positions=64,
        dropout=0.0, vocab_pad_to=pad,
    )
    torch.manual_seed(11)
    stage = GPT2Stage(cfg)
    return cfg, stage


def test_padded_matches_unpadded_exactly():
    cfg0, m0 = _mk(0)
    cfg1, m1 = _mk(128)
    assert cfg1.padded_vocab_size == 384
    # copy the unpadded weights into the padded model (pad rows zero)
    sd0, sd1 = m0.state_dict(), m1.state_dict()
    for k, v in sd0.items():
        if v.shape != sd1[k].shape:
            assert k == "embedding.wte.weight"
            sd1[k].zero_()
            sd1[k][: v.shape[0]] = v
        else:
            sd1[k] = v
    m1.load_state_dict(sd1)

    ids = torch.randint(0, 300, (2, 64))
    labels = torch.randint(0, 300, (2, 64))
    out0 = m0(ids)
    out1 = m1(ids)
    assert out1.shape[-1] == 384
    assert torch.equal(out0, out1[..., :300])
    assert bool(torch.isneginf(out1[..., 300:]).all())

    l0 = causal_lm_loss(out0, labels)
    l1 = causal_lm_loss(out1, labels)
    assert torch.equal(l0, l1)
    l0.backward()
    l1.backward()
    g0 = m0.embedding.wte.weight.grad
    g1 = m1.embedding.wte.weight.grad
    # grads equal up to BLAS blocking (the wgrad GEMM tiles differently
    # at N=384 vs N=300, reordering the fp32 reduction)
    assert torch.allclose(g0, g1[:300], rtol=1e-5, atol=1e-7)
    assert bool((g1[300:] == 0).all())
    for (k0, p0), (k1, p1) in zip(
        m0.named_parameters(), m1.named_parameters()
    ):
        if k0 == "embedding.wte.weight":
            continue
        assert torch.allclose(p0.grad, p1.grad, rtol=1e-5, atol=1e-7), k0


def test_generate_never_emits_pad_tokens():
    _, m = _mk(128)
    m.eval()
    ids = torch.randint(0, 300, (2, 8))
    out = m.generate(ids, max_new_tokens=6)
    assert int(out.max()) < 300
