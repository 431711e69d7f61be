"""Op-layer consistency vs plain PyTorch (CPU reference path).

The same oracles, marked gpu, validate the HIP kernels in
test_ops_gpu.py.
"""

import math

import pytest
import torch
import torch.nn.functional as F

from quintnet_amd import ops


def test_cross_entropy_matches_torch():
    torch.manual_seed(0)
    logits = torch.randn(64, 101, requires_grad=True)
    tgt = torch.randint(0, 101, (64,))
    tgt[::7] = -100
    ours = ops.cross_entropy(logits, tgt, ignore_index=-100)
    ref = F.cross_entropy(logits, tgt, ignore_index=-100)
    assert torch.allclose(ours, ref, atol=1e-5)
    g1 = torch.autograd.grad(ours, logits, retain_graph=True)[0]
    g2 = torch.autograd.grad(ref, logits)[0]
    assert torch.allclose(g1, g2, atol=1e-5)


def test_cross_entropy_all_ignored():
    logits = torch.randn(4, 10)
    tgt = torch.full((4,), -100)
    loss = ops.cross_entropy(logits, tgt)
    assert float(loss) == 0.0


def test_layer_norm_matches_torch():
    torch.manual_seed(1)
    x = torch.randn(8, 50, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    b = torch.randn(64, requires_grad=True)
    y = ops.layer_norm(x, w, b, 1e-5)
    y_ref = F.layer_norm(x, (64,), w, b, 1e-5)
    assert torch.allclose(y, y_ref, atol=1e-5)
    dy = torch.randn_like(y)
    gx, gw, gb = torch.autograd.grad(y, (x, w, b), dy, retain_graph=True)
    rx, rw, rb = torch.autograd.grad(y_ref, (x, w, b), dy)
    assert torch.allclose(gx, rx, atol=1e-4)
    assert torch.allclose(gw, rw, atol=1e-4)
    assert torch.allclose(gb, rb, atol=1e-4)


@pytest.mark.parametrize("causal", [False, True])
def test_attention_matches_sdpa(causal):
    torch.manual_seed(2)
    q = torch.randn(2, 4, 16, 8, requires_grad=True)
    k = torch.randn(2, 4, 16, 8, requires_grad=True)
    v = torch.randn(2, 4, 16, 8, requires_grad=True)
    out = ops.attention(q, k, v, causal=causal)
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=causal)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()
    dy = torch.randn_like(out)
    g = torch.autograd.grad(out, (q, k, v), dy, retain_graph=True)
    r = torch.autograd.grad(ref, (q, k, v), dy)
    for a, b in zip(g, r):
        assert torch.allclose(a, b, atol=1e-4)


@pytest.mark.parametrize("act", [None, "gelu", "relu"])
def test_linear_matches_torch(act):
    torch.manual_seed(3)
    x = torch.randn(6, 32, requires_grad=True)
    w = torch.randn(24, 32, requires_grad=True)
    b = torch.randn(24, requires_grad=True)
    y = ops.linear(x, w, b, activation=act)
    ref = F.linear(x, w, b)
    if act == "gelu":
        ref = F.gelu(ref, approximate="tanh")
    elif act == "relu":
        ref = F.relu(ref)
    assert torch.allclose(y, ref, atol=1e-5)
    dy = torch.randn_like(y)
    g = torch.autograd.grad(y, (x, w, b), dy, retain_graph=True)
    r = torch.autograd.grad(ref, (x, w, b), dy)
    for a, b_ in zip(g, r):
        assert torch.allclose(a, b_, atol=1e-4), (a - b_).abs().max()


def test_softmax_fwd_bwd_helpers():
    torch.manual_seed(4)
    s = torch.randn(2, 3, 8, 8)
    scale = 1.0 / math.sqrt(8)
    p = ops.causal_softmax(s, scale, causal=True)
    # rows sum to 1; upper triangle zero
    assert torch.allclose(p.sum(-1), torch.ones(2, 3, 8), atol=1e-5)
    assert float(p[..., 0, 1:].abs().max()) == 0.0


def test_clip_grad_norm_local():
    p = torch.nn.Parameter(torch.ones(10))
    p.grad = torch.full((10,), 2.0)
    norm = ops.clip_grad_norm_local([p], max_norm=1.0)
    assert abs(float(norm) - math.sqrt(40.0)) < 1e-5
    assert abs(float(ops.l2_norm([p.grad])) - 1.0) < 1e-4
