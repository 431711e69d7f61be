"""Megatron-style sequence parallelism (beyond reference parity):
TP=2 + SP GPT-2 must match the unsharded single model exactly over
spawned gloo, including LayerNorm/bias grad sync."""

import torch

from conftest import run_distributed


def _sp_equivalence(rank, world):
    import copy

    import torch.distributed as dist

    from quintnet_amd import init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss

    pg = init_process_groups("cpu", [1, 2, 1], ["dp", "tp", "pp"])
    tpg = pg.get_group("tp")
    torch.manual_seed(33)
    full_cfg = GPT2Config(
        n_embd=32, n_layer=2, n_head=2, vocab_size=64, n_positions=32, dropout=0.0
    )
    full = GPT2Stage(full_cfg)
    for p in full.parameters():
        dist.broadcast(p.data, src=0)

    sp_cfg = GPT2Config(
        n_embd=32, n_layer=2, n_head=2, vocab_size=64, n_positions=32, dropout=0.0,
        sequence_parallel=True,
    )
    sp = GPT2Stage(sp_cfg, tp_group=tpg)
    # shard the full weights into the SP/TP model
    fsd = full.state_dict()
    with torch.no_grad():
        tgt = sp.state_dict()
        for k, v in tgt.items():
            src = fsd[k]
            if "attn.c_attn." in k:
                q, kk, vv = src.chunk(3, dim=0)
                parts = [t.chunk(2, dim=0)[rank] for t in (q, kk, vv)]
                v.copy_(torch.cat(parts, dim=0))
            elif "mlp.c_fc." in k:
                v.copy_(src.chunk(2, dim=0)[rank])
            elif "c_proj.weight" in k:
                v.copy_(src.chunk(2, dim=1)[rank])
            else:
                v.copy_(src)

    ids = torch.randint(0, 64, (2, 16))
    dist.broadcast(ids, src=0)

    logits_sp = sp(ids)
    logits_full = full(ids)
    assert torch.allclose(logits_sp, logits_full, atol=1e-4), (
        (logits_sp - logits_full).abs().max()
    )

    # backward: grads of replicated (LN) params must match the full model
    loss_sp = causal_lm_loss(logits_sp, ids)
    loss_sp.backward()
    sp.sync_tied_weights_grad()  # runs the SP grad sync
    loss_full = causal_lm_loss(logits_full, ids)
    loss_full.backward()
    assert abs(float(loss_sp) - float(loss_full)) < 1e-5

    g_sp = sp.blocks[0].ln_1.weight.grad
    g_full = full.blocks[0].ln_1.weight.grad
    assert torch.allclose(g_sp, g_full, atol=1e-4), (g_sp - g_full).abs().max()
    b_sp = sp.blocks[0].attn.c_proj.bias.grad
    b_full = full.blocks[0].attn.c_proj.bias.grad
    assert torch.allclose(b_sp, b_full, atol=1e-4)
    # wte grads complete despite the sequence scatter
    assert torch.allclose(
        sp.embedding.wte.weight.grad, full.embedding.wte.weight.grad, atol=1e-4
    )


def test_sequence_parallel_tp2_equivalence():
    run_distributed(_sp_equivalence, 2, timeout=300)
