"""ZeRO-3 parameter sharding: trajectory equivalence with replicated DP
training, shard-only storage, and grad reduce-scatter correctness."""

import copy

import pytest
import torch

from conftest import run_distributed


def test_zero3_single_rank_noop_math():
    """world=1: wrapped model computes identically and grads land on shard."""
    from quintnet_amd.models import Model
    from quintnet_amd.parallel.zero3 import apply_zero3

    torch.manual_seed(0)
    m = Model(hidden_dim=32, n_heads=2, depth=2)
    ref = copy.deepcopy(m)
    apply_zero3(m, dp_group=None)
    x = torch.randn(2, 1, 28, 28)
    out = m(x)
    assert torch.allclose(out, ref(x), atol=1e-5)
    out.sum().backward()
    for blk in m.blocks:
        assert blk.shard.grad is not None and torch.isfinite(blk.shard.grad).all()


def _run_zero3(rank, world):
    import torch.distributed as dist

    from quintnet_amd.models import Model
    from quintnet_amd.parallel.zero3 import apply_zero3

    torch.manual_seed(11)
    model = Model(hidden_dim=32, n_heads=2, depth=4)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    # freeze everything outside the sharded blocks: syncing those is
    # DataParallel's job, and letting them drift per-rank would
    # contaminate the block-gradient comparison
    for name, p in model.named_parameters():
        if not name.startswith("blocks."):
            p.requires_grad_(False)
    ref = copy.deepcopy(model)

    apply_zero3(model, dp_group=dist.group.WORLD)

    # each block's own storage is freed; only the 1/world shard remains
    for blk in model.blocks:
        assert sum(p.numel() for p in blk.module.parameters()) == 0
        assert blk.shard.numel() * world >= sum(n for _, n in blk._metas)

    # DP semantics: different data per rank, grads averaged
    opt = torch.optim.SGD([p for p in model.parameters() if p.requires_grad], lr=0.1)
    ref_opt = torch.optim.SGD([p for p in ref.parameters() if p.requires_grad], lr=0.1)
    for step in range(3):
        torch.manual_seed(500 + step * world + rank)
        x = torch.randn(2, 1, 28, 28)
        y = torch.randint(0, 10, (2,))
        loss = torch.nn.functional.cross_entropy(model(x), y)
        (loss / world).backward()  # ZeRO grads reduce-SUM; scale for mean
        opt.step()
        opt.zero_grad()

        # reference: replicated model sees ALL ranks' batches (DP mean)
        ref_loss = 0.0
        for r in range(world):
            torch.manual_seed(500 + step * world + r)
            xr = torch.randn(2, 1, 28, 28)
            yr = torch.randint(0, 10, (2,))
            ref_loss = ref_loss + torch.nn.functional.cross_entropy(ref(xr), yr)
        (ref_loss / world).backward()
        ref_opt.step()
        ref_opt.zero_grad()

    # compare: re-gather full params of each block vs reference blocks
    for blk, rblk in zip(model.blocks, ref.blocks):
        full = blk.full_state_dict_tensors()
        rparams = dict(rblk.named_parameters())
        for name, t in full.items():
            assert torch.allclose(t, rparams[name], atol=1e-4), (name,)



def test_zero3_world2():
    run_distributed(_run_zero3, 2)


@pytest.mark.slow
def test_zero3_world4():
    run_distributed(_run_zero3, 4)


def _run_zero3_full(rank, world):
    """Whole-model sharding (blocks + embedding + head): no unwrapped
    params left, trajectory matches replicated DP exactly."""
    import torch.distributed as dist

    from quintnet_amd.models import Model
    from quintnet_amd.parallel.zero3 import apply_zero3

    torch.manual_seed(12)
    model = Model(hidden_dim=32, n_heads=2, depth=2)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    ref = copy.deepcopy(model)

    apply_zero3(model, dp_group=dist.group.WORLD,
                extra_attrs=("embedding", "classification_head"))
    # every trainable parameter is now a shard
    for n, p in model.named_parameters():
        if p.requires_grad:
            assert n.endswith("shard"), n

    opt = torch.optim.SGD([p for p in model.parameters() if p.requires_grad], lr=0.1)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)
    for step in range(2):
        torch.manual_seed(900 + step * world + rank)
        x = torch.randn(2, 1, 28, 28)
        y = torch.randint(0, 10, (2,))
        loss = torch.nn.functional.cross_entropy(model(x), y)
        (loss / world).backward()
        opt.step(); opt.zero_grad()

        ref_loss = 0.0
        for r in range(world):
            torch.manual_seed(900 + step * world + r)
            xr = torch.randn(2, 1, 28, 28)
            yr = torch.randint(0, 10, (2,))
            ref_loss = ref_loss + torch.nn.functional.cross_entropy(ref(xr), yr)
        (ref_loss / world).backward()
        ref_opt.step(); ref_opt.zero_grad()

    for blk, rmod in [(model.embedding, ref.embedding),
                      (model.classification_head, ref.classification_head)]:
        full = blk.full_state_dict_tensors()
        rparams = dict(rmod.named_parameters())
        for name, t in full.items():
            assert torch.allclose(t, rparams[name], atol=1e-4), (name,)


def test_zero3_whole_model():
    run_distributed(_run_zero3_full, 2)


def _run_zero3_gpt2(rank, world):
    """ZeRO-3 over GPT2Stage's RESIDUAL-FUSED block chain (stage.forward
    calls blk.forward_fused — ZeRO3Block.forward_fused gathers/
    checkpoints the same way) vs a replicated DP oracle."""
    import torch.distributed as dist

    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel.zero3 import apply_zero3

    torch.manual_seed(7)
    cfg = GPT2Config(n_embd=32, n_layer=2, n_head=2, vocab_size=64,
                     n_positions=32, dropout=0.0)
    model = GPT2Stage(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    for name, p in model.named_parameters():
        if not name.startswith("blocks."):
            p.requires_grad_(False)
    ref = copy.deepcopy(model)
    apply_zero3(model, dp_group=dist.group.WORLD)

    opt = torch.optim.SGD([p for p in model.parameters() if p.requires_grad], lr=0.1)
    ref_opt = torch.optim.SGD([p for p in ref.parameters() if p.requires_grad], lr=0.1)
    for step in range(2):
        torch.manual_seed(900 + step * world + rank)
        ids = torch.randint(0, 64, (2, 16))
        labels = torch.randint(0, 64, (2, 16))
        loss = causal_lm_loss(model(ids), labels)
        (loss / world).backward()
        opt.step()
        opt.zero_grad()

        ref_loss = 0.0
        for r in range(world):
            torch.manual_seed(900 + step * world + r)
            idr = torch.randint(0, 64, (2, 16))
            lbr = torch.randint(0, 64, (2, 16))
            ref_loss = ref_loss + causal_lm_loss(ref(idr), lbr)
        (ref_loss / world).backward()
        ref_opt.step()
        ref_opt.zero_grad()

    for blk, rblk in zip(model.blocks, ref.blocks):
        full = blk.full_state_dict_tensors()
        rparams = dict(rblk.named_parameters())
        for name, t in full.items():
            assert torch.allclose(t, rparams[name], atol=1e-4), (name,)


def test_zero3_gpt2_fused_chain_world2():
    run_distributed(_run_zero3_gpt2, 2)


def test_zero3_checkpoint_save_refuses_sharded_model(tmp_path):
    """A ZeRO-3-wrapped model must not silently save flat shards under
    the named-parameter checkpoint contract."""
    from quintnet_amd.checkpoint import save_sharded_checkpoint
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.parallel.zero3 import apply_zero3

    m = GPT2Stage(GPT2Config(n_embd=32, n_layer=2, n_head=2, vocab_size=64,
                             n_positions=32, dropout=0.0))
    apply_zero3(m, dp_group=None)
    with pytest.raises(ValueError, match="ZeRO-3"):
        save_sharded_checkpoint(m, str(tmp_path))


def test_zero3_skips_double_activation_checkpointing():
    """ZeRO-3 already recomputes inside the gather region; the stage's
    own activation_checkpointing flag must not re-wrap it (grads still
    correct, single recompute)."""
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel.zero3 import apply_zero3

    torch.manual_seed(2)
    cfg = GPT2Config(n_embd=32, n_layer=2, n_head=2, vocab_size=64,
                     n_positions=32, dropout=0.0,
                     activation_checkpointing=True)
    m = GPT2Stage(cfg)
    ref = copy.deepcopy(m)
    apply_zero3(m, dp_group=None)
    m.train(), ref.train()
    ids = torch.randint(0, 64, (2, 16))
    labels = torch.randint(0, 64, (2, 16))
    l1 = causal_lm_loss(m(ids), labels)
    l0 = causal_lm_loss(ref(ids), labels)
    assert torch.allclose(l0, l1, atol=1e-6)
    l1.backward()
    for blk in m.blocks:
        assert blk.shard.grad is not None
        assert torch.isfinite(blk.shard.grad).all()
