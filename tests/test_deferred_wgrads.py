"""Deferred weight gradients (ops/linear.py, zero-bubble building
block): queue dW during backward, flush later — gradients must equal
the normal path exactly."""

import copy

import torch

from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.ops import causal_lm_loss, defer_wgrads, flush_deferred_wgrads


def _model():
    torch.manual_seed(3)
    return GPT2Stage(GPT2Config(
        n_embd=64, n_layer=2, n_head=2, vocab_size=96, n_positions=32,
        dropout=0.0,
    ))


def test_deferred_equals_normal_exactly():
    m0 = _model()
    m1 = copy.deepcopy(m0)
    ids = torch.randint(0, 96, (2, 16))
    labels = torch.randint(0, 96, (2, 16))

    causal_lm_loss(m0(ids), labels).backward()

    with defer_wgrads.scope():
        causal_lm_loss(m1(ids), labels).backward()
    # linear weights got NO grad yet; everything else did
    assert m1.blocks[0].attn.c_attn.weight.grad is None
    assert m1.blocks[0].ln_1.weight.grad is not None
    n = flush_deferred_wgrads()
    assert n > 0
    for (k0, p0), (_, p1) in zip(m0.named_parameters(), m1.named_parameters()):
        assert p1.grad is not None, k0
        assert torch.allclose(p0.grad, p1.grad, rtol=1e-6, atol=1e-7), k0


def test_deferred_accumulates_over_microbatches():
    m0 = _model()
    m1 = copy.deepcopy(m0)
    batches = [
        (torch.randint(0, 96, (2, 16)), torch.randint(0, 96, (2, 16)))
        for _ in range(3)
    ]
    for ids, labels in batches:
        causal_lm_loss(m0(ids), labels).backward()
    with defer_wgrads.scope():
        for ids, labels in batches:
            causal_lm_loss(m1(ids), labels).backward()
    flush_deferred_wgrads()
    for (k0, p0), (_, p1) in zip(m0.named_parameters(), m1.named_parameters()):
        assert torch.allclose(p0.grad, p1.grad, rtol=1e-6, atol=1e-6), k0


def test_flush_is_idempotent_and_scoped():
    m = _model()
    ids = torch.randint(0, 96, (1, 8))
    labels = torch.randint(0, 96, (1, 8))
    with defer_wgrads.scope():
        causal_lm_loss(m(ids), labels).backward()
    assert flush_deferred_wgrads() > 0
    assert flush_deferred_wgrads() == 0  # queue drained
    # outside the scope the normal path is back
    m.zero_grad()
    causal_lm_loss(m(ids), labels).backward()
    assert m.blocks[0].attn.c_attn.weight.grad is not None
    assert flush_deferred_wgrads() == 0


def _ddp_defer(rank, world):
    """Deferred dW under DataParallel: hooks must stay OFF (deferral
    bypasses autograd accumulation), flush fills the bucket views, and
    finalize_gradients reduces complete gradients — equal to plain DDP."""
    import torch.distributed as dist

    from quintnet_amd.parallel import DataParallel, DistributedConfig

    torch.manual_seed(7)
    m_ref = _model()
    for p in m_ref.parameters():
        dist.broadcast(p.data, src=0)
    m_zb = copy.deepcopy(m_ref)

    ddp_ref = DataParallel(m_ref, DistributedConfig(rank, world, None))
    ddp_zb = DataParallel(m_zb, DistributedConfig(rank, world, None))

    torch.manual_seed(100 + rank)  # different data per rank
    ids = torch.randint(0, 96, (2, 16))
    labels = torch.randint(0, 96, (2, 16))

    causal_lm_loss(ddp_ref(ids), labels).backward()
    ddp_ref.finalize_gradients()

    ddp_zb.require_backward_grad_sync = False  # hooks keyed on autograd
    with defer_wgrads.scope():
        causal_lm_loss(ddp_zb(ids), labels).backward()
    flush_deferred_wgrads()
    ddp_zb.require_backward_grad_sync = True
    ddp_zb.finalize_gradients()

    for (k0, p0), (_, p1) in zip(m_ref.named_parameters(), m_zb.named_parameters()):
        assert torch.allclose(p0.grad, p1.grad, rtol=1e-6, atol=1e-6), k0


def test_ddp_deferred_wgrads_world2():
    from conftest import run_distributed

    run_distributed(_ddp_defer, 2)


def _pp2_defer(rank, world):
    """pp=2 1F1B with defer_wgrads: loss matches the normal 1F1B run
    (same seeds/data) — the ZB mechanism is schedule-transparent."""
    from torch.utils.data import DataLoader

    from quintnet_amd import GPT2Trainer, init_process_groups
    from quintnet_amd.parallel import PipelineParallelWrapper
    from quintnet_amd.utils.data import SyntheticCLM

    pg = init_process_groups("cpu", [1, 1, 2], ["dp", "tp", "pp"])
    cfg = GPT2Config(n_embd=32, n_layer=4, n_head=2, vocab_size=96,
                     n_positions=16, dropout=0.0)

    def run(flag):
        # same per-rank seed in both runs -> identical init; no cross-
        # rank broadcast (stage params differ per pp rank by design)
        torch.manual_seed(41 + rank)
        stage = GPT2Stage(cfg, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
                          tied_group=pg.get_tied_embedding_group())
        w = PipelineParallelWrapper(
            stage_module=stage, pp_rank=pg.pp_rank,
            pp_group=pg.get_group("pp"), pp_size=pg.pp_size)
        w.seq_len, w.hidden_dim = 16, cfg.n_embd
        ds = SyntheticCLM(n=8, seq_len=16, vocab_size=96, seed=6)
        tcfg = {"num_epochs": 1, "grad_acc_steps": 2, "zero1": True,
                "schedule": "1f1b", "max_seq_length": 16,
                "task_type": "clm", "defer_wgrads": flag,
                "model_config": {"n_embd": cfg.n_embd}}
        tr = GPT2Trainer(w, DataLoader(ds, batch_size=2), None, tcfg, pg)
        return tr.fit()["train_loss"]

    plain = run(False)
    zb = run(True)
    assert abs(plain - zb) < 1e-6, (plain, zb)


def test_pp2_deferred_wgrads_matches():
    from conftest import run_distributed

    run_distributed(_pp2_defer, 2)


def _interleaved_defer(rank, world):
    """Interleaved 1F1B with defer_wgrads: loss identical to the same
    run without deferral."""
    from torch.utils.data import DataLoader

    from quintnet_amd import GPT2Trainer, get_strategy, init_process_groups
    from quintnet_amd.models import GPT2ForInterleaving
    from quintnet_amd.utils.data import SyntheticCLM

    pg = init_process_groups("cpu", [world], ["pp"])
    cfg = GPT2Config(vocab_size=96, n_positions=16, n_embd=32, n_layer=4,
                     n_head=2, dropout=0.0)

    def run(flag):
        torch.manual_seed(29 + rank)
        model = GPT2ForInterleaving(cfg)
        tcfg = {"schedule": "interleaved", "num_chunks": 2, "num_epochs": 1,
                "grad_acc_steps": 2, "zero1": True, "task_type": "clm",
                "max_seq_length": 16, "defer_wgrads": flag,
                "model_config": {"n_embd": 32, "n_positions": 16}}
        pmodel = get_strategy("pp", pg, tcfg).apply(model)
        ds = SyntheticCLM(n=8, seq_len=16, vocab_size=96, seed=2)
        tr = GPT2Trainer(pmodel, DataLoader(ds, batch_size=2), None, tcfg, pg)
        return tr.fit()["train_loss"]

    assert abs(run(False) - run(True)) < 1e-6


def test_interleaved_deferred_wgrads_matches():
    from conftest import run_distributed

    run_distributed(_interleaved_defer, 2)


def _zb_zero2_pp_dp(rank, world):
    """defer_wgrads x zero_stage=2 x pp2 x dp2: the flush must land the
    dW adds in the reduce-scatter bucket views BEFORE finalize launches
    them; params stay dp-identical and match the non-deferred run."""
    from quintnet_amd import init_process_groups
    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.optim import Zero2AdamW
    from quintnet_amd.parallel import (
        BucketConfig,
        DataParallel,
        DistributedConfig,
        PipelineParallelWrapper,
    )

    pg = init_process_groups("cpu", [2, 2], ["dp", "pp"])
    cfg = GPT2Config(n_embd=64, n_layer=4, n_head=2, vocab_size=128,
                     n_positions=32, dropout=0.0)

    def run(flag):
        torch.manual_seed(5 + pg.pp_rank)  # dp pair identical init
        stage = GPT2Stage(cfg, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
                          tied_group=pg.get_tied_embedding_group())
        stage.seq_len, stage.hidden_dim = 32, 64
        config = {"zero_stage": 2, "learning_rate": 1e-3, "num_epochs": 1,
                  "grad_acc_steps": 2, "max_grad_norm": 1.0,
                  "defer_wgrads": flag,
                  "model_config": {"n_positions": 32, "n_embd": 64}}
        pmodel = PipelineParallelWrapper(
            stage_module=stage, pp_rank=pg.pp_rank,
            pp_group=pg.get_group("pp"), pp_size=pg.pp_size)
        pmodel.seq_len, pmodel.hidden_dim = 32, 64
        pmodel = DataParallel(
            pmodel,
            DistributedConfig(pg.dp_rank, pg.dp_size, pg.get_group("dp")),
            bucket_config=BucketConfig(grad_reduce_op="reduce_scatter"))
        g = torch.Generator().manual_seed(7)
        ids = torch.randint(0, 128, (8, 32), generator=g)
        data = [{"input_ids": ids[i : i + 2], "labels": ids[i : i + 2].clone()}
                for i in range(0, 8, 2)]
        tr = GPT2Trainer(pmodel, data, None, config, pg)
        assert isinstance(tr.optimizer, Zero2AdamW)
        tr._train_epoch(0)
        return [p.detach().clone() for p in stage.parameters()]

    plain = run(False)
    zb = run(True)
    for a, b in zip(plain, zb):
        assert torch.allclose(a, b, atol=1e-6)


def test_zb_zero2_pp_dp_world4():
    from conftest import run_distributed

    run_distributed(_zb_zero2_pp_dp, 4)
