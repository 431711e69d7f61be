"""ZeRO-1 AdamW: equivalence vs torch.optim.AdamW (1 rank) and
shard-consistency at dp=2 over spawned gloo."""

import torch

from conftest import run_distributed


def test_adamw_matches_torch_single_rank():
    import copy

    from quintnet_amd.optim import ZeroRedundancyAdamW

    torch.manual_seed(0)
    m1 = torch.nn.Sequential(torch.nn.Linear(16, 16), torch.nn.Tanh(), torch.nn.Linear(16, 4))
    m2 = copy.deepcopy(m1)
    opt1 = torch.optim.AdamW(m1.parameters(), lr=1e-2, weight_decay=0.01)
    opt2 = ZeroRedundancyAdamW(m2.parameters(), lr=1e-2, weight_decay=0.01)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    for _ in range(5):
        for m, o in ((m1, opt1), (m2, opt2)):
            o.zero_grad()
            ((m(x) - y) ** 2).mean().backward()
            o.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), (p1 - p2).abs().max()


def _zero_dp2(rank, world):
    import copy

    import torch.distributed as dist

    from quintnet_amd.optim import ZeroRedundancyAdamW
    from quintnet_amd.parallel import DataParallel

    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(16, 16), torch.nn.Tanh(), torch.nn.Linear(16, 4))
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    ref = copy.deepcopy(model)

    ddp = DataParallel(model)
    opt = ZeroRedundancyAdamW(
        ddp.parameters(), lr=1e-2, weight_decay=0.01, dp_group=dist.group.WORLD
    )

    ref_opt = torch.optim.AdamW(ref.parameters(), lr=1e-2, weight_decay=0.01)

    for it in range(3):
        g = torch.Generator().manual_seed(1000 * it + rank)
        x = torch.randn(4, 16, generator=g)
        y = torch.randn(4, 4, generator=g)
        ddp.zero_grad()
        ((ddp(x) - y) ** 2).mean().backward()
        ddp.finalize_gradients()
        opt.step()

        # oracle: full AdamW on grads averaged over both ranks' data
        ref_opt.zero_grad()
        for r in range(world):
            gr = torch.Generator().manual_seed(1000 * it + r)
            xr = torch.randn(4, 16, generator=gr)
            yr = torch.randn(4, 4, generator=gr)
            (((ref(xr) - yr) ** 2).mean() / world).backward()
        ref_opt.step()

    for (n, p), rp in zip(model.named_parameters(), ref.parameters()):
        assert torch.allclose(p, rp, atol=1e-4), (n, (p - rp).abs().max())


def test_zero1_dp2_matches_full_adamw():
    run_distributed(_zero_dp2, 2)


def _zero2_matches_zero1(rank, world):
    """ZeRO-2 (reduce-scatter buckets + per-bucket shard step + bucket
    all-gather) must produce the same params as ZeRO-1 (all-reduce +
    contiguous shard) — fp32/gloo makes the comparison exact."""
    import torch
    import torch.nn as nn

    from quintnet_amd.optim import Zero2AdamW, ZeroRedundancyAdamW
    from quintnet_amd.parallel import BucketConfig, DataParallel

    torch.manual_seed(7)
    def mk():
        return nn.Sequential(nn.Linear(64, 256), nn.GELU(), nn.Linear(256, 32))

    m1, m2 = mk(), mk()
    m2.load_state_dict(m1.state_dict())

    ddp1 = DataParallel(m1)
    opt1 = ZeroRedundancyAdamW.from_ddp(ddp1, lr=1e-2, max_grad_norm=1.0)
    ddp2 = DataParallel(
        m2, bucket_config=BucketConfig(capacity_mb=0.05,
                                       grad_reduce_op="reduce_scatter"))
    opt2 = Zero2AdamW(ddp2, lr=1e-2, max_grad_norm=1.0)
    assert len(ddp2.buckets) > 1  # exercise multi-bucket path

    for it in range(3):
        g = torch.Generator().manual_seed(100 + 10 * it + rank)
        x = torch.randn(8, 64, generator=g)
        for ddp, opt in ((ddp1, opt1), (ddp2, opt2)):
            ddp(x).pow(2).mean().backward()
            ddp.finalize_gradients()
            opt.step()
            ddp.zero_grad()
        for p1, p2 in zip(m1.parameters(), m2.parameters()):
            assert torch.allclose(p1, p2, atol=1e-6), (it, (p1 - p2).abs().max())


def test_zero2_matches_zero1_gloo():
    from conftest import run_distributed

    run_distributed(_zero2_matches_zero1, 2)


def _zero2_e2e_trainer(rank, world):
    """zero_stage=2 end to end: strategy -> coordinator builds the
    reduce-scatter DDP -> GPT2Trainer picks Zero2AdamW -> loss drops."""
    import torch

    from quintnet_amd import get_strategy, init_process_groups
    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.optim import Zero2AdamW

    torch.manual_seed(0)
    pg = init_process_groups("cpu", [world], ["dp"])
    cfg = GPT2Config(n_embd=64, n_layer=2, n_head=2, vocab_size=128,
                     n_positions=32, dropout=0.0)
    model = GPT2Stage(cfg)
    config = {"zero_stage": 2, "learning_rate": 1e-3, "num_epochs": 2,
              "grad_acc_steps": 2, "max_grad_norm": 1.0}
    pmodel = get_strategy("dp", pg, config).apply(model)
    assert getattr(pmodel, "_reduce_scatter", False)

    g = torch.Generator().manual_seed(5)
    ids = torch.randint(0, 128, (8, 32), generator=g)
    data = [{"input_ids": ids[i : i + 2], "labels": ids[i : i + 2].clone()}
            for i in range(0, 8, 2)]
    tr = GPT2Trainer(pmodel, data, None, config, pg)
    assert isinstance(tr.optimizer, Zero2AdamW)
    hist1 = tr._train_epoch_plain()
    hist2 = tr._train_epoch_plain()
    assert hist2["loss"] < hist1["loss"]
    # params identical across ranks after the bucket all-gathers
    import torch.distributed as dist

    for p in model.parameters():
        t = p.detach().clone()
        dist.broadcast(t, src=0)
        assert torch.allclose(t, p.detach(), atol=1e-6)


def test_zero2_e2e_trainer_gloo():
    from conftest import run_distributed

    run_distributed(_zero2_e2e_trainer, 2)


def _zero2_with_pipeline(rank, world):
    """zero_stage=2 composed with pp=2 x dp=2 (the bench-like assembly):
    schedule + reduce-scatter DDP + Zero2AdamW step must run and keep
    params dp-identical."""
    import torch

    from quintnet_amd import init_process_groups
    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.optim import Zero2AdamW

    from quintnet_amd.parallel import (
        BucketConfig,
        DataParallel,
        DistributedConfig,
        PipelineParallelWrapper,
    )

    torch.manual_seed(0)
    pg = init_process_groups("cpu", [2, 2], ["dp", "pp"])
    cfg = GPT2Config(n_embd=64, n_layer=4, n_head=2, vocab_size=128,
                     n_positions=32, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
                      tied_group=pg.get_tied_embedding_group())
    stage.seq_len, stage.hidden_dim = 32, 64
    config = {"zero_stage": 2, "learning_rate": 1e-3, "num_epochs": 1,
              "grad_acc_steps": 2, "max_grad_norm": 1.0,
              "model_config": {"n_positions": 32, "n_embd": 64}}
    # the bench-style assembly: stage_module wrapper + reduce-scatter DDP
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
    m = tr._train_epoch(0)
    # dp pair holds identical params after the bucket all-gathers
    import torch.distributed as dist

    for p in stage.parameters():
        t = p.detach().clone()
        dist.broadcast(t, src=dist.get_global_rank(pg.get_group("dp"), 0),
                       group=pg.get_group("dp"))
        assert torch.allclose(t, p.detach(), atol=1e-6)


def test_zero2_with_pipeline_gloo():
    from conftest import run_distributed

    run_distributed(_zero2_with_pipeline, 4)


def _kitchen_sink_compose(rank, world):
    """Feature-interaction smoke: ZeRO-2 + cosine LR warmup + activation
    checkpointing + padded vocab + NaN guard + metrics, in one run —
    loss drops, ranks stay in sync."""
    import os
    import tempfile

    import torch
    import torch.distributed as dist

    from quintnet_amd import get_strategy, init_process_groups
    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.optim import Zero2AdamW

    torch.manual_seed(0)
    pg = init_process_groups("cpu", [world], ["dp"])
    cfg = GPT2Config(n_embd=64, n_layer=2, n_head=2, vocab_size=100,
                     n_positions=32, dropout=0.0, vocab_pad_to=64,
                     activation_checkpointing=True)
    model = GPT2Stage(cfg)
    config = {"zero_stage": 2, "learning_rate": 1e-3, "num_epochs": 2,
              "grad_acc_steps": 2, "max_grad_norm": 1.0,
              "lr_schedule": "cosine", "warmup_steps": 2,
              "detect_nan_grads": True}
    pmodel = get_strategy("dp", pg, config).apply(model)
    g = torch.Generator().manual_seed(5)
    ids = torch.randint(0, 100, (8, 32), generator=g)
    data = [{"input_ids": ids[i : i + 2], "labels": ids[i : i + 2].clone()}
            for i in range(0, 8, 2)]
    tr = GPT2Trainer(pmodel, data, None, config, pg)
    assert isinstance(tr.optimizer, Zero2AdamW)
    hist = tr.fit()
    assert torch.isfinite(torch.tensor(hist["train_loss"]))
    assert tr.lr_scheduler is not None and tr.lr_scheduler._step > 0
    for p in model.parameters():
        t = p.detach().clone()
        dist.broadcast(t, src=0)
        assert torch.allclose(t, p.detach(), atol=1e-6)
    # padded rows never trained
    assert bool((model.embedding.wte.weight[100:] == 0).all())


def test_kitchen_sink_compose_gloo():
    from conftest import run_distributed

    run_distributed(_kitchen_sink_compose, 2)
