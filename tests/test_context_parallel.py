"""Context parallelism: CP-sharded attention equals full attention
(forward + gradients) on gloo; also the q_offset attention fallback."""

import math

import pytest
import torch

from conftest import run_distributed


def _ref_attention(q, k, v, causal, q_offset=0):
    scale = 1.0 / math.sqrt(q.shape[-1])
    s = torch.matmul(q.float(), k.float().transpose(-2, -1)) * scale
    if causal:
        Tq, Tk = s.shape[-2], s.shape[-1]
        mask = torch.ones(Tq, Tk, dtype=torch.bool).tril(q_offset)
        s = s.masked_fill(~mask, float("-inf"))
    return torch.matmul(torch.softmax(s, -1), v.float())


def test_attention_q_offset_cpu():
    from quintnet_amd.ops.attention import attention

    torch.manual_seed(0)
    B, H, Tq, Tk, D = 2, 3, 8, 16, 16
    q = torch.randn(B, H, Tq, D, requires_grad=True)
    k = torch.randn(B, H, Tk, D, requires_grad=True)
    v = torch.randn(B, H, Tk, D, requires_grad=True)
    out = attention(q, k, v, causal=True, q_offset=8)
    ref = _ref_attention(q, k, v, True, 8)
    assert torch.allclose(out, ref.to(out.dtype), atol=1e-5)
    out.sum().backward()
    assert q.grad is not None and k.grad is not None and torch.isfinite(k.grad).all()


def test_q_offset_matches_slice_of_full():
    """attention(q_shard, k_full, v_full, offset) == rows of the full result."""
    from quintnet_amd.ops.attention import attention

    torch.manual_seed(1)
    B, H, T, D = 2, 2, 32, 16
    q = torch.randn(B, H, T, D)
    k = torch.randn(B, H, T, D)
    v = torch.randn(B, H, T, D)
    full = attention(q, k, v, causal=True)
    for cp in (2, 4):
        tl = T // cp
        for r in range(cp):
            shard = attention(q[:, :, r * tl : (r + 1) * tl], k, v,
                              causal=True, q_offset=r * tl)
            assert torch.allclose(shard, full[:, :, r * tl : (r + 1) * tl], atol=1e-5)


def _run_cp(rank, world):
    import torch.distributed as dist

    from quintnet_amd.parallel import context_parallel_attention, scatter_to_context

    torch.manual_seed(7)
    B, H, T, D = 2, 2, 64, 16
    q = torch.randn(B, H, T, D)
    k = torch.randn(B, H, T, D)
    v = torch.randn(B, H, T, D)
    for t in (q, k, v):
        dist.broadcast(t, src=0)

    # full-sequence reference with grads
    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref = _ref_attention(qr, kr, vr, True)
    ref.square().sum().backward()

    # CP: each rank holds a T/world shard of q, k, v
    ql = scatter_to_context(q, dist.group.WORLD, dim=2).requires_grad_(True)
    kl = scatter_to_context(k, dist.group.WORLD, dim=2).requires_grad_(True)
    vl = scatter_to_context(v, dist.group.WORLD, dim=2).requires_grad_(True)
    out = context_parallel_attention(ql, kl, vl, dist.group.WORLD, causal=True)
    tl = T // world
    sl = slice(rank * tl, (rank + 1) * tl)
    assert torch.allclose(out, ref.detach()[:, :, sl].to(out.dtype), atol=1e-5)

    # grads: the loss sum over the full output = sum of per-rank shard
    # losses, so local backward + the all-gather's reduce-scatter must
    # reproduce the reference grads' shards
    out.square().sum().backward()
    assert torch.allclose(ql.grad, qr.grad[:, :, sl], atol=1e-5)
    assert torch.allclose(kl.grad, kr.grad[:, :, sl], atol=1e-5)
    assert torch.allclose(vl.grad, vr.grad[:, :, sl], atol=1e-5)


def test_context_parallel_cp2():
    run_distributed(_run_cp, 2)


def test_context_parallel_cp4():
    run_distributed(_run_cp, 4)


def _run_ring(rank, world, causal):
    import torch.distributed as dist

    from quintnet_amd.parallel import ring_attention, scatter_to_context

    torch.manual_seed(13)
    B, H, T, D = 2, 2, 64, 16
    q = torch.randn(B, H, T, D)
    k = torch.randn(B, H, T, D)
    v = torch.randn(B, H, T, D)
    for t in (q, k, v):
        dist.broadcast(t, src=0)

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref = _ref_attention(qr, kr, vr, causal)
    ref.square().sum().backward()

    ql = scatter_to_context(q, dist.group.WORLD, dim=2).requires_grad_(True)
    kl = scatter_to_context(k, dist.group.WORLD, dim=2).requires_grad_(True)
    vl = scatter_to_context(v, dist.group.WORLD, dim=2).requires_grad_(True)
    out = ring_attention(ql, kl, vl, dist.group.WORLD, causal=causal)
    tl = T // world
    sl = slice(rank * tl, (rank + 1) * tl)
    assert torch.allclose(out, ref.detach()[:, :, sl].to(out.dtype), atol=1e-5), (
        (out - ref.detach()[:, :, sl]).abs().max()
    )
    out.square().sum().backward()
    assert torch.allclose(ql.grad, qr.grad[:, :, sl], atol=1e-5)
    assert torch.allclose(kl.grad, kr.grad[:, :, sl], atol=1e-5)
    assert torch.allclose(vl.grad, vr.grad[:, :, sl], atol=1e-5)


def _ring_causal(rank, world):
    _run_ring(rank, world, True)


def _ring_full(rank, world):
    _run_ring(rank, world, False)


def test_ring_attention_cp2_causal():
    run_distributed(_ring_causal, 2)


def test_ring_attention_cp4_causal():
    run_distributed(_ring_causal, 4)


def test_ring_attention_cp2_noncausal():
    run_distributed(_ring_full, 2)


def _run_gpt2_cp(rank, world):
    """End-to-end CP training step for GPT-2: sequence-sharded blocks +
    partitioned loss + DataParallel(cp) MEAN reduction == full-sequence
    single-process gradients exactly."""
    import copy

    import torch.distributed as dist

    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import (
        DataParallel,
        DistributedConfig,
        cp_causal_lm_loss,
        scatter_clm_targets,
        scatter_to_context,
    )

    torch.manual_seed(31)
    cfg = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    full = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    for p in full.parameters():
        dist.broadcast(p.data, src=0)
    ref = copy.deepcopy(full)

    cp_stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                         cp_group=dist.group.WORLD)
    cp_stage.load_state_dict(full.state_dict())
    model = DataParallel(
        cp_stage, DistributedConfig(rank, world, dist.group.WORLD)
    )

    ids = torch.randint(0, 96, (2, 32))
    labels = ids.clone()
    dist.broadcast(ids, src=0)
    dist.broadcast(labels, src=0)

    ids_shard = scatter_to_context(ids, dist.group.WORLD, dim=1)
    tgt_shard = scatter_clm_targets(labels, dist.group.WORLD)
    logits_shard = model(ids_shard)
    loss_bwd, true_loss = cp_causal_lm_loss(
        logits_shard, tgt_shard, dist.group.WORLD
    )
    loss_bwd.backward()
    model.finalize_gradients()

    # reference
    out = ref(ids)
    ref_loss = causal_lm_loss(out, labels, ignore_index=-100)
    ref_loss.backward()

    assert abs(float(true_loss) - float(ref_loss)) < 1e-5
    rp = dict(ref.named_parameters())
    for name, p in cp_stage.named_parameters():
        if p.grad is None:
            continue
        assert torch.allclose(p.grad, rp[name].grad, atol=2e-5), (
            name, (p.grad - rp[name].grad).abs().max()
        )


def test_gpt2_context_parallel_training_step():
    run_distributed(_run_gpt2_cp, 2)


def test_gpt2_context_parallel_cp4():
    run_distributed(_run_gpt2_cp, 4)


def _run_gpt2_cp_tp(rank, world):
    """CP x TP composition: heads sharded by tp, sequence by cp; grads of
    the TP shards must match slices of the full-model gradients."""
    import torch.distributed as dist

    from test_gpt2_pipeline import _shard_full_stage_sd

    from quintnet_amd import init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import (
        DataParallel,
        DistributedConfig,
        cp_causal_lm_loss,
        scatter_clm_targets,
        scatter_to_context,
    )

    pg = init_process_groups("cpu", [2, 2], ["dp", "tp"])  # dp axis = cp here
    cp_group = pg.get_group("dp")
    cp_rank, cp_size = pg.dp_rank, pg.dp_size

    torch.manual_seed(41)
    cfg = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    full = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    for p in full.parameters():
        dist.broadcast(p.data, src=0)

    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=pg.get_group("tp"),
                      cp_group=cp_group)
    stage.load_state_dict(
        _shard_full_stage_sd(full.state_dict(), cfg, 0, 1, pg.tp_rank,
                             pg.tp_size, stage),
        strict=False,
    )
    model = DataParallel(stage, DistributedConfig(cp_rank, cp_size, cp_group))

    ids = torch.randint(0, 96, (2, 32))
    labels = ids.clone()
    dist.broadcast(ids, src=0)
    dist.broadcast(labels, src=0)
    ids_shard = scatter_to_context(ids, cp_group, dim=1)
    tgt_shard = scatter_clm_targets(labels, cp_group)
    logits_shard = model(ids_shard)
    loss_bwd, true_loss = cp_causal_lm_loss(logits_shard, tgt_shard, cp_group)
    loss_bwd.backward()
    model.finalize_gradients()

    out = full(ids)
    ref_loss = causal_lm_loss(out, labels, ignore_index=-100)
    ref_loss.backward()
    assert abs(float(true_loss) - float(ref_loss)) < 1e-5

    # spot-check a TP-sharded grad (c_fc rows) and a replicated one (wte)
    rp = dict(full.named_parameters())
    inner_loc = cfg.n_inner // pg.tp_size
    isl = slice(pg.tp_rank * inner_loc, (pg.tp_rank + 1) * inner_loc)
    got = dict(stage.named_parameters())
    assert torch.allclose(
        got["blocks.0.mlp.c_fc.weight"].grad,
        rp["blocks.0.mlp.c_fc.weight"].grad[isl], atol=2e-5,
    )
    assert torch.allclose(
        got["embedding.wte.weight"].grad,
        rp["embedding.wte.weight"].grad, atol=2e-5,
    )


def test_gpt2_cp_tp_composition():
    run_distributed(_run_gpt2_cp_tp, 4)


def _run_cp_trainer(rank, world):
    """Turnkey CP through GPT2Trainer: mesh with a 'cp' axis, config flag,
    loss matches a plain single-process epoch."""
    import copy

    import torch.distributed as dist

    from quintnet_amd import GPT2Trainer, init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import DataParallel, DistributedConfig
    from quintnet_amd.utils.data import SyntheticCLM
    from torch.utils.data import DataLoader

    pg = init_process_groups("cpu", [world], ["cp"])
    torch.manual_seed(61)
    cfg = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      cp_group=pg.get_group("cp"))
    for p in stage.parameters():
        dist.broadcast(p.data, src=0)
    ref = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    ref.load_state_dict(stage.state_dict())

    model = DataParallel(
        stage, DistributedConfig(rank, world, pg.get_group("cp"))
    )
    ds = SyntheticCLM(n=4, seq_len=32, vocab_size=96, seed=8)
    tcfg = {"num_epochs": 1, "grad_acc_steps": 2, "zero1": False,
            "context_parallel": True, "learning_rate": 1e-3,
            "max_grad_norm": None, "task_type": "clm"}
    tr = GPT2Trainer(model, DataLoader(ds, batch_size=2), None, tcfg, pg)
    hist = tr.fit()

    # oracle: identical single-process epoch (Adam default in base Trainer
    # is replaced by AdamW in GPT2Trainer; replicate exactly)
    opt = torch.optim.AdamW(ref.parameters(), lr=1e-3, weight_decay=0.01)
    dl = DataLoader(ds, batch_size=2)
    tot, steps = 0.0, 0
    accum = 0
    for b in dl:
        loss = causal_lm_loss(ref(b["input_ids"]), b["labels"], ignore_index=-100)
        (loss / 2).backward()
        tot += float(loss.detach()); steps += 1
        accum += 1
        if accum == 2:
            accum = 0
            ref.sync_tied_weights_grad()
            opt.step(); opt.zero_grad()
    assert abs(hist["train_loss"] - tot / steps) < 5e-4, (hist, tot / steps)


def test_gpt2_trainer_context_parallel():
    run_distributed(_run_cp_trainer, 2)


def _run_cp_pp(rank, world):
    """[cp=2, pp=2]: pipeline over sequence shards; inter-stage tensors
    are [B, T/cp, H]; loss trajectory matches single process."""
    import torch.distributed as dist

    from test_gpt2_pipeline import _shard_full_stage_sd

    from quintnet_amd import init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import (
        DataParallel,
        DistributedConfig,
        PipelineDataLoader,
        PipelineParallelWrapper,
        PipelineTrainer,
    )
    from quintnet_amd.utils.data import SyntheticCLM
    from torch.utils.data import DataLoader

    pg = init_process_groups("cpu", [2, 2], ["dp", "pp"])  # dp axis = cp
    cp_group = pg.get_group("dp")
    torch.manual_seed(81)
    seq = 32
    cfg = GPT2Config(vocab_size=96, n_positions=seq, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    full = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    for p in full.parameters():
        dist.broadcast(p.data, src=0)

    stage = GPT2Stage(cfg, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
                      tp_group=None, cp_group=cp_group,
                      tied_group=pg.get_tied_embedding_group())
    stage.load_state_dict(
        _shard_full_stage_sd(full.state_dict(), cfg, pg.pp_rank, pg.pp_size,
                             0, 1, stage),
        strict=False,
    )
    stage.seq_len, stage.hidden_dim = seq, cfg.n_embd
    wrapper = PipelineParallelWrapper(
        stage_module=stage, pp_rank=pg.pp_rank, pp_group=pg.get_group("pp"),
        pp_size=pg.pp_size,
    )
    model = DataParallel(
        wrapper, DistributedConfig(pg.dp_rank, pg.dp_size, cp_group)
    )
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    pt = PipelineTrainer(
        model=model, optimizer=opt, criterion=None,
        pp_rank=pg.pp_rank, pp_size=pg.pp_size, pp_group=pg.get_group("pp"),
        pp_group_ranks=pg.get_group_ranks("pp"), schedule="1f1b",
        task_type="clm", max_grad_norm=None, cp_group=cp_group,
    )
    ds = SyntheticCLM(n=8, seq_len=seq, vocab_size=96, seed=12)
    dl = DataLoader(ds, batch_size=2, shuffle=False)
    loader = PipelineDataLoader(dl, grad_acc_steps=2, task_type="clm")
    shapes = (2, seq // pg.dp_size, cfg.n_embd)  # inter-stage seq SHARD
    losses = []
    for _ in range(2):
        m = pt.train_step(loader, shapes, torch.device("cpu"), torch.float32)
        if pg.pp_rank == pg.pp_size - 1:
            losses.append(m["loss"])

    if pg.pp_rank == pg.pp_size - 1 and pg.dp_rank == 0:
        opt_r = torch.optim.Adam(full.parameters(), lr=1e-3)
        it = iter(PipelineDataLoader(dl, 2, "clm"))
        ref_losses = []
        for _ in range(2):
            opt_r.zero_grad()
            tot = 0.0
            for _ in range(2):
                b = next(it)
                loss = causal_lm_loss(full(b["input_ids"]), b["labels"], ignore_index=-100)
                (loss / 2).backward()
                tot += float(loss.detach())
            full.sync_tied_weights_grad()
            opt_r.step()
            ref_losses.append(tot / 2)
        for a, b in zip(losses, ref_losses):
            assert abs(a - b) < 5e-4, (losses, ref_losses)


def test_gpt2_cp_pp_composition():
    run_distributed(_run_cp_pp, 4)


def _run_cp_pp_trainer(rank, world):
    """Turnkey CP x PP via GPT2Trainer: mesh [cp, pp] + config flag."""
    import torch.distributed as dist

    from quintnet_amd import GPT2Trainer, init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.parallel import (
        DataParallel,
        DistributedConfig,
        PipelineParallelWrapper,
    )
    from quintnet_amd.utils.data import SyntheticCLM
    from torch.utils.data import DataLoader

    pg = init_process_groups("cpu", [2, 2], ["cp", "pp"])
    torch.manual_seed(91)
    seq = 32
    cfg = GPT2Config(vocab_size=96, n_positions=seq, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
                      tp_group=None, cp_group=pg.get_group("cp"),
                      tied_group=pg.get_tied_embedding_group())
    stage.seq_len, stage.hidden_dim = seq, cfg.n_embd
    wrapper = PipelineParallelWrapper(
        stage_module=stage, pp_rank=pg.pp_rank, pp_group=pg.get_group("pp"),
        pp_size=pg.pp_size,
    )
    wrapper.seq_len, wrapper.hidden_dim = seq, cfg.n_embd
    model = DataParallel(
        wrapper,
        DistributedConfig(pg.axis_rank("cp"), pg.axis_size("cp"), pg.get_group("cp")),
    )
    ds = SyntheticCLM(n=8, seq_len=seq, vocab_size=96, seed=13)
    tcfg = {"num_epochs": 1, "grad_acc_steps": 2, "zero1": True,
            "context_parallel": True, "task_type": "clm",
            "max_seq_length": seq,
            "model_config": {"n_embd": cfg.n_embd, "n_positions": seq}}
    tr = GPT2Trainer(model, DataLoader(ds, batch_size=2), None, tcfg, pg)
    hist = tr.fit()
    assert torch.isfinite(torch.tensor(hist["train_loss"]))


def test_gpt2_trainer_cp_pp():
    run_distributed(_run_cp_pp_trainer, 4)


def _run_gpt2_cp_ring(rank, world):
    """cp_ring flavor produces the same logits as the all-gather flavor."""
    import torch.distributed as dist

    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.parallel import scatter_to_context

    torch.manual_seed(51)
    cfg_a = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                       n_head=2, dropout=0.0)
    cfg_r = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                       n_head=2, dropout=0.0, cp_ring=True)
    a = GPT2Stage(cfg_a, pp_rank=0, pp_size=1, tp_group=None,
                  cp_group=dist.group.WORLD)
    for p in a.parameters():
        dist.broadcast(p.data, src=0)
    r = GPT2Stage(cfg_r, pp_rank=0, pp_size=1, tp_group=None,
                  cp_group=dist.group.WORLD)
    r.load_state_dict(a.state_dict())

    ids = torch.randint(0, 96, (2, 32))
    dist.broadcast(ids, src=0)
    shard = scatter_to_context(ids, dist.group.WORLD, dim=1)
    with torch.no_grad():
        la = a(shard)
        lr = r(shard)
    assert torch.allclose(la, lr, atol=1e-4), (la - lr).abs().max()


def test_gpt2_cp_ring_flavor():
    run_distributed(_run_gpt2_cp_ring, 2)


def _run_zigzag(rank, world, causal):
    import torch.distributed as dist

    from quintnet_amd.parallel import (
        zigzag_positions,
        zigzag_ring_attention,
        zigzag_to_context,
    )

    torch.manual_seed(17)
    B, H, T, D = 2, 2, 64, 16
    q = torch.randn(B, H, T, D)
    k = torch.randn(B, H, T, D)
    v = torch.randn(B, H, T, D)
    for t in (q, k, v):
        dist.broadcast(t, src=0)

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref = _ref_attention(qr, kr, vr, causal)
    ref.square().sum().backward()

    g = dist.group.WORLD
    ql = zigzag_to_context(q, g, dim=2).requires_grad_(True)
    kl = zigzag_to_context(k, g, dim=2).requires_grad_(True)
    vl = zigzag_to_context(v, g, dim=2).requires_grad_(True)
    out = zigzag_ring_attention(ql, kl, vl, g, causal=causal)
    pos = zigzag_positions(T, g, q.device)
    assert torch.allclose(out, ref.detach()[:, :, pos].to(out.dtype), atol=1e-5), (
        (out - ref.detach()[:, :, pos]).abs().max()
    )
    out.square().sum().backward()
    assert torch.allclose(ql.grad, qr.grad[:, :, pos], atol=1e-5)
    assert torch.allclose(kl.grad, kr.grad[:, :, pos], atol=1e-5)
    assert torch.allclose(vl.grad, vr.grad[:, :, pos], atol=1e-5)


def _zigzag_causal(rank, world):
    _run_zigzag(rank, world, True)


def _zigzag_full(rank, world):
    _run_zigzag(rank, world, False)


def test_zigzag_ring_cp2_causal():
    run_distributed(_zigzag_causal, 2)


def test_zigzag_ring_cp4_causal():
    run_distributed(_zigzag_causal, 4)


def test_zigzag_ring_cp2_noncausal():
    run_distributed(_zigzag_full, 2)


def _run_gpt2_cp_zigzag(rank, world):
    """End-to-end ZIGZAG-CP GPT-2 step == full-sequence gradients exactly
    (balanced chunk map: positions, attention blocks and loss shards all
    follow the zigzag layout)."""
    import copy

    import torch.distributed as dist

    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import (
        DataParallel,
        DistributedConfig,
        cp_causal_lm_loss,
        zigzag_clm_targets,
        zigzag_to_context,
    )

    torch.manual_seed(33)
    cfg = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0, cp_zigzag=True)
    full_cfg = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                          n_head=2, dropout=0.0)
    full = GPT2Stage(full_cfg, pp_rank=0, pp_size=1, tp_group=None)
    for p in full.parameters():
        dist.broadcast(p.data, src=0)
    ref = copy.deepcopy(full)

    cp_stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                         cp_group=dist.group.WORLD)
    cp_stage.load_state_dict(full.state_dict())
    model = DataParallel(
        cp_stage, DistributedConfig(rank, world, dist.group.WORLD)
    )

    ids = torch.randint(0, 96, (2, 32))
    labels = ids.clone()
    dist.broadcast(ids, src=0)
    dist.broadcast(labels, src=0)

    ids_shard = zigzag_to_context(ids, dist.group.WORLD, dim=1)
    tgt_shard = zigzag_clm_targets(labels, dist.group.WORLD)
    logits_shard = model(ids_shard)
    loss_bwd, true_loss = cp_causal_lm_loss(
        logits_shard, tgt_shard, dist.group.WORLD
    )
    loss_bwd.backward()
    model.finalize_gradients()

    out = ref(ids)
    ref_loss = causal_lm_loss(out, labels, ignore_index=-100)
    ref_loss.backward()

    assert abs(float(true_loss) - float(ref_loss)) < 1e-5
    rp = dict(ref.named_parameters())
    for name, p in cp_stage.named_parameters():
        if p.grad is None:
            continue
        assert torch.allclose(p.grad, rp[name].grad, atol=2e-5), (
            name, (p.grad - rp[name].grad).abs().max())


def test_gpt2_cp_zigzag_cp2():
    run_distributed(_run_gpt2_cp_zigzag, 2)


def test_gpt2_cp_zigzag_cp4():
    run_distributed(_run_gpt2_cp_zigzag, 4)


def _run_cp_trainer_zigzag(rank, world, vocab_pad_to=0):
    """GPT2Trainer with a cp_zigzag config: the trainer must scatter
    ZIGZAG shards (chunks (r, 2cp-1-r)) to match the stage's position
    math; loss matches the plain single-process epoch.  vocab_pad_to
    composes the padded-vocab layout with CP."""
    import torch.distributed as dist
    from torch.utils.data import DataLoader

    from quintnet_amd import GPT2Trainer, init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import DataParallel, DistributedConfig
    from quintnet_amd.utils.data import SyntheticCLM

    pg = init_process_groups("cpu", [world], ["cp"])
    torch.manual_seed(62)
    cfg = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0, cp_zigzag=True,
                     vocab_pad_to=vocab_pad_to)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      cp_group=pg.get_group("cp"))
    for p in stage.parameters():
        dist.broadcast(p.data, src=0)
    ref = GPT2Stage(GPT2Config(vocab_size=96, n_positions=32, n_embd=32,
                               n_layer=2, n_head=2, dropout=0.0,
                               vocab_pad_to=vocab_pad_to))
    ref.load_state_dict(stage.state_dict())

    model = DataParallel(
        stage, DistributedConfig(rank, world, pg.get_group("cp"))
    )
    ds = SyntheticCLM(n=4, seq_len=32, vocab_size=96, seed=8)
    tcfg = {"num_epochs": 1, "grad_acc_steps": 2, "zero1": False,
            "context_parallel": True, "learning_rate": 1e-3,
            "max_grad_norm": None, "task_type": "clm"}
    tr = GPT2Trainer(model, DataLoader(ds, batch_size=2), None, tcfg, pg)
    hist = tr.fit()

    opt = torch.optim.AdamW(ref.parameters(), lr=1e-3, weight_decay=0.01)
    dl = DataLoader(ds, batch_size=2)
    tot, steps, accum = 0.0, 0, 0
    for b in dl:
        loss = causal_lm_loss(ref(b["input_ids"]), b["labels"], ignore_index=-100)
        (loss / 2).backward()
        tot += float(loss.detach()); steps += 1
        accum += 1
        if accum == 2:
            accum = 0
            opt.step(); opt.zero_grad()
    assert abs(hist["train_loss"] - tot / steps) < 5e-4, (hist, tot / steps)


def test_gpt2_trainer_context_parallel_zigzag():
    run_distributed(_run_cp_trainer_zigzag, 2)


def test_gpt2_trainer_zigzag_padded_vocab():
    run_distributed(_run_cp_trainer_zigzag, 2, 64)
