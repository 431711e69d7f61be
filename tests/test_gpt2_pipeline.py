"""GPT-2 CLM through the full pipeline stack over spawned gloo:
PP=2 stages + tied-weight grad sync + GPT2Trainer epoch, vs a
single-process oracle."""

import pytest
import torch

from conftest import run_distributed


def _gpt2_pp2(rank, world):
    import copy

    import torch.distributed as dist

    from quintnet_amd import GPT2Trainer, init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import PipelineParallelWrapper
    from quintnet_amd.utils.data import SyntheticCLM

    pg = init_process_groups("cpu", [1, 1, 2], ["dp", "tp", "pp"])
    cfg = GPT2Config(
        n_embd=32, n_layer=4, n_head=2, vocab_size=96, n_positions=32, dropout=0.0
    )
    seq = 16
    torch.manual_seed(21)
    # build the full single-process model for the oracle, then per-rank stages
    full = GPT2Stage(cfg, pp_rank=0, pp_size=1)
    for p in full.parameters():
        dist.broadcast(p.data, src=0)

    stage = GPT2Stage(
        cfg,
        pp_rank=pg.pp_rank,
        pp_size=pg.pp_size,
        tied_group=pg.get_tied_embedding_group(),
    )
    # copy matching weights from the full model
    with torch.no_grad():
        sd = full.state_dict()
        offset = stage.layer_distribution[pg.pp_rank][0]
        tgt = {}
        for i, _ in enumerate(stage.my_layers):
            for k in (
                "ln_1.weight", "ln_1.bias", "ln_2.weight", "ln_2.bias",
                "attn.c_attn.weight", "attn.c_attn.bias", "attn.c_proj.weight",
                "attn.c_proj.bias", "mlp.c_fc.weight", "mlp.c_fc.bias",
                "mlp.c_proj.weight", "mlp.c_proj.bias",
            ):
                tgt[f"blocks.{i}.{k}"] = sd[f"blocks.{i + offset}.{k}"]
        if stage.is_first_stage:
            tgt["embedding.wte.weight"] = sd["embedding.wte.weight"]
            tgt["embedding.wpe.weight"] = sd["embedding.wpe.weight"]
        if stage.is_last_stage and not stage.is_first_stage:
            tgt["ln_f.weight"] = sd["ln_f.weight"]
            tgt["ln_f.bias"] = sd["ln_f.bias"]
            tgt["lm_head"] = sd["embedding.wte.weight"].clone()  # tied copy
        stage.load_state_dict(tgt, strict=False)

    wrapper = PipelineParallelWrapper(
        stage_module=stage, pp_rank=pg.pp_rank, pp_group=pg.get_group("pp"),
        pp_size=pg.pp_size,
    )
    wrapper.seq_len, wrapper.hidden_dim = seq, cfg.n_embd

    ds = SyntheticCLM(n=16, seq_len=seq, vocab_size=96, seed=3)
    dl = torch.utils.data.DataLoader(ds, batch_size=2, shuffle=False)
    tcfg = {
        "batch_size": 2,
        "num_epochs": 1,
        "learning_rate": 1e-3,
        "grad_acc_steps": 2,
        "max_grad_norm": None,
        "schedule": "1f1b",
        "zero1": True,
        "max_seq_length": seq,
        "model_config": {"n_embd": cfg.n_embd},
    }
    trainer = GPT2Trainer(wrapper, dl, None, tcfg, pg)
    metrics = trainer.fit()
    assert torch.isfinite(torch.tensor(metrics["train_loss"]))

    # oracle: single process, ZeRO-1 AdamW (dp=1 degenerate), same stream
    if rank == 1:  # last stage holds the loss metric
        from quintnet_amd.optim import ZeroRedundancyAdamW

        opt = ZeroRedundancyAdamW(full.parameters(), lr=1e-3, weight_decay=0.01)
        num_steps = len(dl) // 2
        it = iter(dl)
        step_losses = []
        for _ in range(num_steps):
            tot = 0.0
            for _ in range(2):
                try:
                    b = next(it)
                except StopIteration:
                    it = iter(dl)
                    b = next(it)
                loss = causal_lm_loss(full(b["input_ids"]), b["labels"])
                (loss / 2).backward()
                tot += float(loss.detach())
            opt.step()
            opt.zero_grad()
            step_losses.append(tot / 2)
        ref = sum(step_losses) / len(step_losses)
        assert abs(metrics["train_loss"] - ref) < 2e-3, (metrics["train_loss"], ref)


def test_gpt2_pp2_matches_single_process():
    run_distributed(_gpt2_pp2, 2, timeout=300)


def test_gpt2_stage_tied_weights_pp1():
    """pp=1: lm_head IS wte (true tying), logits = h @ wte^T."""
    from quintnet_amd.models import GPT2Config, GPT2Stage

    cfg = GPT2Config(n_embd=16, n_layer=1, n_head=2, vocab_size=64, n_positions=16, dropout=0.0)
    s = GPT2Stage(cfg)
    ids = torch.randint(0, 64, (2, 8))
    logits = s(ids)
    logits.sum().backward()
    assert s.embedding.wte.weight.grad is not None
    assert s.lm_head is None


def _shard_full_stage_sd(sd, cfg, pp_rank, pp_size, tp_rank, tp_size, stage):
    """Slice a full (pp=1, tp=1) GPT2Stage state dict down to this rank's
    shard — same layouts as checkpoint/distributed_loading.py (per-head
    fused QKV for column-parallel c_attn, input-dim slices for the
    row-parallel projections)."""
    import torch

    h_loc = cfg.n_embd // tp_size
    inner_loc = cfg.n_inner // tp_size
    sl = slice(tp_rank * h_loc, (tp_rank + 1) * h_loc)
    isl = slice(tp_rank * inner_loc, (tp_rank + 1) * inner_loc)
    off = stage.layer_distribution[pp_rank][0]
    out = {}
    for i, _ in enumerate(stage.my_layers):
        src = f"blocks.{i + off}"
        dst = f"blocks.{i}"
        for ln in ("ln_1", "ln_2"):
            out[f"{dst}.{ln}.weight"] = sd[f"{src}.{ln}.weight"]
            out[f"{dst}.{ln}.bias"] = sd[f"{src}.{ln}.bias"]
        qw, kw, vw = sd[f"{src}.attn.c_attn.weight"].chunk(3, dim=0)
        qb, kb, vb = sd[f"{src}.attn.c_attn.bias"].chunk(3, dim=0)
        out[f"{dst}.attn.c_attn.weight"] = torch.cat([qw[sl], kw[sl], vw[sl]], 0)
        out[f"{dst}.attn.c_attn.bias"] = torch.cat([qb[sl], kb[sl], vb[sl]], 0)
        out[f"{dst}.attn.c_proj.weight"] = sd[f"{src}.attn.c_proj.weight"][:, sl].contiguous()
        out[f"{dst}.attn.c_proj.bias"] = sd[f"{src}.attn.c_proj.bias"]
        out[f"{dst}.mlp.c_fc.weight"] = sd[f"{src}.mlp.c_fc.weight"][isl].contiguous()
        out[f"{dst}.mlp.c_fc.bias"] = sd[f"{src}.mlp.c_fc.bias"][isl].contiguous()
        out[f"{dst}.mlp.c_proj.weight"] = sd[f"{src}.mlp.c_proj.weight"][:, isl].contiguous()
        out[f"{dst}.mlp.c_proj.bias"] = sd[f"{src}.mlp.c_proj.bias"]
    if pp_rank == 0:
        out["embedding.wte.weight"] = sd["embedding.wte.weight"]
        out["embedding.wpe.weight"] = sd["embedding.wpe.weight"]
    if pp_rank == pp_size - 1:
        out["ln_f.weight"] = sd["ln_f.weight"]
        out["ln_f.bias"] = sd["ln_f.bias"]
        if pp_size > 1:
            out["lm_head"] = sd["embedding.wte.weight"].clone()
    return out


def _gpt2_3d(rank, world, vocab=128, vocab_pad_to=0):
    """The bench/finetune assembly at mesh [2,2,2] (dp x tp x pp) with
    ZeRO-1: tiny GPT-2, one epoch, loss must match a single-process run
    (TP/PP are exact decompositions; both DP replicas see the same
    stream here).  vocab_pad_to>0 runs the padded-vocab layout the GPU
    bench defaults to (bench.py QN_VOCAB_PAD) through the full 3D
    assembly."""
    import copy

    import torch.distributed as dist

    from quintnet_amd import GPT2Trainer, init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import (
        DataParallel,
        DistributedConfig,
        PipelineParallelWrapper,
    )
    from quintnet_amd.utils.data import SyntheticCLM

    pg = init_process_groups("cpu", [2, 2, 2], ["dp", "tp", "pp"])
    torch.manual_seed(77)
    cfg = GPT2Config(vocab_size=vocab, n_positions=16, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0, vocab_pad_to=vocab_pad_to)
    stage = GPT2Stage(
        cfg, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
        tp_group=pg.get_group("tp"), tied_group=pg.get_tied_embedding_group(),
    )
    # all ranks must agree on the FULL model weights: broadcast a master
    # copy and load the per-rank (pp, tp) shards from it
    master = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    for p in master.parameters():
        dist.broadcast(p.data, src=0)
    stage.load_state_dict(
        _shard_full_stage_sd(master.state_dict(), cfg, pg.pp_rank, pg.pp_size,
                             pg.tp_rank, pg.tp_size, stage),
        strict=False,
    )
    stage.seq_len, stage.hidden_dim = 16, cfg.n_embd
    pmodel = PipelineParallelWrapper(
        stage_module=stage, pp_rank=pg.pp_rank, pp_group=pg.get_group("pp"),
        pp_size=pg.pp_size,
    )
    pmodel.seq_len, pmodel.hidden_dim = 16, cfg.n_embd
    pmodel = DataParallel(
        pmodel, DistributedConfig(pg.dp_rank, pg.dp_size, pg.get_group("dp"))
    )
    tcfg = {"batch_size": 2, "num_epochs": 1, "learning_rate": 1e-3,
            "grad_acc_steps": 2, "max_grad_norm": None, "zero1": True,
            "task_type": "clm", "max_seq_length": 16,
            "model_config": {"n_embd": cfg.n_embd, "n_positions": 16}}
    ds = SyntheticCLM(n=8, seq_len=16, vocab_size=vocab, seed=9)  # same per dp!
    dl = torch.utils.data.DataLoader(ds, batch_size=2, shuffle=False)
    trainer = GPT2Trainer(pmodel, dl, None, tcfg, pg)
    metrics = trainer.fit()
    assert torch.isfinite(torch.tensor(metrics["train_loss"]))

    if rank == 0:
        ref = master
        opt = torch.optim.AdamW(ref.parameters(), lr=1e-3, weight_decay=0.01)
        it = iter(dl)
        step_losses = []
        for _ in range(2):  # 8 samples / (2 micro x 2 acc) = 2 steps
            opt.zero_grad()
            tot = 0.0
            for _ in range(2):
                b = next(it)
                loss = causal_lm_loss(ref(b["input_ids"]), b["labels"], ignore_index=-100)
                (loss / 2).backward()
                tot += float(loss.detach())
            ref.sync_tied_weights_grad()
            opt.step()
            step_losses.append(tot / 2)
        ref_mean = sum(step_losses) / len(step_losses)
        assert abs(metrics["train_loss"] - ref_mean) < 5e-3, (
            metrics["train_loss"], ref_mean,
        )


@pytest.mark.slow
def test_gpt2_3d_zero1_matches_single_process():
    run_distributed(_gpt2_3d, 8, timeout=300)


@pytest.mark.slow
def test_gpt2_3d_padded_vocab():
    """mesh [2,2,2] with the padded-vocab layout (the GPU bench's
    default config) — exact vs the single-process oracle."""
    run_distributed(_gpt2_3d, 8, 100, 64, timeout=300)


def _gpt2_3d_sp(rank, world):
    """Same [2,2,2] assembly with Megatron sequence parallelism enabled:
    inter-stage activations are sequence shards; loss must still match."""
    _gpt2_3d_impl(rank, world, sequence_parallel=True)


def _gpt2_3d_impl(rank, world, sequence_parallel=False):
    import torch.distributed as dist

    from quintnet_amd import GPT2Trainer, init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import (
        DataParallel,
        DistributedConfig,
        PipelineParallelWrapper,
    )
    from quintnet_amd.utils.data import SyntheticCLM

    pg = init_process_groups("cpu", [2, 2, 2], ["dp", "tp", "pp"])
    torch.manual_seed(78)
    cfg = GPT2Config(vocab_size=128, n_positions=16, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0, sequence_parallel=sequence_parallel)
    full_cfg = GPT2Config(vocab_size=128, n_positions=16, n_embd=32, n_layer=2,
                          n_head=2, dropout=0.0)
    stage = GPT2Stage(
        cfg, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
        tp_group=pg.get_group("tp"), tied_group=pg.get_tied_embedding_group(),
    )
    master = GPT2Stage(full_cfg, pp_rank=0, pp_size=1, tp_group=None)
    for p in master.parameters():
        dist.broadcast(p.data, src=0)
    stage.load_state_dict(
        _shard_full_stage_sd(master.state_dict(), cfg, pg.pp_rank, pg.pp_size,
                             pg.tp_rank, pg.tp_size, stage),
        strict=False,
    )
    seq = 16
    stage.seq_len, stage.hidden_dim = seq, cfg.n_embd
    pmodel = PipelineParallelWrapper(
        stage_module=stage, pp_rank=pg.pp_rank, pp_group=pg.get_group("pp"),
        pp_size=pg.pp_size,
    )
    pmodel.seq_len, pmodel.hidden_dim = seq, cfg.n_embd
    pmodel = DataParallel(
        pmodel, DistributedConfig(pg.dp_rank, pg.dp_size, pg.get_group("dp"))
    )
    tcfg = {"batch_size": 2, "num_epochs": 1, "learning_rate": 1e-3,
            "grad_acc_steps": 2, "max_grad_norm": None, "zero1": True,
            "task_type": "clm", "max_seq_length": seq,
            "model_config": {"n_embd": cfg.n_embd, "n_positions": seq,
                             "sequence_parallel": sequence_parallel}}
    ds = SyntheticCLM(n=8, seq_len=seq, vocab_size=128, seed=9)
    dl = torch.utils.data.DataLoader(ds, batch_size=2, shuffle=False)
    trainer = GPT2Trainer(pmodel, dl, None, tcfg, pg)
    metrics = trainer.fit()
    assert torch.isfinite(torch.tensor(metrics["train_loss"]))

    if rank == 0:
        opt = torch.optim.AdamW(master.parameters(), lr=1e-3, weight_decay=0.01)
        it = iter(dl)
        step_losses = []
        for _ in range(2):
            opt.zero_grad()
            tot = 0.0
            for _ in range(2):
                b = next(it)
                loss = causal_lm_loss(master(b["input_ids"]), b["labels"], ignore_index=-100)
                (loss / 2).backward()
                tot += float(loss.detach())
            master.sync_tied_weights_grad()
            opt.step()
            step_losses.append(tot / 2)
        ref_mean = sum(step_losses) / len(step_losses)
        assert abs(metrics["train_loss"] - ref_mean) < 5e-3, (
            metrics["train_loss"], ref_mean,
        )


@pytest.mark.slow
def test_gpt2_3d_sequence_parallel():
    run_distributed(_gpt2_3d_sp, 8, timeout=300)
