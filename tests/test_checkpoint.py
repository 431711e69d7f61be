"""Checkpoint save → merge → reload round-trip (the reference had no
such test) + staged distributed loading equivalence."""

import os

import torch

from conftest import run_distributed


def _save_and_merge(rank, world, tmpdir):
    import torch.distributed as dist

    from quintnet_amd import init_process_groups
    from quintnet_amd.checkpoint import merge_checkpoints, save_sharded_checkpoint
    from quintnet_amd.models import GPT2Config, GPT2Stage

    pg = init_process_groups("cpu", [2, 2], ["tp", "pp"])
    cfg = GPT2Config(
        n_embd=32, n_layer=4, n_head=2, vocab_size=64, n_positions=32, dropout=0.0
    )
    torch.manual_seed(100 + rank)  # distinct shards per rank
    stage = GPT2Stage(
        cfg,
        pp_rank=pg.pp_rank,
        pp_size=pg.pp_size,
        tp_group=pg.get_group("tp"),
        tied_group=pg.get_tied_embedding_group(),
    )
    save_sharded_checkpoint(stage, tmpdir, name="final_model", pg_manager=pg)
    dist.barrier()
    if rank == 0:
        out = os.path.join(tmpdir, "merged.pt")
        merge_checkpoints(tmpdir, out, prefix="final_model")
        merged = torch.load(out, map_location="cpu", weights_only=False)["model_state_dict"]
        # HF-format keys present with full shapes
        assert merged["transformer.wte.weight"].shape == (64, 32)
        assert merged["transformer.h.0.attn.c_attn.weight"].shape == (32, 96)  # Conv1D [in, 3H]
        assert merged["transformer.h.3.mlp.c_proj.weight"].shape == (128, 32)
        assert merged["transformer.ln_f.weight"].shape == (32,)
        assert merged["lm_head.weight"].shape == (64, 32)
        assert "transformer.h.2.ln_1.weight" in merged  # PP offset remap worked


def test_save_merge_roundtrip(tmp_path):
    run_distributed(_save_and_merge, 4, str(tmp_path))


def test_staged_load_matches_full_model(tmp_path):
    """single-rank staged load of an HF-style checkpoint == direct load."""
    from quintnet_amd.checkpoint.distributed_loading import load_gpt2_distributed
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import cross_entropy

    cfg = GPT2Config(
        n_embd=32, n_layer=2, n_head=2, vocab_size=64, n_positions=32, dropout=0.0
    )
    torch.manual_seed(0)
    # build an HF-style state dict (Conv1D layout: [in, out])
    hf = {}
    hf["wte.weight"] = torch.randn(64, 32)
    hf["wpe.weight"] = torch.randn(32, 32)
    for i in range(2):
        hf[f"h.{i}.ln_1.weight"] = torch.randn(32)
        hf[f"h.{i}.ln_1.bias"] = torch.randn(32)
        hf[f"h.{i}.ln_2.weight"] = torch.randn(32)
        hf[f"h.{i}.ln_2.bias"] = torch.randn(32)
        hf[f"h.{i}.attn.c_attn.weight"] = torch.randn(32, 96)
        hf[f"h.{i}.attn.c_attn.bias"] = torch.randn(96)
        hf[f"h.{i}.attn.c_proj.weight"] = torch.randn(32, 32)
        hf[f"h.{i}.attn.c_proj.bias"] = torch.randn(32)
        hf[f"h.{i}.mlp.c_fc.weight"] = torch.randn(32, 128)
        hf[f"h.{i}.mlp.c_fc.bias"] = torch.randn(128)
        hf[f"h.{i}.mlp.c_proj.weight"] = torch.randn(128, 32)
        hf[f"h.{i}.mlp.c_proj.bias"] = torch.randn(32)
    hf["ln_f.weight"] = torch.randn(32)
    hf["ln_f.bias"] = torch.randn(32)
    path = str(tmp_path / "ckpt.pt")
    torch.save(hf, path)

    state = load_gpt2_distributed(path, cfg, pp_rank=0, pp_size=1, tp_rank=0, tp_size=1)
    stage = GPT2Stage.from_sharded_state_dict(cfg, state, pp_rank=0, pp_size=1)
    ids = torch.randint(0, 64, (2, 8))
    logits = stage(ids)
    assert logits.shape == (2, 8, 64)
    loss = cross_entropy(logits[:, :-1], ids[:, 1:])
    assert torch.isfinite(loss)

    # manual reference forward of layer 0 ln_1 + qkv to validate slicing
    x = stage.embedding(ids)
    import torch.nn.functional as F

    ln = F.layer_norm(x, (32,), hf["h.0.ln_1.weight"], hf["h.0.ln_1.bias"])
    qkv_ref = ln @ hf["h.0.attn.c_attn.weight"] + hf["h.0.attn.c_attn.bias"]
    qkv_ours = stage.blocks[0].attn.c_attn(stage.blocks[0].ln_1(x))
    assert torch.allclose(qkv_ours, qkv_ref, atol=1e-4), (qkv_ours - qkv_ref).abs().max()


def test_resume_roundtrip(tmp_path):
    """Mid-training resume: save model+optimizer, reload, training
    continues with identical state (beyond-reference capability —
    SURVEY.md §5.4 notes the reference never loads optimizer state)."""
    import copy

    from quintnet_amd.checkpoint import load_sharded_checkpoint, save_sharded_checkpoint
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.optim import ZeroRedundancyAdamW

    torch.manual_seed(0)
    cfg = GPT2Config(n_embd=32, n_layer=2, n_head=2, vocab_size=64, n_positions=32, dropout=0.0)
    stage = GPT2Stage(cfg)
    opt = ZeroRedundancyAdamW(stage.parameters(), lr=1e-3)
    ids = torch.randint(0, 64, (2, 16))
    for _ in range(3):
        loss = causal_lm_loss(stage(ids), ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
    save_sharded_checkpoint(stage, str(tmp_path), name="ck", optimizer=opt)

    stage2 = GPT2Stage(cfg)  # fresh random init
    opt2 = ZeroRedundancyAdamW(stage2.parameters(), lr=1e-3)
    load_sharded_checkpoint(stage2, str(tmp_path), name="ck", optimizer=opt2)
    assert opt2.step_count == 3
    assert torch.allclose(opt2.master, opt.master)
    assert torch.allclose(opt2.exp_avg, opt.exp_avg)

    # both continue identically
    for m, o in ((stage, opt), (stage2, opt2)):
        loss = causal_lm_loss(m(ids), ids)
        loss.backward()
        o.step()
        o.zero_grad()
    for p1, p2 in zip(stage.parameters(), stage2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_unwrap_peels_pipeline_wrapper():
    """Trainer._save_checkpoint must save stage-canonical keys even when the
    stage is nested in PipelineParallelWrapper (regression: shards were saved
    with a ``local_module.`` prefix, which the merge CLI silently passed
    through unmapped)."""
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.trainer import _unwrap

    cfg = GPT2Config(vocab_size=64, n_positions=32, n_embd=16, n_layer=2, n_head=2)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)

    class FakePPWrapper(torch.nn.Module):
        def __init__(self, m):
            super().__init__()
            self.local_module = m

    class FakeDP(torch.nn.Module):
        def __init__(self, m):
            super().__init__()
            self.module = m

    wrapped = FakeDP(FakePPWrapper(stage))
    inner = _unwrap(wrapped)
    assert inner is stage
    assert all(not k.startswith("local_module.") for k in inner.state_dict())
    assert any(k.startswith("embedding.wte") for k in inner.state_dict())


def test_auto_resume(tmp_path):
    """resume_from='auto': fresh start without shards, resume when the
    checkpoint_dir holds them (elastic-restart convenience)."""
    from torch.utils.data import DataLoader

    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.utils.data import SyntheticCLM

    cfg = GPT2Config(vocab_size=64, n_positions=16, n_embd=16, n_layer=1, n_head=2)
    ds = SyntheticCLM(n=4, seq_len=16, vocab_size=64, seed=0)
    tcfg = {"num_epochs": 1, "grad_acc_steps": 2, "zero1": False,
            "checkpoint_dir": str(tmp_path), "resume_from": "auto",
            "task_type": "clm"}
    # run 1: nothing to resume -> trains and saves
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    GPT2Trainer(stage, DataLoader(ds, batch_size=2), None, tcfg, None).fit()
    assert (tmp_path / "final_model_pp0_tp0.pt").exists()
    w0 = stage.state_dict()["embedding.wte.weight"].clone()
    # run 2: auto-resume loads the saved weights into a fresh model
    stage2 = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    tr2 = GPT2Trainer(stage2, DataLoader(ds, batch_size=2), None, tcfg, None)
    assert torch.allclose(stage2.state_dict()["embedding.wte.weight"], w0)


def _run_moe_ckpt(rank, world, tmpdir):
    """MoE [ep] mesh: per-ep shard files don't collide; save->resume
    round-trips the rank-local expert weights exactly."""
    import os

    import torch.distributed as dist

    from quintnet_amd import init_process_groups
    from quintnet_amd.checkpoint import (
        load_sharded_checkpoint,
        save_sharded_checkpoint,
    )
    from quintnet_amd.models import GPT2Config, GPT2Stage

    pg = init_process_groups("cpu", [world], ["ep"])
    torch.manual_seed(rank)  # deliberately DIFFERENT experts per rank
    cfg = GPT2Config(vocab_size=64, n_positions=16, n_embd=16, n_layer=1,
                     n_head=2, dropout=0.0, n_experts=2, moe_top_k=1)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      ep_group=pg.get_group("ep"))
    save_sharded_checkpoint(stage, tmpdir, name="moe", pg_manager=pg)
    assert os.path.exists(os.path.join(tmpdir, f"moe_pp0_tp0_ep{rank}.pt"))

    want = stage.blocks[0].mlp.experts[0].fc1.weight.clone()
    fresh = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      ep_group=pg.get_group("ep"))
    load_sharded_checkpoint(fresh, tmpdir, name="moe", pg_manager=pg)
    assert torch.allclose(fresh.blocks[0].mlp.experts[0].fc1.weight, want)


def test_moe_checkpoint_shards(tmp_path):
    run_distributed(_run_moe_ckpt, 2, str(tmp_path))


def test_save_best_keeps_best_val_shards(tmp_path):
    """save_best: a separate *_best shard set tracks the lowest val
    loss across epochs (base trainer semantics through GPT2Trainer)."""
    import os

    import torch
    from torch.utils.data import DataLoader

    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.utils.data import SyntheticCLM

    cfg = GPT2Config(vocab_size=64, n_positions=32, n_embd=16, n_layer=1,
                     n_head=2, dropout=0.0)
    ds = SyntheticCLM(n=4, seq_len=16, vocab_size=64, seed=0)
    vs = SyntheticCLM(n=2, seq_len=16, vocab_size=64, seed=7)
    tr = GPT2Trainer(
        GPT2Stage(cfg), DataLoader(ds, batch_size=2),
        DataLoader(vs, batch_size=2),
        {"num_epochs": 2, "grad_acc_steps": 1, "zero1": False,
         "checkpoint_dir": str(tmp_path), "save_best": True,
         "learning_rate": 1e-3},
        None,
    )
    tr.fit()
    assert os.path.exists(tmp_path / "final_model_pp0_tp0.pt")
    assert os.path.exists(tmp_path / "final_model_best_pp0_tp0.pt")
    assert tr._best_metric is not None


def test_early_stopping_breaks_epoch_loop(tmp_path):
    import torch
    from torch.utils.data import DataLoader

    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.utils.data import SyntheticCLM

    cfg = GPT2Config(vocab_size=64, n_positions=32, n_embd=16, n_layer=1,
                     n_head=2, dropout=0.0)
    ds = SyntheticCLM(n=4, seq_len=16, vocab_size=64, seed=0)
    vs = SyntheticCLM(n=2, seq_len=16, vocab_size=64, seed=7)
    tr = GPT2Trainer(
        GPT2Stage(cfg), DataLoader(ds, batch_size=2),
        DataLoader(vs, batch_size=2),
        {"num_epochs": 50, "grad_acc_steps": 1, "zero1": False,
         # lr=0: val loss can never improve -> stop after patience
         "learning_rate": 0.0, "early_stop_patience": 2,
         "metrics_file": str(tmp_path / "m.jsonl")},
        None,
    )
    tr.fit()
    n_epochs_run = len((tmp_path / "m.jsonl").read_text().splitlines())
    assert n_epochs_run <= 4, n_epochs_run  # 1 best + 2 flat, not 50


def test_resume_continues_lr_schedule(tmp_path):
    """After auto-resume the cosine schedule restarts from the ZeRO
    optimizer's checkpointed step counter, not from zero."""
    import torch
    from torch.utils.data import DataLoader

    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.utils.data import SyntheticCLM

    cfg = GPT2Config(vocab_size=64, n_positions=32, n_embd=16, n_layer=1,
                     n_head=2, dropout=0.0)
    ds = SyntheticCLM(n=4, seq_len=16, vocab_size=64, seed=0)
    base = {"num_epochs": 1, "grad_acc_steps": 1, "zero1": True,
            "learning_rate": 1e-3, "lr_schedule": "cosine",
            "total_steps": 10, "checkpoint_dir": str(tmp_path)}
    torch.manual_seed(0)
    tr1 = GPT2Trainer(GPT2Stage(cfg), DataLoader(ds, batch_size=2), None,
                      dict(base), None)
    tr1.fit()  # 2 optimizer steps, then shards saved
    assert tr1.optimizer.step_count == 2

    torch.manual_seed(0)
    tr2 = GPT2Trainer(GPT2Stage(cfg), DataLoader(ds, batch_size=2), None,
                      dict(base, resume_from="auto"), None)
    assert tr2.optimizer.step_count == 2  # state restored
    sched = tr2._build_lr_schedule()
    assert sched._step == 2  # curve continues, not restarted
    assert sched.lr_at(2) < sched.lr_at(0)
