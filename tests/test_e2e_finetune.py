"""End-to-end CLI round trip: 8-rank [2,2,2] gpt2_finetune -> shard
checkpoints -> merge_checkpoints.py -> HF-format keys.  (This exact flow
caught the local_module-prefix checkpoint bug.)"""

import os
import subprocess
import sys

import torch
import yaml


import pytest


@pytest.mark.slow
@pytest.mark.parametrize("mesh,nproc", [([2, 2, 2], 8), ([1, 2, 2], 4)])
def test_finetune_merge_roundtrip(tmp_path, mesh, nproc):
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ckdir = tmp_path / "ck"
    cfg = {
        "mesh_dim": mesh,
        "mesh_name": ["dp", "tp", "pp"],
        "batch_size": 2,
        "max_seq_length": 16,
        "num_epochs": 1,
        "grad_acc_steps": 2,
        "learning_rate": 1e-4,
        "n_train": 8,
        "checkpoint_dir": str(ckdir),
        "model_config": {
            "vocab_size": 128, "n_positions": 16, "n_embd": 32,
            "n_layer": 2, "n_head": 2, "dropout": 0.0,
        },
    }
    cfg_path = tmp_path / "tiny.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))

    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
         "--master-port", str(29781 + nproc), "-m", "examples.gpt2_finetune",
         "--config", str(cfg_path)],
        cwd=root, env=env, capture_output=True, text=True, timeout=420,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert (ckdir / "final_model_pp0_tp0.pt").exists()
    assert (ckdir / "final_model_pp1_tp1.pt").exists()

    merged = tmp_path / "merged.pt"
    r = subprocess.run(
        [sys.executable, "merge_checkpoints.py", "--input-dir", str(ckdir),
         "--output", str(merged), "--prefix", "final_model"],
        cwd=root, capture_output=True, text=True, timeout=120,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    sd = torch.load(merged, map_location="cpu", weights_only=False)["model_state_dict"]
    assert sd["transformer.wte.weight"].shape == (128, 32)
    assert sd["transformer.h.1.attn.c_attn.weight"].shape == (32, 96)
    assert "transformer.ln_f.weight" in sd and "lm_head.weight" in sd

    # numeric round trip: the merged model's val ppl must be in the same
    # ballpark as the training run's (weights survived shard + merge)
    r = subprocess.run(
        [sys.executable, "-m", "examples.verify_model", "--checkpoint",
         str(merged), "--n", "32"],
        cwd=root, capture_output=True, text=True, timeout=180,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert "ppl:" in r.stdout
    ppl = float(r.stdout.split("ppl:")[1].split()[0])
    assert ppl < 200, r.stdout  # random-init would be ~vocab_size
