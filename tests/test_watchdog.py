"""Hang-detection watchdog: fires on a missed heartbeat, stays quiet
with beats, rearms."""

import time


def test_watchdog_fires_and_rearms(capsys):
    from quintnet_amd.utils import Watchdog

    with Watchdog(timeout_s=0.3) as wd:
        for _ in range(4):
            wd.beat()
            time.sleep(0.05)
        assert not wd.fired
        time.sleep(0.8)  # miss heartbeats
        assert wd.fired


def test_watchdog_quiet_with_beats():
    from quintnet_amd.utils import Watchdog

    with Watchdog(timeout_s=0.5) as wd:
        for _ in range(8):
            wd.beat()
            time.sleep(0.05)
        assert not wd.fired


def test_watchdog_kill_on_hang(tmp_path):
    """kill_on_hang exits the process (code 42) so an elastic agent can
    restart it instead of waiting out the collective timeout."""
    import subprocess
    import sys

    import os

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = tmp_path / "hang.py"
    script.write_text(
        "import sys, time\n"
        "sys.path.insert(0, %r)\n"
        "from quintnet_amd.utils import Watchdog\n"
        "Watchdog(timeout_s=0.4, kill_on_hang=True).start()\n"
        "time.sleep(30)\n" % root
    )
    r = subprocess.run([sys.executable, str(script)], cwd=root,
                       capture_output=True, text=True, timeout=20)
    assert r.returncode == 42, (r.returncode, r.stderr[-500:])
    assert "no heartbeat" in r.stderr


def test_assert_finite_grads_names_offender():
    import pytest
    import torch

    from quintnet_amd.utils.watchdog import assert_finite_grads

    m = torch.nn.Linear(4, 4)
    m.weight.grad = torch.zeros(4, 4)
    m.bias.grad = torch.tensor([0.0, float("nan"), 0.0, 0.0])
    with pytest.raises(FloatingPointError, match="bias"):
        assert_finite_grads(m)
    m.bias.grad.zero_()
    assert_finite_grads(m)  # clean pass


def test_trainer_detect_nan_grads_flag():
    import pytest
    import torch
    from torch.utils.data import DataLoader

    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.utils.data import SyntheticCLM

    cfg = GPT2Config(vocab_size=64, n_positions=32, n_embd=16, n_layer=1,
                     n_head=2, dropout=0.0)
    stage = GPT2Stage(cfg)
    # poison one weight so the first backward produces NaN grads
    with torch.no_grad():
        stage.embedding.wte.weight[0, 0] = float("nan")
    ds = SyntheticCLM(n=2, seq_len=16, vocab_size=64, seed=0)
    tr = GPT2Trainer(stage, DataLoader(ds, batch_size=2), None,
                     {"num_epochs": 1, "grad_acc_steps": 1, "zero1": False,
                      "detect_nan_grads": True, "max_grad_norm": None},
                     None)
    with pytest.raises(FloatingPointError):
        tr.fit()
