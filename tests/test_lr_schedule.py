"""LR schedules (optim/schedule.py): shape of the curve, optimizer
plumbing for both APIs (ZeRO ``optimizer.lr`` and torch param_groups),
and end-to-end through GPT2Trainer."""

import math

import torch

from quintnet_amd.optim import LRSchedule


class _FakeZeroOpt:
    def __init__(self, lr):
        self.lr = lr

    @property
    def param_groups(self):
        return [{"params": [], "lr": self.lr}]


def test_warmup_then_cosine():
    opt = _FakeZeroOpt(1e-3)
    s = LRSchedule(opt, base_lr=1e-3, total_steps=100, warmup_steps=10,
                   kind="cosine", min_lr=1e-5)
    lrs = [s.step() for _ in range(100)]
    # warmup: linear 0 -> base over 10 steps
    assert abs(lrs[0] - 1e-4) < 1e-12
    assert abs(lrs[9] - 1e-3) < 1e-12
    # peak right after warmup; approaches min at the end (exactly min
    # at step == total_steps)
    assert lrs[10] <= 1e-3 + 1e-12 and lrs[-1] < 3e-5
    assert abs(s.lr_at(100) - 1e-5) < 1e-10
    # monotone decay after warmup
    assert all(a >= b - 1e-12 for a, b in zip(lrs[10:], lrs[11:]))
    # midpoint of cosine ~ average of base and min
    mid = s.lr_at(10 + (100 - 10) // 2)
    assert abs(mid - (1e-3 + 1e-5) / 2) < 2e-5
    assert abs(opt.lr - lrs[-1]) < 1e-12  # applied to the ZeRO API


def test_linear_and_constant():
    opt = _FakeZeroOpt(2e-3)
    lin = LRSchedule(opt, 2e-3, total_steps=20, kind="linear")
    vals = [lin.step() for _ in range(20)]
    assert abs(vals[0] - 2e-3) < 1e-12 and vals[-1] < 2e-4
    const = LRSchedule(opt, 2e-3, total_steps=20, warmup_steps=5,
                       kind="constant")
    cv = [const.step() for _ in range(20)]
    assert cv[4] == 2e-3 and all(v == 2e-3 for v in cv[5:])


def test_torch_optimizer_param_groups_updated():
    p = torch.nn.Parameter(torch.zeros(3))
    opt = torch.optim.AdamW([p], lr=1e-3)
    s = LRSchedule(opt, 1e-3, total_steps=10, kind="linear")
    s.step()
    s.step()
    assert opt.param_groups[0]["lr"] == s.lr_at(1)


def test_gpt2_trainer_applies_schedule():
    from quintnet_amd import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.utils.data import SyntheticCLM

    cfg = GPT2Config(n_embd=32, n_layer=2, n_head=2, vocab_size=64,
                     n_positions=32, dropout=0.0)
    torch.manual_seed(0)
    model = GPT2Stage(cfg)
    ds = SyntheticCLM(n=8, seq_len=16, vocab_size=64, seed=1)
    dl = torch.utils.data.DataLoader(ds, batch_size=2)
    tcfg = {
        "batch_size": 2, "num_epochs": 1, "learning_rate": 1e-3,
        "grad_acc_steps": 1, "max_grad_norm": None, "zero1": True,
        "task_type": "clm", "max_seq_length": 16,
        "lr_schedule": "cosine", "warmup_steps": 1, "min_lr": 1e-5,
        "model_config": {"n_embd": 32},
    }
    tr = GPT2Trainer(model, dl, None, tcfg, None)
    tr.fit()
    # 4 optimizer steps happened; optimizer.lr must sit on the curve
    assert tr.lr_scheduler is not None
    assert abs(tr.optimizer.lr - tr.lr_scheduler.lr_at(3)) < 1e-12
