"""DP gradient-sync oracle over spawned gloo (fixes the reference's
broken tests/test_data_parallel.py — its DDP never reduced)."""

import torch

from conftest import run_distributed


def _grad_sync_oracle(rank, world):
    import torch.nn as nn

    from quintnet_amd.parallel import DataParallel

    torch.manual_seed(1234)  # same init everywhere (broadcast also enforces)
    model = nn.Sequential(nn.Linear(10, 8), nn.ReLU(), nn.Linear(8, 5))
    ref = nn.Sequential(nn.Linear(10, 8), nn.ReLU(), nn.Linear(8, 5))
    ref.load_state_dict(model.state_dict())

    ddp = DataParallel(model)
    # per-rank data
    g = torch.Generator().manual_seed(100 + rank)
    x = torch.randn(4, 10, generator=g)
    y = torch.randn(4, 5, generator=g)
    out = ddp(x)
    loss = ((out - y) ** 2).mean()
    loss.backward()
    ddp.finalize_gradients()

    # reference: grads averaged over ALL ranks' data, computed locally
    ref.zero_grad()
    for r in range(world):
        gr = torch.Generator().manual_seed(100 + r)
        xr = torch.randn(4, 10, generator=gr)
        yr = torch.randn(4, 5, generator=gr)
        lr = ((ref(xr) - yr) ** 2).mean() / world
        lr.backward()

    for (n, p), (rn, rp) in zip(ddp.module.named_parameters(), ref.named_parameters()):
        assert p.grad is not None, n
        assert torch.allclose(p.grad, rp.grad, atol=1e-6), (n, (p.grad - rp.grad).abs().max())


def _grad_identity_across_ranks(rank, world):
    import torch.distributed as dist
    import torch.nn as nn

    from quintnet_amd.parallel import DataParallel

    torch.manual_seed(7)
    ddp = DataParallel(nn.Linear(10, 5))
    g = torch.Generator().manual_seed(rank)
    x = torch.randn(3, 10, generator=g)
    ddp(x).sum().backward()
    ddp.finalize_gradients()
    w = ddp.module.weight.grad.clone()
    gathered = [torch.empty_like(w) for _ in range(world)]
    dist.all_gather(gathered, w)
    for gw in gathered:
        assert torch.allclose(gw, w, atol=1e-6)


def _no_sync_and_reset(rank, world):
    import torch.nn as nn

    from quintnet_amd.parallel import DataParallel

    torch.manual_seed(3)
    ddp = DataParallel(nn.Linear(4, 4))
    x = torch.randn(2, 4)
    # two iterations: the bucket counter must reset (reference bug §8.2)
    for _ in range(2):
        ddp(x).sum().backward()
        ddp.finalize_gradients()
        assert ddp.module.weight.grad is not None
        ddp.zero_grad()
        assert float(ddp.module.weight.grad.abs().sum()) == 0.0


def test_grad_sync_oracle():
    run_distributed(_grad_sync_oracle, 2)


def test_grad_identity():
    run_distributed(_grad_identity_across_ranks, 2)


def test_counter_reset_two_iters():
    run_distributed(_no_sync_and_reset, 2)


def test_local_backend_single_process():
    """DataParallel constructible without distributed init."""
    import torch.nn as nn

    from quintnet_amd.parallel import DataParallel

    ddp = DataParallel(nn.Linear(4, 2))
    x = torch.randn(3, 4)
    ddp(x).sum().backward()
    ddp.finalize_gradients()
    assert ddp.module.weight.grad is not None


def test_ddp_factories():
    """C14 parity: create_local_ddp / create_distributed_ddp."""
    import torch.nn as nn

    from quintnet_amd.parallel import DataParallel, create_local_ddp

    m = nn.Linear(8, 8)
    ddp = create_local_ddp(m)
    assert isinstance(ddp, DataParallel)
    out = ddp(torch.randn(2, 8))
    out.sum().backward()
    ddp.finalize_gradients()
    assert m.weight.grad is not None


def test_plain_trainer_grad_accumulation_oracle():
    """Base Trainer classification path honors grad_acc_steps: one
    optimizer step per window, mean-of-window gradient (was silently
    ignored before — stepped per micro-batch)."""
    import copy

    import torch

    from quintnet_amd.models import Model as ViT
    from quintnet_amd.trainer import Trainer
    from quintnet_amd.utils.data import SyntheticMNIST

    torch.manual_seed(5)
    m = ViT(depth=2, hidden_dim=32, n_heads=2)
    ref = copy.deepcopy(m)
    ds = SyntheticMNIST(n=8, seed=3)
    dl = torch.utils.data.DataLoader(ds, batch_size=2, shuffle=False)
    tr = Trainer(m, dl, None,
                 {"num_epochs": 1, "grad_acc_steps": 2, "max_grad_norm": None,
                  "learning_rate": 1e-3},
                 None)
    tr.fit()

    opt = torch.optim.Adam(ref.parameters(), lr=1e-3)
    crit = torch.nn.CrossEntropyLoss()
    it = iter(torch.utils.data.DataLoader(ds, batch_size=2, shuffle=False))
    for _ in range(2):  # 4 batches / window 2 = 2 steps
        opt.zero_grad()
        for _ in range(2):
            b = next(it)
            (crit(ref(b["images"]), b["labels"]) / 2).backward()
        opt.step()
    for (k, p), (_, r) in zip(m.named_parameters(), ref.named_parameters()):
        assert torch.allclose(p, r, atol=1e-6), k
