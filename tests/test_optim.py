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
