"""Strategy registry + 2D coordinator compositions over spawned gloo
(the reference only ever tested 1D paths)."""

import pytest
import torch

from conftest import run_distributed


def test_registry_names():
    from quintnet_amd.strategy import _REGISTRY, get_strategy

    assert set(_REGISTRY) == {"dp", "tp", "pp", "dp_tp", "dp_pp", "tp_pp", "3d"}
    with pytest.raises(ValueError):
        get_strategy("nope", None, {})


def _compose(rank, world, name, mesh, axes):
    import copy

    import torch.distributed as dist

    from quintnet_amd import get_strategy, init_process_groups
    from quintnet_amd.models import Model

    pg = init_process_groups("cpu", mesh, axes)
    torch.manual_seed(5)
    model = Model(hidden_dim=32, n_heads=2, depth=4)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    ref = copy.deepcopy(model)
    pmodel = get_strategy(name, pg, {}).apply(model)

    x = torch.randn(2, 1, 28, 28)
    dist.broadcast(x, src=0)
    if pg.pp_size > 1:
        # forward via manual stage chaining for the check
        h = x
        for r in range(pg.pp_size):
            if pg.pp_rank == r:
                out = pmodel(h if r == 0 else recv)
                if r < pg.pp_size - 1:
                    dist.send(out.detach(), dst=pg.get_group_ranks("pp")[r + 1], group=None)
            if pg.pp_rank == r + 1:
                shape = (2, 50, 32)
                recv = torch.empty(shape)
                dist.recv(recv, src=pg.get_group_ranks("pp")[r])
        if pg.pp_rank == pg.pp_size - 1:
            assert torch.allclose(out, ref(x), atol=1e-4), (out - ref(x)).abs().max()
    else:
        out = pmodel(x)
        assert torch.allclose(out, ref(x), atol=1e-4), (out - ref(x)).abs().max()


def _dp_tp(rank, world):
    _compose(rank, world, "dp_tp", [2, 2], ["dp", "tp"])


def _tp_pp(rank, world):
    _compose(rank, world, "tp_pp", [2, 2], ["tp", "pp"])


def _dp_pp(rank, world):
    _compose(rank, world, "dp_pp", [2, 2], ["dp", "pp"])


def test_dp_tp_composition():
    run_distributed(_dp_tp, 4)


def test_tp_pp_composition():
    run_distributed(_tp_pp, 4)


def test_dp_pp_composition():
    run_distributed(_dp_pp, 4)


def _pp_interleaved_strategy(rank, world):
    import torch.distributed as dist

    from quintnet_amd import get_strategy, init_process_groups
    from quintnet_amd.models import Model
    from quintnet_amd.parallel import InterleavedPipelineWrapper

    pg = init_process_groups("cpu", [world], ["pp"])
    torch.manual_seed(5)
    model = Model(hidden_dim=32, n_heads=2, depth=8)
    pmodel = get_strategy("pp", pg, {"schedule": "interleaved", "num_chunks": 2}).apply(model)
    assert isinstance(pmodel, InterleavedPipelineWrapper)
    assert pmodel.num_chunks == 2
    # chunked forward works
    x = torch.randn(2, 50, 32)
    y = pmodel(x, chunk_id=1)
    assert y.shape[0] == 2


def test_pp_strategy_interleaved_wrapping():
    run_distributed(_pp_interleaved_strategy, 2)
