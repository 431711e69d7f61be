"""TP numerical equivalence vs unsharded nn.Linear over spawned gloo
(reference tests/test_tensor_parallel.py pattern)."""

import torch

from conftest import run_distributed


def _column_parallel_equiv(rank, world):
    import torch.distributed as dist
    import torch.nn as nn

    from quintnet_amd.parallel import ColumnParallelLinear

    torch.manual_seed(0)
    lin = nn.Linear(16, 8)
    # broadcast the full weights so every rank shards the same linear
    for t in (lin.weight, lin.bias):
        dist.broadcast(t.data, src=0)
    col = ColumnParallelLinear.from_linear(lin, tp_group=dist.group.WORLD, gather_output=True)
    x = torch.randn(4, 16)
    dist.broadcast(x, src=0)
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y_ref = lin(x1)
    y_tp = col(x2)
    assert torch.allclose(y_tp, y_ref, atol=1e-5), (y_tp - y_ref).abs().max()
    # backward equivalence on input grads
    y_ref.sum().backward()
    y_tp.sum().backward()
    assert torch.allclose(x2.grad, x1.grad, atol=1e-5)
    # weight shard grad == corresponding slice of full grad
    sl = slice(rank * col.out_per_rank, (rank + 1) * col.out_per_rank)
    assert torch.allclose(col.weight.grad, lin.weight.grad[sl], atol=1e-5)


def _row_parallel_equiv(rank, world):
    import torch.distributed as dist
    import torch.nn as nn

    from quintnet_amd.parallel import RowParallelLinear

    torch.manual_seed(1)
    lin = nn.Linear(16, 8)
    for t in (lin.weight, lin.bias):
        dist.broadcast(t.data, src=0)
    row = RowParallelLinear.from_linear(lin, tp_group=dist.group.WORLD, input_is_parallel=False)
    x = torch.randn(4, 16)
    dist.broadcast(x, src=0)
    y_ref = lin(x)
    y_tp = row(x)
    assert torch.allclose(y_tp, y_ref, atol=1e-5), (y_tp - y_ref).abs().max()


def _vocab_parallel_equiv(rank, world):
    import torch.distributed as dist
    import torch.nn as nn

    from quintnet_amd.parallel import VocabParallelEmbedding

    torch.manual_seed(2)
    full = nn.Embedding(32, 8)
    dist.broadcast(full.weight.data, src=0)
    vp = VocabParallelEmbedding(32, 8, tp_group=dist.group.WORLD)
    with torch.no_grad():
        vp.weight.copy_(full.weight[vp.vocab_start : vp.vocab_end])
    ids = torch.randint(0, 32, (3, 5))
    dist.broadcast(ids, src=0)
    assert torch.allclose(vp(ids), full(ids), atol=1e-6)


def _rewriter(rank, world):
    import torch.distributed as dist
    import torch.nn as nn

    from quintnet_amd.parallel import ColumnParallelLinear, apply_tensor_parallel

    torch.manual_seed(3)
    model = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 8), nn.Linear(8, 3))
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    import copy

    ref = copy.deepcopy(model)
    apply_tensor_parallel(model, tp_size=world, tp_rank=rank, tp_group=dist.group.WORLD)
    # divisible layers replaced, non-divisible (out=3) left alone
    assert isinstance(model[0], ColumnParallelLinear)
    assert isinstance(model[3], nn.Linear)
    x = torch.randn(2, 8)
    dist.broadcast(x, src=0)
    assert torch.allclose(model(x), ref(x), atol=1e-5)


def test_column_parallel():
    run_distributed(_column_parallel_equiv, 2)


def test_row_parallel():
    run_distributed(_row_parallel_equiv, 2)


def test_vocab_parallel():
    run_distributed(_vocab_parallel_equiv, 2)


def test_rewriter():
    run_distributed(_rewriter, 2)
