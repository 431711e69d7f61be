"""Mesh + process-group tests over spawned gloo (reference tests/test_mesh.py)."""

from conftest import run_distributed


def _check_2x2(rank, world):
    from quintnet_amd import init_process_groups

    pg = init_process_groups("cpu", [2, 2], ["dp", "tp"])
    assert pg.world_size == 4
    assert pg.dp_size == 2 and pg.tp_size == 2
    coords = pg.get_coordinates_tensor_search(rank)
    assert pg.mesh[tuple(coords)].item() == rank
    # dp groups: {0,2},{1,3}; tp groups: {0,1},{2,3} for mesh arange(4).view(2,2)
    tp_ranks = pg.get_group_ranks("tp")
    dp_ranks = pg.get_group_ranks("dp")
    assert rank in tp_ranks and rank in dp_ranks
    assert len(tp_ranks) == 2 and len(dp_ranks) == 2
    expected_tp = [rank - rank % 2, rank - rank % 2 + 1]
    assert tp_ranks == expected_tp


def _check_2x2x2(rank, world):
    from quintnet_amd import init_process_groups

    pg = init_process_groups("cpu", [2, 2, 2], ["dp", "tp", "pp"])
    coords = pg.get_coordinates_tensor_search(rank)
    d, t, p = coords
    assert pg.mesh[d, t, p].item() == rank
    assert pg.dp_rank == d and pg.tp_rank == t and pg.pp_rank == p
    # pp axis is last: pp pair = {rank with p=0, rank with p=1}
    assert pg.get_group_ranks("pp") == [rank - p, rank - p + 1]
    # tied embedding group exists for pp=2
    assert pg.get_tied_embedding_group() is not None


def test_mesh_2x2():
    run_distributed(_check_2x2, 4)


def test_mesh_2x2x2():
    run_distributed(_check_2x2x2, 8)
