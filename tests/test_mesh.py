"""Mesh + process-group tests over spawned gloo.

Full 8-rank coordinate/membership matrix, matching the reference's
tests/test_mesh.py:36-140 coverage (every rank's coords, every axis
group's exact membership, subgroup sizes, cross-axis disjointness).
"""

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
    """The full [2,2,2] membership matrix (reference test_mesh.py:36-140).

    mesh = arange(8).view(2,2,2) with axes (dp, tp, pp):
      rank = dp*4 + tp*2 + pp
    """
    from quintnet_amd import init_process_groups

    pg = init_process_groups("cpu", [2, 2, 2], ["dp", "tp", "pp"])

    # -- coordinates: closed-form for every rank, not just this one
    for r in range(8):
        d, t, p = r // 4, (r // 2) % 2, r % 2
        assert pg.get_coordinates_tensor_search(r) == [d, t, p], r
        assert pg.mesh[d, t, p].item() == r

    d, t, p = rank // 4, (rank // 2) % 2, rank % 2
    assert pg.dp_rank == d and pg.tp_rank == t and pg.pp_rank == p
    assert (pg.dp_size, pg.tp_size, pg.pp_size) == (2, 2, 2)

    # -- exact axis-group membership for this rank
    assert pg.get_group_ranks("pp") == [rank - p, rank - p + 1]
    assert pg.get_group_ranks("tp") == sorted({rank - 2 * t, rank - 2 * t + 2})
    assert pg.get_group_ranks("dp") == sorted({rank - 4 * d, rank - 4 * d + 4})

    # -- every rank is in exactly one group per axis; groups partition
    # the world (verified via the closed forms above on all 8 ranks)
    for r in range(8):
        rd, rt, rp = r // 4, (r // 2) % 2, r % 2
        pp_row = [r - rp, r - rp + 1]
        tp_row = sorted({r - 2 * rt, r - 2 * rt + 2})
        dp_row = sorted({r - 4 * rd, r - 4 * rd + 4})
        assert r in pp_row and r in tp_row and r in dp_row
        # cross-axis groups intersect only at r itself
        assert set(pp_row) & set(tp_row) == {r}
        assert set(pp_row) & set(dp_row) == {r}
        assert set(tp_row) & set(dp_row) == {r}

    # -- distributed sanity: the axis subgroup actually communicates
    # among exactly its members (rank-sum over the group)
    import torch
    import torch.distributed as dist

    for axis in ("dp", "tp", "pp"):
        x = torch.tensor([float(rank)])
        dist.all_reduce(x, group=pg.get_group(axis))
        assert float(x) == sum(pg.get_group_ranks(axis)), axis

    # tied embedding group exists for pp=2 and equals the pp pair
    assert pg.get_tied_embedding_group() is not None

    # duplicate pp communicators for interleaved 1F1B direction split
    groups = pg.get_all_groups()
    assert groups.get("pp_fwd") is not None and groups.get("pp_bwd") is not None
    assert pg.get_group_ranks("pp_fwd") == pg.get_group_ranks("pp")


def _check_4x2_tied(rank, world):
    """pp=4: the tied-embedding subgroup must be exactly {first, last}
    of this rank's pp row."""
    from quintnet_amd import init_process_groups

    pg = init_process_groups("cpu", [2, 4], ["dp", "pp"])
    row = pg.get_group_ranks("pp")
    assert len(row) == 4
    tied = pg.get_tied_embedding_group()
    if rank in (row[0], row[-1]):
        assert tied is not None
        import torch
        import torch.distributed as dist

        x = torch.tensor([float(rank)])
        dist.all_reduce(x, group=tied)
        assert float(x) == row[0] + row[-1]
    # middle stages hold no tied weight; their group handle may exist but
    # is never used (sync_tied_weights_grad returns early)


def test_mesh_2x2():
    run_distributed(_check_2x2, 4)


def test_mesh_2x2x2():
    run_distributed(_check_2x2x2, 8)


def test_mesh_tied_pp4():
    run_distributed(_check_4x2_tied, 8)


def test_mesh_shape_validation():
    """Constructor rejects mismatched names/dims without distributed init."""
    import pytest
    import torch

    from quintnet_amd.core.mesh import MeshGenerator

    with pytest.raises(ValueError):
        MeshGenerator("cpu", torch.arange(4).view(2, 2), ["dp"])
