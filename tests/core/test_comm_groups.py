"""Comm-group fabric unit tests (single-process: group membership math only;
multi-process gloo: actual group creation).  Reference spec:
comm_groups.py:69-183 docstring examples."""
import pytest

from hetu_galvatron_amd.config.strategy import LayerStrategy
from hetu_galvatron_amd.core.comm_groups import (
    CommGroupCache, build_stage_coords, gen_layer_comm_groups,
    pp_neighbor_ranks, pp_stage_of_rank,
)


def groups_of(rank, world, strategies):
    layer_groups, cache = gen_layer_comm_groups(strategies, world, rank)
    return layer_groups, cache


def test_coords_consecutive():
    coords = build_stage_coords(8, tsp=2, cp=2, consecutive=True)
    # rank = (dp*cp + cp_idx)*tsp + tp_idx
    assert (coords[0].dp_idx, coords[0].cp_idx, coords[0].tp_idx) == (0, 0, 0)
    assert (coords[1].dp_idx, coords[1].cp_idx, coords[1].tp_idx) == (0, 0, 1)
    assert (coords[2].dp_idx, coords[2].cp_idx, coords[2].tp_idx) == (0, 1, 0)
    assert (coords[4].dp_idx, coords[4].cp_idx, coords[4].tp_idx) == (1, 0, 0)


def test_coords_nonconsecutive():
    coords = build_stage_coords(8, tsp=2, cp=1, consecutive=False)
    # dp fastest: tp groups are strided {0,4},{1,5},...
    assert coords[0].tp_idx == 0 and coords[4].tp_idx == 1
    assert coords[0].dp_idx == 0 and coords[1].dp_idx == 1


def test_tp_dp_groups_tp2_dp4():
    s = LayerStrategy(pp_deg=1, tp=2, dp=4)
    lg, _ = groups_of(rank=0, world=8, strategies=[s])
    g = lg[0]
    assert list(g.tp_group.ranks) == [0, 1]
    assert list(g.dp_group.ranks) == [0, 2, 4, 6]
    assert list(g.sdp_group.ranks) == [0, 2, 4, 6]
    lg3, _ = groups_of(rank=3, world=8, strategies=[s])
    assert list(lg3[0].tp_group.ranks) == [2, 3]
    assert list(lg3[0].dp_group.ranks) == [1, 3, 5, 7]


def test_nonconsecutive_tp():
    s = LayerStrategy(pp_deg=1, tp=2, dp=4, tp_consecutive=False)
    lg, _ = groups_of(rank=0, world=8, strategies=[s])
    assert list(lg[0].tp_group.ranks) == [0, 4]
    assert list(lg[0].dp_group.ranks) == [0, 1, 2, 3]


def test_cp_and_sdp():
    s = LayerStrategy(pp_deg=1, tp=2, cp=2, dp=2)
    lg, _ = groups_of(rank=0, world=8, strategies=[s])
    g = lg[0]
    assert list(g.tp_group.ranks) == [0, 1]
    assert list(g.cp_group.ranks) == [0, 2]
    assert list(g.dp_group.ranks) == [0, 4]
    assert list(g.sdp_group.ranks) == [0, 2, 4, 6]   # dp x cp, same tp idx
    assert list(g.tsp_cp_group.ranks) == [0, 1, 2, 3]


def test_pp_groups():
    s = LayerStrategy(pp_deg=2, tp=2, dp=2)
    lg, _ = groups_of(rank=5, world=8, strategies=[s])
    g = lg[0]
    # stage 1 holds ranks 4..7
    assert list(g.tp_group.ranks) == [4, 5]
    assert list(g.dp_group.ranks) == [5, 7]
    assert pp_stage_of_rank(5, 8, 2) == 1
    assert pp_neighbor_ranks(5, 8, 2) == (1, None)
    assert pp_neighbor_ranks(1, 8, 2) == (None, 5)


def test_group_cache_shared_across_layers():
    s = LayerStrategy(pp_deg=1, tp=2, dp=4)
    lg, cache = groups_of(rank=0, world=8, strategies=[s, s, s, s])
    assert lg[0].tp_group is lg[3].tp_group
    n_unique = len(cache)
    lg2, cache2 = groups_of(rank=0, world=8, strategies=[s])
    assert len(cache2) == n_unique  # one layer creates the same set


def test_mixed_layer_strategies():
    s1 = LayerStrategy(pp_deg=1, tp=4, dp=2)
    s2 = LayerStrategy(pp_deg=1, tp=1, sp=2, dp=4)  # ulysses layer
    lg, _ = groups_of(rank=0, world=8, strategies=[s1, s2])
    assert list(lg[0].tp_group.ranks) == [0, 1, 2, 3]
    assert list(lg[1].sp_group.ranks) == [0, 1]
    assert lg[1].strategy.use_ulysses


def test_moe_groups():
    s = LayerStrategy(pp_deg=1, tp=2, dp=4, ep=2)
    lg, _ = groups_of(rank=0, world=8, strategies=[s])
    g = lg[0]
    assert g.ep_group is not None and g.edp_group is not None
    assert len(g.ep_group.ranks) == 2
    assert len(g.edp_group.ranks) == 2
    # ep x edp spans the sdp domain
    all_ranks = set(g.ep_group.ranks) | set(g.edp_group.ranks)
    assert all_ranks <= set(g.sdp_group.ranks)


def _dist_group_creation(rank, world):
    import torch.distributed as dist
    from hetu_galvatron_amd.core.initialize import _initialize_distributed
    _initialize_distributed("gloo")
    import torch
    s = LayerStrategy(pp_deg=1, tp=2, dp=world // 2)
    lg, _ = gen_layer_comm_groups([s], world, rank)
    g = lg[0]
    # allreduce over the tp group: ranks in the same group share a sum
    t = torch.tensor([float(rank)])
    dist.all_reduce(t, group=g.tp_group.group)
    return t.item()


@pytest.mark.distributed
def test_group_creation_gloo_world2():
    from tests.utils import run_distributed
    res = run_distributed(_dist_group_creation, world_size=2)
    assert res == [1.0, 1.0]  # ranks 0+1 in the same tp group


@pytest.mark.distributed
def test_group_creation_gloo_world4():
    from tests.utils import run_distributed
    res = run_distributed(_dist_group_creation, world_size=4)
    assert res == [1.0, 1.0, 5.0, 5.0]
