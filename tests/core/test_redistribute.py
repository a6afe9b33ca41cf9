"""Direct redistribute() round-trip tests between activation layouts.

Reference: tests/core/test_redistributed.py (per-layer tp-list changes ->
relocation correctness) — here at the primitive level: for layout pairs
A→B on gloo world 4, the full canonical tensor reassembled from B's
shards must equal the original, B→A must invert exactly, and the
backward must be the exact adjoint (grad of identity round trip = ones).
"""
import pytest
import torch

from hetu_galvatron_amd.config.strategy import LayerStrategy


def _shard_of(full, groups, rank, batch_global):
    """Slice the canonical [S,B,h] tensor to `rank`'s shard under layout."""
    from hetu_galvatron_amd.runtime.redistribute import natural_rows
    s = groups.strategy
    c = groups.coord_of(rank)
    S = full.shape[0]
    rows = natural_rows(S, s.cp, s.tp_sp, c.cp_idx, c.tp_idx, full.device)
    b_loc = batch_global // s.dp
    return full[rows, c.dp_idx * b_loc:(c.dp_idx + 1) * b_loc].contiguous()


def _worker(rank, world, sa_kw, sb_kw):
    import torch.distributed as dist
    from hetu_galvatron_amd.core.comm_groups import gen_layer_comm_groups
    from hetu_galvatron_amd.runtime.redistribute import redistribute

    dist.init_process_group("gloo", rank=rank, world_size=world)
    sa = LayerStrategy(**sa_kw)
    sb = LayerStrategy(**sb_kw)
    (ga, gb), _ = gen_layer_comm_groups([sa, sb], world, rank)
    torch.manual_seed(11)  # identical canonical tensor on every rank
    S, B, h = 16, 4, 6
    full = torch.randn(S, B, h)

    xa = _shard_of(full, ga, rank, B).requires_grad_(True)
    xb = redistribute(xa, ga, gb, B)
    torch.testing.assert_close(xb, _shard_of(full, gb, rank, B))
    back = redistribute(xb, gb, ga, B)
    torch.testing.assert_close(back, xa.detach())
    # adjoint: identity round trip -> grad of ones everywhere
    back.sum().backward()
    torch.testing.assert_close(xa.grad, torch.ones_like(xa))
    return True


PAIRS = [
    # tp2xdp2 -> cp2xdp2 (seq layout flips tp->zigzag)
    (dict(tp=2, dp=2), dict(cp=2, dp=2)),
    # pure dp -> pure tp (allgather over everyone, reslice)
    (dict(dp=4), dict(tp=4)),
    # tp2xcp2 -> dp4
    (dict(tp=2, cp=2), dict(dp=4)),
    # ulysses sp2 x dp2 -> tp2 x dp2 (same seq sharding degree, different
    # group semantics)
    (dict(sp=2, dp=2), dict(tp=2, dp=2)),
]


@pytest.mark.distributed
@pytest.mark.parametrize("sa_kw,sb_kw", PAIRS)
def test_redistribute_roundtrip_world4(sa_kw, sb_kw):
    from tests.utils import run_distributed
    res = run_distributed(_worker, world_size=4, args=(sa_kw, sb_kw))
    assert all(res)
