"""Direct autograd round-trip tests for the TP/SP collective mappings.

Reference: tensor_parallel/mappings.py:18-546 (copy/reduce/scatter/gather
to/from model-parallel region, sequence-parallel gather/reduce-scatter,
all_to_all).  gloo world-2; every forward is checked against a local
reference built from the full tensor, and every backward against the
matching adjoint collective.
"""
import torch
import torch.distributed as dist

import hetu_galvatron_amd.runtime.tensor_parallel.mappings as M


def _full(n=4, h=6, seed=7):
    torch.manual_seed(seed)
    return torch.randn(n, 2, h)


def _worker(rank, world):
    dist.init_process_group("gloo", rank=rank, world_size=world)
    g = dist.group.WORLD
    out = {}

    # copy fwd = identity; bwd = allreduce of grads
    x = _full().requires_grad_(True)
    y = M.copy_to_tensor_model_parallel_region(x, g)
    torch.testing.assert_close(y, x)
    (y * (rank + 1)).sum().backward()
    # grads: sum over ranks of (rank+1) = 1+2 = 3
    torch.testing.assert_close(x.grad, torch.full_like(x, 3.0))

    # reduce fwd = allreduce; bwd = identity
    x = _full(seed=8 + rank).requires_grad_(True)
    y = M.reduce_from_tensor_model_parallel_region(x, g)
    ref = _full(seed=8) + _full(seed=9)
    torch.testing.assert_close(y, ref)
    y.sum().backward()
    torch.testing.assert_close(x.grad, torch.ones_like(x))

    # scatter (split last dim) <-> gather round trip with autograd
    full = _full(seed=11).requires_grad_(True)
    mine = M.scatter_to_tensor_model_parallel_region(full, g)
    h = full.shape[-1] // world
    torch.testing.assert_close(
        mine, full[..., rank * h:(rank + 1) * h])
    back = M.gather_from_tensor_model_parallel_region(mine, g)
    torch.testing.assert_close(back, full)
    back.sum().backward()
    # gather bwd splits, scatter bwd gathers -> identity on the full grad
    torch.testing.assert_close(full.grad, torch.ones_like(full))

    # SP gather (first dim) fwd; bwd = reduce-scatter
    shard = _full(seed=20 + rank).requires_grad_(True)
    gathered = M.gather_from_sequence_parallel_region(shard, g)
    ref = torch.cat([_full(seed=20), _full(seed=21)], dim=0)
    torch.testing.assert_close(gathered, ref)
    go = torch.ones_like(gathered)
    gathered.backward(go)
    # tensor_parallel_output_grad=True: bwd reduce-scatters — every rank
    # feeds a full ones-grad, so each shard grad sums to world (=2)
    torch.testing.assert_close(shard.grad, torch.full_like(shard, 2.0))

    # SP reduce-scatter fwd; bwd = all-gather
    x = _full(seed=30 + rank).requires_grad_(True)
    y = M.reduce_scatter_to_sequence_parallel_region(x, g)
    full_sum = _full(seed=30) + _full(seed=31)
    n = x.shape[0] // world
    torch.testing.assert_close(y, full_sum[rank * n:(rank + 1) * n])
    y.sum().backward()
    torch.testing.assert_close(x.grad, torch.ones_like(x))

    # all_to_all scatter dim 0 / gather dim -1 round trip
    x = _full(seed=40 + rank)
    y = M.all_to_all(x, g, scatter_dim=0, gather_dim=2)
    assert y.shape == (x.shape[0] // world, 2, x.shape[2] * world)
    z = M.all_to_all(y, g, scatter_dim=2, gather_dim=0)
    torch.testing.assert_close(z, x)
    out["ok"] = True
    return out


def test_mappings_world2():
    from tests.utils.dist_helpers import run_distributed
    res = run_distributed(_worker, world_size=2)
    assert all(r["ok"] for r in res)
