"""Vocab-parallel cross-entropy numerics vs torch.nn.functional.

Reference behavior: tensor_parallel/triton_cross_entropy.py (max / sum-exp /
target-pick with TP all-reduces between passes).  CPU path exercises the
reference_ops fallbacks; the HIP kernels are compared on GPU in
tests/ops/test_gpu_kernels.py.
"""
import torch
import torch.distributed as dist
import torch.nn.functional as F

from hetu_galvatron_amd.runtime.tensor_parallel.cross_entropy import (
    vocab_parallel_cross_entropy)


def _ref(logits, target):
    l = logits.detach().clone().requires_grad_(True)
    loss = F.cross_entropy(l.float(), target, reduction="none")
    loss.sum().backward()
    return loss.detach(), l.grad.detach()


def test_vocab_ce_single_process_matches_torch():
    torch.manual_seed(0)
    n, v = 64, 203
    logits = torch.randn(n, v, requires_grad=True)
    target = torch.randint(0, v, (n,))
    loss = vocab_parallel_cross_entropy(logits, target, None)
    loss.sum().backward()
    ref_loss, ref_grad = _ref(logits, target)
    torch.testing.assert_close(loss, ref_loss, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(logits.grad, ref_grad, atol=1e-5, rtol=1e-5)


def test_vocab_ce_nd_target_shape():
    torch.manual_seed(1)
    s, b, v = 7, 3, 50
    logits = torch.randn(s, b, v)
    target = torch.randint(0, v, (s, b))
    loss = vocab_parallel_cross_entropy(logits, target, None)
    assert loss.shape == (s, b)
    ref_loss, _ = _ref(logits.reshape(-1, v), target.reshape(-1))
    torch.testing.assert_close(loss.reshape(-1), ref_loss,
                               atol=1e-5, rtol=1e-5)


def _shard_worker(rank, world):
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(42)  # same full tensors on every rank
    n, v = 32, 128
    full = torch.randn(n, v)
    target = torch.randint(0, v, (n,))
    v_loc = v // world
    shard = full[:, rank * v_loc:(rank + 1) * v_loc] \
        .detach().clone().requires_grad_(True)
    loss = vocab_parallel_cross_entropy(shard, target,
                                        dist.group.WORLD)
    loss.sum().backward()
    ref_loss, ref_grad = _ref(full, target)
    torch.testing.assert_close(loss, ref_loss, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(
        shard.grad, ref_grad[:, rank * v_loc:(rank + 1) * v_loc],
        atol=1e-5, rtol=1e-5)
    return float(loss.sum())


def test_vocab_ce_sharded_world2_matches_unsharded():
    from tests.utils.dist_helpers import run_distributed
    res = run_distributed(_shard_worker, world_size=2)
    assert abs(res[0] - res[1]) < 1e-4  # identical loss on both ranks
