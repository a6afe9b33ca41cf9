"""T5 encoder-decoder correctness: single-process training + tp2/zero3
world-2 runs vs the 1-process baseline."""
import os

import pytest
import torch

from hetu_galvatron_amd.config import HybridParallelPlan, load_config

STEPS = 3
TOL = 0.03

BASE = {
    "model": {"model_name": "tiny-t5"},
    "train": {"global_train_batch_size": 4, "train_iters": STEPS, "lr": 1e-3,
              "lr_decay_style": "constant", "distributed_backend": "gloo"},
}


def make_cfg(extra=None):
    import copy
    base = copy.deepcopy(BASE)
    for k, v in (extra or {}).items():
        base.setdefault(k, {}).update(v)
    return load_config(base=base)


def train_steps(model, cfg, steps=STEPS):
    from hetu_galvatron_amd.runtime import (
        get_optimizer_and_param_scheduler, get_train_iterator)
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    it = get_train_iterator(model.cfg, torch.device("cpu"))
    losses = []
    for _ in range(steps):
        opt.zero_grad()
        stats = model.forward_backward(next(it))
        opt.step()
        sched.step()
        losses.append(model.global_loss(stats))
    return losses


_BASELINE = {}


def get_baseline(tmp_dir="/tmp/galvatron_t5_test"):
    if "v" not in _BASELINE:
        from hetu_galvatron_amd.runtime import GalvatronModel
        from hetu_galvatron_amd.runtime.checkpoint.state import (
            canonical_state_from_stage)
        os.makedirs(tmp_dir, exist_ok=True)
        cfg = make_cfg()
        torch.manual_seed(0)
        model = GalvatronModel(cfg)
        state = canonical_state_from_stage(model.stage_model)
        path = os.path.join(tmp_dir, "tiny_t5_state.pt")
        torch.save(state, path)
        _BASELINE["v"] = (train_steps(model, cfg), path)
    return _BASELINE["v"]


def test_t5_overfits_one_batch():
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler, get_train_iterator)
    cfg = make_cfg({"train": {"lr": 5e-3}})
    torch.manual_seed(0)
    device = torch.device("cuda", 0) if torch.cuda.is_available() \
        else torch.device("cpu")
    m = GalvatronModel(cfg, device=device)
    opt, sched = get_optimizer_and_param_scheduler(m.stage_model, cfg)
    it = get_train_iterator(cfg, device)
    batch = next(it)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        st = m.forward_backward(batch)
        opt.step()
        sched.step()
        losses.append(st.loss)
    assert losses[-1] < losses[0] - 0.05, losses


def _t5_worker(rank, world, plan_dict, state_path):
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state

    cfg = make_cfg()
    initialize_galvatron(cfg, backend="gloo")
    plan = HybridParallelPlan.from_config_dict(plan_dict)
    model = GalvatronModel(cfg, plan)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    return train_steps(model, cfg)


@pytest.mark.distributed
@pytest.mark.parametrize("tp,dp_type", [(2, "ddp"), (1, "zero3"),
                                        (1, "zero2")])
def test_t5_dist_vs_baseline(tp, dp_type):
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=4, world_size=2, pp=1, tp=tp, dp_type=dp_type,
        global_bsz=4, chunks=1)
    res = run_distributed(_t5_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
@pytest.mark.parametrize("division,chunks", [([2, 2], 1), ([3, 1], 2),
                                             ([1, 3], 1)])
def test_t5_pp2_vs_baseline(division, chunks):
    """pp=2 cuts: [2,2] boundary inside the encoder (encoder-shaped act),
    [3,1] boundary past the bridge (decoder act + memory ride-along, 1F1B
    with 2 chunks), [1,3] mid-encoder cut."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=4, world_size=2, pp=2, tp=1, dp_type="ddp",
        global_bsz=4, chunks=chunks)
    plan.pp_division = division
    res = run_distributed(_t5_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_t5_ulysses_vs_baseline():
    """Ulysses SP on both stacks: encoder/decoder self-attention bias
    sliced post-a2a, cross-attention q a2a + kv head-chunking."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=4, world_size=2, pp=1, tp=2, use_sp=True, dp_type="ddp",
        global_bsz=4, chunks=1)
    res = run_distributed(_t5_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_t5_cp2_vs_baseline():
    """Ring-CP on both T5 stacks: bidirectional ring + bias on the
    encoder, causal ring + bias on the decoder self-attention,
    cross-attention over zigzag-sharded queries vs the full memory."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=4, world_size=2, pp=1, tp=1, cp=2, dp_type="ddp",
        global_bsz=4, chunks=1)
    plan.vcp = 2  # vocab layers share the cp split (uniform dp domain)
    res = run_distributed(_t5_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_t5_world4_tp2_cp2():
    """tp=2 x cp=2 composed on T5: head-sliced bias rows over the full
    cp-local zigzag pair (the SP allgather precedes attention)."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=4, world_size=4, pp=1, tp=2, cp=2, dp_type="ddp",
        global_bsz=4, chunks=1, vtp=2)
    plan.vcp = 2
    res = run_distributed(_t5_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_t5_world4_pp2_tp2():
    """t5 with pp=2 x tp=2: encoder-shaped boundary under megatron-SP
    seq sharding + head-sliced relative bias on both stages."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=4, world_size=4, pp=2, tp=2, dp_type="ddp",
        global_bsz=4, chunks=2, vtp=2)
    res = run_distributed(_t5_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_t5_world4_ulysses2_cp2():
    """ulysses sp=2 composed with ring-CP cp=2 on T5: post-a2a head
    chunks carry the bias into the ring inner attention."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=4, world_size=4, pp=1, tp=2, use_sp=True, cp=2,
        dp_type="ddp", global_bsz=4, chunks=1)
    plan.vcp = 2
    res = run_distributed(_t5_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_t5_pp2_ckpt_memory_boundary():
    """Activation checkpointing on all t5 layers with the memory-carrying
    pp cut ([3,1]): the encoder memory captured in the non-reentrant ckpt
    closure must re-materialize correctly."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=4, world_size=2, pp=2, tp=1, dp_type="ddp",
        checkpoint=True, global_bsz=4, chunks=2)
    plan.pp_division = [3, 1]
    res = run_distributed(_t5_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)
