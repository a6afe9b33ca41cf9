"""Replicated-param consistency checker (reference test_mode realtime
weight checks)."""
import pytest
import torch

from hetu_galvatron_amd.config import HybridParallelPlan, load_config


def _worker(rank, world, perturb):
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.utils.consistency import check_param_consistency

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 4, "train_iters": 1,
                  "distributed_backend": "gloo"}})
    initialize_galvatron(cfg, backend="gloo")
    plan = HybridParallelPlan.uniform(num_layers=2, world_size=2, pp=1,
                                      tp=1, dp_type="ddp", global_bsz=4)
    torch.manual_seed(0)
    model = GalvatronModel(cfg, plan)
    clean = check_param_consistency(model.stage_model)
    if perturb and rank == 1:
        with torch.no_grad():
            next(model.stage_model.blocks[1].inner.parameters()).add_(1.0)
    dirty = check_param_consistency(model.stage_model)
    return {"clean": clean, "dirty": dirty}


@pytest.mark.distributed
@pytest.mark.parametrize("perturb", [False, True])
def test_consistency_detects_divergence(perturb):
    from tests.utils import run_distributed
    res = run_distributed(_worker, world_size=2, args=(perturb,))
    for r in res:
        assert r["clean"] == []
        if perturb:
            assert r["dirty"], "divergence not detected"
        else:
            assert r["dirty"] == []


def test_accuracy_alignment_tool(tmp_path):
    """tools/accuracy_alignment.py (reference scripts/accuracy_alignment/):
    1-process self-alignment is exact."""
    import json
    import subprocess
    import sys
    out = tmp_path / "align.json"
    r = subprocess.run(
        [sys.executable, "tools/accuracy_alignment.py", "--model",
         "tiny-llama", "--iters", "2", "--out", str(out)],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    d = json.load(open(out))
    assert d["aligned"] and d["max_abs_delta"] == 0.0


def _zero2_worker(rank, world):
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler,
        get_train_iterator)
    from hetu_galvatron_amd.utils.consistency import check_param_consistency

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 4, "train_iters": 2,
                  "lr": 1e-3, "distributed_backend": "gloo"}})
    initialize_galvatron(cfg, backend="gloo")
    plan = HybridParallelPlan.uniform(num_layers=2, world_size=2, pp=1,
                                      tp=1, dp_type="zero2", global_bsz=4)
    torch.manual_seed(0)
    model = GalvatronModel(cfg, plan)
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    it = get_train_iterator(cfg, torch.device("cpu"))
    for _ in range(2):
        opt.zero_grad()
        model.forward_backward(next(it))
        opt.step()
        sched.step()
    return check_param_consistency(model.stage_model)


@pytest.mark.distributed
def test_consistency_zero2_post_step_clean():
    """zero2's post-step allgather re-replicates the bf16 params: the
    checker (extended to zero2 domains) must stay clean after steps."""
    from tests.utils import run_distributed
    res = run_distributed(_zero2_worker, world_size=2)
    assert all(r == [] for r in res), res
