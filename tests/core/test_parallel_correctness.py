"""Distributed-correctness tests: every parallel mode vs a 1-process baseline.

Reference test strategy: tests/core/test_tp.py / test_pp.py / test_fsdp.py /
test_hybrid.py / test_redistributed.py — both models start from the same
weights, train a few steps on identical synthetic data, losses must match.
Runs on CPU over gloo (backend-pluggable runtime; BASELINE config 1).
"""
import os

import pytest
import torch

from hetu_galvatron_amd.config import GalvatronConfig, HybridParallelPlan, load_config

STEPS = 3
TOL = 0.02

BASE = {
    "model": {"model_name": "tiny-llama"},
    "train": {"global_train_batch_size": 4, "train_iters": STEPS, "lr": 1e-3,
              "lr_decay_style": "constant", "distributed_backend": "gloo"},
}


def make_cfg(extra=None):
    base = dict(BASE)
    if extra:
        import copy
        base = copy.deepcopy(base)
        for k, v in extra.items():
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
        ctx = next(it)
        stats = model.forward_backward(ctx)
        opt.step()
        sched.step()
        losses.append(model.global_loss(stats))
    return losses


def baseline_run(tmp_path_factory_dir=None):
    """1-process run; returns (losses, canonical_state_path)."""
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import canonical_state_from_stage

    cfg = make_cfg()
    model = GalvatronModel(cfg)
    state = canonical_state_from_stage(model.stage_model)
    path = os.path.join(tmp_path_factory_dir or "/tmp", "tiny_llama_state.pt")
    torch.save(state, path)
    losses = train_steps(model, cfg)
    return losses, path


_BASELINE = {}


def get_baseline(tmp_dir="/tmp/galvatron_test"):
    if "v" not in _BASELINE:
        os.makedirs(tmp_dir, exist_ok=True)
        _BASELINE["v"] = baseline_run(tmp_dir)
    return _BASELINE["v"]


def _dist_worker(rank, world, plan_dict, state_path, cfg_extra):
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state

    cfg = make_cfg(cfg_extra)
    initialize_galvatron(cfg, backend="gloo")
    plan = HybridParallelPlan.from_config_dict(plan_dict)
    model = GalvatronModel(cfg, plan)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    return train_steps(model, cfg)


def run_case(world, plan: HybridParallelPlan, cfg_extra=None):
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    res = run_distributed(_dist_worker, world_size=world,
                          args=(plan.to_config_dict(), state_path,
                                cfg_extra or {}))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: dist {a:.4f} vs baseline {b:.4f} " \
                f"(all: {losses} vs {base_losses})"
    return res


N_LAYERS = 2  # tiny-llama


@pytest.mark.distributed
def test_dp2_ddp():
    run_case(2, HybridParallelPlan.uniform(N_LAYERS, 2, dp_type="ddp",
                                           global_bsz=4))


@pytest.mark.distributed
def test_dp2_zero2():
    run_case(2, HybridParallelPlan.uniform(N_LAYERS, 2, dp_type="zero2",
                                           global_bsz=4))


@pytest.mark.distributed
def test_dp2_zero3():
    run_case(2, HybridParallelPlan.uniform(N_LAYERS, 2, dp_type="zero3",
                                           global_bsz=4))


@pytest.mark.distributed
def test_dp2_zero3_ckpt():
    run_case(2, HybridParallelPlan.uniform(N_LAYERS, 2, dp_type="zero3",
                                           checkpoint=True, global_bsz=4))


@pytest.mark.distributed
def test_tp2_megatron_sp():
    run_case(2, HybridParallelPlan.uniform(N_LAYERS, 2, tp=2, vtp=2,
                                           global_bsz=4))


@pytest.mark.distributed
def test_ulysses_sp2():
    run_case(2, HybridParallelPlan.uniform(N_LAYERS, 2, tp=2, use_sp=True,
                                           vtp=2, vsp=True, global_bsz=4))


@pytest.mark.distributed
def test_cp2():
    run_case(2, HybridParallelPlan.uniform(N_LAYERS, 2, cp=1, global_bsz=4,
                                           dp_type="ddp").__class__(
        pp_deg=1, tp_sizes_enc=[1] * N_LAYERS,
        tp_consecutive_flags=[1] * N_LAYERS, cp_sizes_enc=[2] * N_LAYERS,
        dp_types_enc=[0] * N_LAYERS, use_sp=[0] * N_LAYERS,
        checkpoint_flags=[0] * N_LAYERS, global_bsz=4, chunks=1,
        default_dp_type="ddp", vtp=1, vsp=0, vcp=2))


@pytest.mark.distributed
def test_pp2_gpipe():
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, pp=2, chunks=2,
                                      global_bsz=4,
                                      pipeline_type="gpipe")
    run_case(2, plan)


@pytest.mark.distributed
def test_pp2_1f1b():
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, pp=2, chunks=2,
                                      global_bsz=4,
                                      pipeline_type="pipedream_flush")
    run_case(2, plan)


@pytest.mark.distributed
def test_pp2_1f1b_uneven_chunks():
    """chunks=3 with 4 rows -> microbatches [2,1,1]: the remainder shape
    negotiation (reference pipeline.py:275) across 1F1B."""
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, pp=2, chunks=3,
                                      global_bsz=4,
                                      pipeline_type="pipedream_flush")
    run_case(2, plan)


@pytest.mark.distributed
def test_pp2_gpipe_uneven_chunks():
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, pp=2, chunks=3,
                                      global_bsz=4, pipeline_type="gpipe")
    run_case(2, plan)


@pytest.mark.distributed
@pytest.mark.parametrize("dp_type", ["zero2", "zero3"])
def test_dp2_reduce_each_microbatch(dp_type):
    """Per-microbatch shard-domain grad accumulation (the 70B memory mode:
    no full fp32 accumulator) must train identically to the default
    accumulate-then-reduce path."""
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, dp_type=dp_type,
                                      global_bsz=4, chunks=2)
    run_case(2, plan,
             {"parallel": {"reduce_grads_each_microbatch": True}})


@pytest.mark.distributed
def test_world4_zero3_reduce_mb():
    """Compound: zero3 (+ both-direction prefetch) x per-microbatch shard
    accumulation on 4 ranks (chunks clamp to the 1 row/rank; the
    multi-microbatch accumulation case is test_dp2_reduce_each_microbatch)."""
    plan = HybridParallelPlan.uniform(N_LAYERS, 4, dp_type="zero3",
                                      global_bsz=4, chunks=3)
    run_case(4, plan,
             {"parallel": {"reduce_grads_each_microbatch": True}})


def test_uneven_chunks_no_pipeline():
    """1-process path: global batch 4, chunks=3 -> [2,1,1] must reproduce
    the chunks=1 loss exactly (loss normalized by global tokens)."""
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state
    base_losses, state_path = get_baseline()
    cfg = make_cfg()
    plan = HybridParallelPlan.uniform(N_LAYERS, 1, chunks=3, global_bsz=4)
    model = GalvatronModel(cfg, plan)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    losses = train_steps(model, cfg)
    for a, b in zip(losses, base_losses):
        # fp32 grad accumulation order differs across microbatch splits and
        # Adam amplifies the epsilon-scale grad diffs over steps
        assert abs(a - b) < 5e-3, (losses, base_losses)


@pytest.mark.distributed
def test_mixed_per_layer_tp_dp():
    """layer0 tp2, layer1 dp2(zero3) -> exercises redistribution."""
    plan = HybridParallelPlan(
        pp_deg=1, tp_sizes_enc=[2, 1], tp_consecutive_flags=[1, 1],
        cp_sizes_enc=[1, 1], dp_types_enc=[0, 1], use_sp=[0, 0],
        checkpoint_flags=[0, 1], global_bsz=4, chunks=1,
        default_dp_type="ddp", vtp=2, vsp=0, vcp=1, embed_sdp=0)
    run_case(2, plan)


@pytest.mark.distributed
@pytest.mark.slow
def test_world4_tp2_dp2():
    run_case(4, HybridParallelPlan.uniform(N_LAYERS, 4, tp=2, vtp=2,
                                           dp_type="zero2", global_bsz=4))


@pytest.mark.distributed
@pytest.mark.slow
def test_world4_pp2_tp2():
    plan = HybridParallelPlan.uniform(N_LAYERS, 4, pp=2, tp=2, vtp=2,
                                      chunks=2, global_bsz=4)
    run_case(4, plan)


@pytest.mark.distributed
@pytest.mark.slow
def test_world4_mixed_ulysses_cp():
    """layer0 ulysses sp2+dp2, layer1 cp2+dp2."""
    plan = HybridParallelPlan(
        pp_deg=1, tp_sizes_enc=[2, 1], tp_consecutive_flags=[1, 1],
        cp_sizes_enc=[1, 2], dp_types_enc=[0, 0], use_sp=[1, 0],
        checkpoint_flags=[0, 0], global_bsz=4, chunks=1,
        default_dp_type="ddp", vtp=1, vsp=0, vcp=1)
    run_case(4, plan)


@pytest.mark.distributed
def test_world4_pp2_zero3_ckpt():
    """pp2 x dp2-zero3 x activation ckpt x 2 chunks — the deepest dense
    composition (sharded params re-gathered inside checkpointed
    recompute, 1F1B boundaries)."""
    run_case(4, HybridParallelPlan.uniform(
        N_LAYERS, 4, pp=2, tp=1, dp_type="zero3",
        checkpoint=True, global_bsz=4, chunks=2))


@pytest.mark.distributed
def test_world4_pp2_cp2():
    """pp2 x cp2: zigzag-sharded boundary activations + ring attention
    across pipeline stages."""
    run_case(4, HybridParallelPlan.uniform(
        N_LAYERS, 4, pp=2, cp=2, dp_type="ddp", global_bsz=4, chunks=2,
        vtp=1))


@pytest.mark.distributed
def test_world4_cp4():
    """cp=4 zigzag ring: 3-hop KV rotation fwd + dkv ring bwd (the cp2
    tests only exercise a single exchange; multi-hop covers the ring
    schedule and LSE merges at depth)."""
    run_case(4, HybridParallelPlan.uniform(
        N_LAYERS, 4, cp=4, dp_type="ddp", global_bsz=4, chunks=1, vtp=1))


def get_baseline_with(cfg_extra):
    """1-process baseline under extra model config (e.g. sliding window)."""
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import (
        canonical_state_from_stage)
    cfg = make_cfg(cfg_extra)
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    state = canonical_state_from_stage(model.stage_model)
    tag = abs(hash(str(sorted(str(cfg_extra))))) % 10**8
    path = f"/tmp/galvatron_test/tiny_llama_state_{tag}.pt"
    os.makedirs(os.path.dirname(path), exist_ok=True)
    torch.save(state, path)
    return train_steps(model, cfg), path


@pytest.mark.distributed
def test_ulysses_sp2_sliding_window():
    """Sliding window under ulysses (window semantics apply to the
    post-a2a full-seq inner attention)."""
    from tests.utils import run_distributed
    cfg_extra = {"model": {"sliding_window": 48}}
    base_losses, state_path = get_baseline_with(cfg_extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, tp=2, use_sp=True,
                                      dp_type="ddp", global_bsz=4)
    res = run_distributed(_dist_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path,
                                cfg_extra))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
def test_tp2_sliding_window():
    from tests.utils import run_distributed
    cfg_extra = {"model": {"sliding_window": 48}}
    base_losses, state_path = get_baseline_with(cfg_extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, tp=2, dp_type="ddp",
                                      global_bsz=4)
    res = run_distributed(_dist_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path,
                                cfg_extra))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
@pytest.mark.slow
def test_world8_dp8_zero2():
    """The round-end scale-bench topology (dp8 zero2, bf16-compressed
    reduction) at world 8 over gloo."""
    from tests.utils import run_distributed
    cfg_extra = {"train": {"global_train_batch_size": 8}}
    base_losses, state_path = get_baseline_with(cfg_extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 8, dp_type="zero2",
                                      global_bsz=8)
    res = run_distributed(_dist_worker, world_size=8,
                          args=(plan.to_config_dict(), state_path,
                                cfg_extra))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
def test_dp2_fp16_loss_scaling_matches_single_process():
    """fp16 + dynamic loss scaling under dp2/zero2: the scale is folded
    into the fused-optimizer gscale and overflow decisions come off the
    all-reduced grad norm, so two ranks must track a single fp16 process
    exactly (same init state, same data)."""
    from tests.utils import run_distributed
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state

    _, state_path = get_baseline()
    extra = {"parallel": {"mixed_precision": "fp16",
                          "loss_scale_init": 1024.0}}
    cfg = make_cfg(extra)
    model = GalvatronModel(cfg)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    base_losses = train_steps(model, cfg)

    plan = HybridParallelPlan.uniform(N_LAYERS, 2, dp_type="zero2",
                                      global_bsz=4)
    res = run_distributed(_dist_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path, extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < 5e-3, \
                f"rank {r} step {s}: dist {a:.4f} vs fp16 baseline {b:.4f}"


@pytest.mark.distributed
def test_world8_pp2_tp2_dp2():
    """Full 3D composition (pp2 x tp2 x dp2) on 8 ranks: boundary
    redistribution between tp shards across pipeline stages plus dp
    gradient reduction — the classic hybrid topology the search emits
    for larger models."""
    from tests.utils import run_distributed
    cfg_extra = {"train": {"global_train_batch_size": 8}}
    base_losses, state_path = get_baseline_with(cfg_extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 8, pp=2, tp=2,
                                      dp_type="zero2", global_bsz=8,
                                      chunks=2, vtp=2)
    res = run_distributed(_dist_worker, world_size=8,
                          args=(plan.to_config_dict(), state_path,
                                cfg_extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_world4_ulysses2_pp2():
    """Ulysses SP across pipeline stages (pp2 x sp2): sequence-sharded
    boundary activations + head-scatter a2a inside each stage."""
    run_case(4, HybridParallelPlan.uniform(
        N_LAYERS, 4, pp=2, tp=2, use_sp=True, dp_type="ddp",
        global_bsz=4, chunks=2, vtp=2, vsp=True))


@pytest.mark.distributed
def test_world8_pp2_cp2_dp2():
    """pp2 x cp2 x dp2 on 8 ranks: ring attention and dp reduction under
    a pipeline split."""
    from tests.utils import run_distributed
    cfg_extra = {"train": {"global_train_batch_size": 8}}
    base_losses, state_path = get_baseline_with(cfg_extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 8, pp=2, cp=2,
                                      dp_type="ddp", global_bsz=8,
                                      chunks=2, vtp=1)
    res = run_distributed(_dist_worker, world_size=8,
                          args=(plan.to_config_dict(), state_path,
                                cfg_extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_tp2_qk_layernorm():
    """qk_layernorm under Megatron-TP: the [head_dim] norm weights are
    tp-replicated (each rank normalizes its own heads), so their grads
    must be summed over the tp group — the tp_replicated tag path."""
    extra = {"model": {"qk_layernorm": True}}
    base_losses, state_path = get_baseline_with(extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, tp=2, vtp=2,
                                      global_bsz=4)
    from tests.utils import run_distributed
    res = run_distributed(_dist_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path, extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_dp2_eod_mask_loss():
    """eod_mask_loss under dp2: identical masked normalization on every
    rank (loss_denom is computed from the full global batch)."""
    eod = 5
    extra = {"data": {"eod_mask_loss": True, "eod_token_id": eod}}
    base_losses, state_path = get_baseline_with(extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, dp_type="ddp",
                                      global_bsz=4)
    from tests.utils import run_distributed
    res = run_distributed(_dist_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path, extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f}"


def _eval_worker(rank, world, plan_dict, state_path):
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import GalvatronModel, get_train_iterator
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state
    cfg = make_cfg()
    initialize_galvatron(cfg, backend="gloo")
    plan = HybridParallelPlan.from_config_dict(plan_dict)
    model = GalvatronModel(cfg, plan)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    it = get_train_iterator(cfg, torch.device("cpu"), split="valid")
    stats = model.evaluate(next(it))
    return model.global_loss(stats)


@pytest.mark.distributed
def test_pp2_evaluate_matches_single_process():
    """Forward-only validation across a pipeline split equals the
    1-process eval loss on the same weights/valid batch."""
    from tests.utils import run_distributed
    from hetu_galvatron_amd.runtime import GalvatronModel, get_train_iterator
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state
    _, state_path = get_baseline()
    cfg = make_cfg()
    model = GalvatronModel(cfg)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    it = get_train_iterator(cfg, torch.device("cpu"), split="valid")
    want = model.global_loss(model.evaluate(next(it)))
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, pp=2, chunks=2,
                                      global_bsz=4)
    res = run_distributed(_eval_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path))
    for r, got in enumerate(res):
        assert abs(got - want) < 1e-3, (r, got, want)


def _eval_then_train_worker(rank, world, plan_dict, state_path):
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import GalvatronModel, get_train_iterator
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state
    cfg = make_cfg()
    initialize_galvatron(cfg, backend="gloo")
    plan = HybridParallelPlan.from_config_dict(plan_dict)
    model = GalvatronModel(cfg, plan)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    # interleave an eval pass before every train step: the forward-only
    # path must leave zero3 param/gather state ready for training
    from hetu_galvatron_amd.runtime import get_optimizer_and_param_scheduler
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    it = get_train_iterator(model.cfg, torch.device("cpu"))
    vit = get_train_iterator(model.cfg, torch.device("cpu"), split="valid")
    losses = []
    for _ in range(STEPS):
        model.evaluate(next(vit))
        opt.zero_grad()
        stats = model.forward_backward(next(it))
        opt.step()
        sched.step()
        losses.append(model.global_loss(stats))
    return losses


@pytest.mark.distributed
def test_zero3_eval_interleaved_with_training():
    """evaluate() between zero3 train steps must not disturb gather /
    reshard bookkeeping: losses still match the plain baseline."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, dp_type="zero3",
                                      global_bsz=4)
    res = run_distributed(_eval_then_train_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f}"


@pytest.mark.distributed
def test_pp2_eod_mask_loss():
    """eod masking with the head on the last pipeline stage (loss_mask
    rides the per-rank batch context, microbatch-sliced by chunk_batch)."""
    eod = 5
    extra = {"data": {"eod_mask_loss": True, "eod_token_id": eod}}
    base_losses, state_path = get_baseline_with(extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 2, pp=2, chunks=2,
                                      global_bsz=4)
    from tests.utils import run_distributed
    res = run_distributed(_dist_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path, extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f}"


@pytest.mark.distributed
def test_world8_ulysses2_dp4():
    """Ulysses sp2 composed with dp4 on 8 ranks (tiny-llama has 2 heads,
    capping the ulysses degree at 2)."""
    from tests.utils import run_distributed
    cfg_extra = {"train": {"global_train_batch_size": 8}}
    base_losses, state_path = get_baseline_with(cfg_extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 8, tp=2, use_sp=True,
                                      dp_type="ddp", global_bsz=8,
                                      vtp=2, vsp=True)
    res = run_distributed(_dist_worker, world_size=8,
                          args=(plan.to_config_dict(), state_path,
                                cfg_extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f}"


@pytest.mark.distributed
def test_world4_ulysses4_deep_a2a():
    """Ulysses at its full degree (sp=4 over 4 heads): every head lands
    on a different rank — the deepest head-scatter the a2a supports."""
    extra = {"model": {"num_attention_heads": 4,
                       "num_key_value_heads": 2}}
    base_losses, state_path = get_baseline_with(extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 4, tp=4, use_sp=True,
                                      dp_type="ddp", global_bsz=4,
                                      vtp=4, vsp=True)
    from tests.utils import run_distributed
    res = run_distributed(_dist_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path, extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f}"


@pytest.mark.distributed
def test_world4_tp2_cp2():
    """Megatron-TP x ring-CP on the same layers (tp2 x cp2, world 4):
    sequence sharded by BOTH the zigzag pairs and the SP split, kv rings
    inside each tp shard."""
    run_case(4, HybridParallelPlan.uniform(
        N_LAYERS, 4, tp=2, cp=2, dp_type="ddp", global_bsz=4, vtp=2))


@pytest.mark.distributed
def test_world8_pp2_tp2_cp2():
    """4-way composition pp2 x tp2 x cp2 (dp=1) on 8 ranks: pipeline
    boundaries carry tp-and-zigzag-sharded activations, rings and SP
    collectives nest inside each stage."""
    from tests.utils import run_distributed
    cfg_extra = {"train": {"global_train_batch_size": 8}}
    base_losses, state_path = get_baseline_with(cfg_extra)
    plan = HybridParallelPlan.uniform(N_LAYERS, 8, pp=2, tp=2, cp=2,
                                      dp_type="ddp", global_bsz=8,
                                      chunks=2, vtp=2)
    res = run_distributed(_dist_worker, world_size=8,
                          args=(plan.to_config_dict(), state_path,
                                cfg_extra))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f}"
