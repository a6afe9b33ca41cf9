"""MoE correctness: router invariants + EP distributed runs vs 1-process
baseline (reference test style: tests/core/test_ep.py:245)."""
import os

import pytest
import torch

from hetu_galvatron_amd.config import HybridParallelPlan, load_config

STEPS = 3
TOL = 0.03

BASE = {
    "model": {"model_name": "tiny-moe"},
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


def test_router_topk_and_aux():
    cfg = make_cfg()
    from hetu_galvatron_amd.runtime.moe.router import TopKRouter
    torch.manual_seed(0)
    r = TopKRouter(cfg.model)
    r.train()
    x = torch.randn(64, cfg.model.hidden_size)
    probs, idx, aux = r(x)
    assert probs.shape == (64, cfg.model.moe_router_topk)
    assert torch.allclose(probs.float().sum(-1), torch.ones(64), atol=1e-3)
    assert idx.max() < cfg.model.num_experts
    assert float(aux.detach()) > 0  # load-balancing loss active in training


def test_dispatcher_roundtrip_identity_experts():
    """dispatch -> identity experts -> combine == prob-weighted passthrough."""
    from hetu_galvatron_amd.runtime.moe.dispatcher import AlltoAllDispatcher
    torch.manual_seed(1)
    n, h, E, k = 32, 16, 4, 2
    x = torch.randn(n, h)
    probs = torch.softmax(torch.randn(n, k), dim=-1)
    idx = torch.randint(0, E, (n, k))
    d = AlltoAllDispatcher(None, E)

    class G:  # fake group of size 1
        size = 1
    d.ep_group = None
    inp, counts = d.dispatch(x, probs, idx)
    assert int(counts.sum()) == n * k
    out = d.combine(inp, n, k)
    want = x * probs.sum(-1, keepdim=True)
    assert torch.allclose(out, want, atol=1e-5)


def _moe_worker(rank, world, plan_dict, state_path, cfg_extra):
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


_BASELINE = {}


def get_baseline(tmp_dir="/tmp/galvatron_moe_test"):
    if "v" not in _BASELINE:
        from hetu_galvatron_amd.runtime import GalvatronModel
        from hetu_galvatron_amd.runtime.checkpoint.state import (
            canonical_state_from_stage)
        os.makedirs(tmp_dir, exist_ok=True)
        cfg = make_cfg()
        torch.manual_seed(0)
        model = GalvatronModel(cfg)
        state = canonical_state_from_stage(model.stage_model)
        path = os.path.join(tmp_dir, "tiny_moe_state.pt")
        torch.save(state, path)
        _BASELINE["v"] = (train_steps(model, cfg), path)
    return _BASELINE["v"]


@pytest.mark.distributed
@pytest.mark.parametrize("ep,dispatcher", [(2, "alltoall"), (2, "allgather"),
                                           (1, "alltoall")])
def test_moe_ep_vs_baseline(ep, dispatcher):
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=2, pp=1, tp=1, dp_type="ddp",
        global_bsz=4, chunks=1, ep=ep)
    res = run_distributed(
        _moe_worker, world_size=2,
        args=(plan.to_config_dict(), state_path,
              {"model": {"moe_token_dispatcher_type": dispatcher}}))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: dist {a:.4f} vs base {b:.4f} " \
                f"({losses} vs {base_losses})"


@pytest.mark.distributed
def test_moe_ep_zero3():
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=2, pp=1, tp=1, dp_type="zero3",
        global_bsz=4, chunks=1, ep=2)
    res = run_distributed(_moe_worker, world_size=2,
                          args=(plan.to_config_dict(), state_path, {}))
    for losses in res:
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
def test_moe_world4_ep2_dp2():
    """ep=2 x dp=2 on 4 ranks: experts split over half the sdp group,
    expert grads reduced over edp."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=4, pp=1, tp=1, dp_type="ddp",
        global_bsz=4, chunks=1, ep=2)
    res = run_distributed(_moe_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path, {}))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
@pytest.mark.parametrize("world,tp,ep", [(2, 2, 1), (4, 2, 2)])
def test_moe_etp_vs_baseline(world, tp, ep):
    """Expert tensor parallelism: megatron tp shards the expert ffn
    (partial outputs reduce-scattered to the seq shard); tp=2 alone and
    composed with ep=2 on 4 ranks."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=world, pp=1, tp=tp, dp_type="ddp",
        global_bsz=4, chunks=1, ep=ep)
    res = run_distributed(_moe_worker, world_size=world,
                          args=(plan.to_config_dict(), state_path, {}))
    for r, losses in enumerate(res):
        for s, (a, b) in enumerate(zip(losses, base_losses)):
            assert abs(a - b) < TOL, \
                f"rank {r} step {s}: {a:.4f} vs {b:.4f} " \
                f"({losses} vs {base_losses})"


def test_router_group_limited_topk():
    """Experts must come only from the selected groups."""
    cfg = make_cfg({"model": {"moe_router_num_groups": 2,
                              "moe_router_group_topk": 1}})
    from hetu_galvatron_amd.runtime.moe.router import TopKRouter
    torch.manual_seed(0)
    r = TopKRouter(cfg.model)
    x = torch.randn(64, cfg.model.hidden_size)
    probs, idx, _ = r(x)
    E, G = cfg.model.num_experts, 2
    per_g = E // G
    groups = idx // per_g
    # with group_topk=1 every token's experts are in ONE group
    assert (groups == groups[:, :1]).all()


def test_router_seq_aux_and_capacity():
    from hetu_galvatron_amd.runtime.moe.router import TopKRouter
    cfg = make_cfg({"model": {"moe_aux_loss_type": "seq_aux_loss"}})
    torch.manual_seed(1)
    r = TopKRouter(cfg.model)
    r.train()
    x = torch.randn(8 * 4, cfg.model.hidden_size)
    _, _, aux_seq = r(x, seq_len=8)
    assert float(aux_seq) > 0
    # capacity: tight factor zeroes some probs
    cfg2 = make_cfg({"model": {"moe_expert_capacity_factor": 0.5}})
    r2 = TopKRouter(cfg2.model)
    torch.manual_seed(2)
    probs, idx, _ = r2(torch.randn(64, cfg2.model.hidden_size))
    assert (probs == 0).any()          # overflow tokens dropped
    assert (probs.sum(-1) > 0).any()   # within-capacity tokens kept


def test_aux_loss_tracker():
    from hetu_galvatron_amd.runtime.moe import tracker
    tracker.clear()
    tracker.save_aux_loss("load_balancing", 0, torch.tensor(0.5))
    tracker.save_aux_loss("load_balancing", 0, torch.tensor(1.5))
    tracker.save_aux_loss("z_loss", 1, 2.0)
    got = tracker.reduce_and_get()
    assert abs(got["load_balancing/layer_0"] - 1.0) < 1e-9
    assert abs(got["z_loss/layer_1"] - 2.0) < 1e-9
    tracker.clear()
    assert tracker.reduce_and_get() == {}


def test_moe_shared_expert_group_router_trains():
    """tiny-moe-shared (shared expert + sigmoid scores + aux-free bias +
    group-limited routing, deepseek-style) overfits one batch."""
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler,
        get_train_iterator)
    cfg = make_cfg({"model": {"model_name": "tiny-moe-shared"},
                    "train": {"lr": 5e-3}})
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    it = get_train_iterator(cfg, torch.device("cpu"))
    batch = next(it)
    losses = []
    for _ in range(8):
        opt.zero_grad()
        st = model.forward_backward(batch)
        opt.step()
        sched.step()
        losses.append(st.loss)
    assert losses[-1] < losses[0] - 0.05, losses


@pytest.mark.distributed
def test_moe_shared_expert_etp2():
    """shared expert under etp: applied on the local seq shard with
    tp-summed replicated params."""
    import copy
    from tests.utils import run_distributed
    base = copy.deepcopy(BASE)
    base["model"] = {"model_name": "tiny-moe-shared"}
    # 1-proc baseline with the shared preset
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import (
        canonical_state_from_stage)
    cfg = make_cfg({"model": {"model_name": "tiny-moe-shared"}})
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    state = canonical_state_from_stage(model.stage_model)
    path = "/tmp/galvatron_moe_test/tiny_moe_shared_state.pt"
    os.makedirs(os.path.dirname(path), exist_ok=True)
    torch.save(state, path)
    base_losses = train_steps(model, cfg)
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=2, pp=1, tp=2, dp_type="ddp",
        global_bsz=4, chunks=1)
    res = run_distributed(
        _moe_worker, world_size=2,
        args=(plan.to_config_dict(), path,
              {"model": {"model_name": "tiny-moe-shared"}}))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
def test_moe_world4_pp2_ep2():
    """MoE composed with pipeline parallelism: 2 stages x (ep=2 within
    each stage's sdp domain); expert grads reduce over edp while boundary
    activations flow 1F1B."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=4, pp=2, tp=1, dp_type="ddp",
        global_bsz=4, chunks=2, ep=2)
    res = run_distributed(_moe_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path, {}))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
def test_moe_etp2_sequential_experts():
    """etp with SequentialMLP experts (per-expert nn.Linear fc1/fc2
    sliced by the gated-halves rule)."""
    from tests.utils import run_distributed
    import copy
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import (
        canonical_state_from_stage)
    cfg = make_cfg({"model": {"moe_grouped_gemm": False}})
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    state = canonical_state_from_stage(model.stage_model)
    path = "/tmp/galvatron_moe_test/tiny_moe_seq_state.pt"
    os.makedirs(os.path.dirname(path), exist_ok=True)
    torch.save(state, path)
    base_losses = train_steps(model, cfg)
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=2, pp=1, tp=2, dp_type="ddp",
        global_bsz=4, chunks=1)
    res = run_distributed(
        _moe_worker, world_size=2,
        args=(plan.to_config_dict(), path,
              {"model": {"moe_grouped_gemm": False}}))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
def test_moe_world4_ulysses2_ep2():
    """ulysses sp=2 composed with ep=2: each sp rank routes its sequence
    shard through its own ep plane (ep groups keyed by sp index)."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=4, pp=1, tp=2, use_sp=True,
        dp_type="ddp", global_bsz=4, chunks=1, ep=2)
    res = run_distributed(_moe_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path, {}))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
def test_moe_world4_cp2_ep2():
    """ring-CP composed with ep: experts shard over the (dp x cp) sdp
    domain while attention rings over cp."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=4, pp=1, tp=1, cp=2, dp_type="ddp",
        global_bsz=4, chunks=1, ep=2)
    res = run_distributed(_moe_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path, {}))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


@pytest.mark.distributed
def test_moe_world4_pp2_etp2():
    """pipeline x expert-TP: tp=2 within each of 2 stages."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=4, pp=2, tp=2, dp_type="ddp",
        global_bsz=4, chunks=2, vtp=2)
    res = run_distributed(_moe_worker, world_size=4,
                          args=(plan.to_config_dict(), state_path, {}))
    for losses in res:
        for a, b in zip(losses, base_losses):
            assert abs(a - b) < TOL, (losses, base_losses)


def test_router_sinkhorn_balances_selection():
    """Sinkhorn routing (reference router.py:140): with training-mode
    sinkhorn the expert choice is load-balanced even when raw logits all
    prefer one expert; eval mode routes by raw scores again."""
    import torch
    from hetu_galvatron_amd.runtime.moe.router import TopKRouter, sinkhorn

    class M:
        hidden_size = 16
        num_experts = 4
        moe_router_topk = 1
        moe_aux_loss_coeff = 0.0
        moe_z_loss_coeff = 0.0
        moe_router_score_function = "softmax"
        moe_router_pre_softmax = False
        moe_aux_loss_free = False
        moe_router_bias_update_rate = 0.0
        moe_aux_loss_type = "aux_loss"
        moe_router_num_groups = None
        moe_router_group_topk = None
        moe_expert_capacity_factor = None
        moe_router_load_balancing_type = "sinkhorn"

    torch.manual_seed(0)
    r = TopKRouter(M())
    # bias every logit toward expert 0
    with torch.no_grad():
        r.weight.zero_()
        r.weight[0] += 1.0
    x = torch.randn(64, 16).abs()  # positive -> expert-0 logit dominates
    r.train()
    _, idx, _ = r(x)
    counts = torch.bincount(idx.flatten(), minlength=4)
    # sinkhorn spreads the load: no expert takes everything
    assert counts.max() < 64, counts.tolist()
    assert (counts > 0).sum() >= 2, counts.tolist()
    r.eval()
    _, idx_eval, _ = r(x)
    # raw routing sends everything to the dominant expert
    assert torch.bincount(idx_eval.flatten(), minlength=4)[0] == 64

    # sinkhorn output is (approximately) doubly stochastic
    m = sinkhorn(torch.randn(32, 4))
    torch.testing.assert_close(m.sum(1), torch.full((32,), 1.0 / 32),
                               atol=1e-2, rtol=1e-2)


def test_router_deterministic_mode():
    """train.deterministic_mode: router topk breaks ties by expert index
    (stable argsort path, reference deterministic_mode arg)."""
    cfg = make_cfg({"train": {"deterministic_mode": True}})
    from hetu_galvatron_amd.runtime import GalvatronModel
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    routers = [b.inner.mlp.router for b in model.stage_model.blocks
               if hasattr(getattr(b.inner, "mlp", None), "router")]
    assert routers and all(r.deterministic for r in routers)
    r = routers[0]
    # tied scores: deterministic mode must pick the LOWEST expert indices
    x = torch.zeros(4, cfg.model.hidden_size)
    with torch.no_grad():
        r.weight.zero_()
    _, idx, _ = r(x)
    assert idx.tolist() == [[0, 1]] * 4, idx.tolist()


def test_dispatcher_pad_to_capacity_static_shapes():
    """moe_pad_expert_input_to_capacity (reference token_dispatcher):
    every expert sees exactly `cap` rows; kept tokens round-trip
    unchanged, dropped tokens contribute zero."""
    from hetu_galvatron_amd.runtime.moe.dispatcher import AlltoAllDispatcher
    import math
    torch.manual_seed(3)
    n, h, E, k, cf = 16, 8, 4, 2, 1.0
    x = torch.randn(n, h)
    probs = torch.softmax(torch.randn(n, k), dim=-1)
    idx = torch.randint(0, E, (n, k))
    d = AlltoAllDispatcher(None, E, capacity_factor=cf,
                           pad_to_capacity=True)
    inp, counts = d.dispatch(x, probs, idx)
    cap = math.ceil(n * k / E * cf)
    assert counts.tolist() == [cap] * E
    assert inp.shape == (E * cap, h)
    out = d.combine(inp, n, k)  # identity experts
    # kept tokens: prob-weighted passthrough; dropped: that expert's
    # contribution missing.  Compute the expected kept mask per (tok, j).
    flat = idx.reshape(-1)
    order = torch.argsort(flat, stable=True)
    kept = torch.zeros(n * k, dtype=torch.bool)
    for e in range(E):
        rows = order[flat[order] == e]
        kept[rows[:cap]] = True
    want = torch.zeros_like(x)
    for t in range(n):
        for j in range(k):
            if kept[t * k + j]:
                want[t] += probs[t, j] * x[t]
    torch.testing.assert_close(out, want, atol=1e-5, rtol=1e-5)


@pytest.mark.distributed
def test_moe_ep2_pad_to_capacity_runs():
    """pad-to-capacity under ep=2: static shapes through the a2a
    regrouped buffer, training still converges with the baseline
    (generous tol — capacity drops change the math slightly)."""
    from tests.utils import run_distributed
    _, state_path = get_baseline()
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=2, pp=1, tp=1, dp_type="ddp",
        global_bsz=4, chunks=1, ep=2)
    res = run_distributed(
        _moe_worker, world_size=2,
        args=(plan.to_config_dict(), state_path,
              {"model": {"moe_expert_capacity_factor": 2.0,
                         "moe_pad_expert_input_to_capacity": True}}))
    for losses in res:
        assert all(l == l for l in losses)  # finite
        assert losses[-1] < losses[0] + 0.5


def test_aux_loss_grad_invariant_to_chunks():
    """MoEAuxLossAutoScaler (reference moe_utils.py:166): the aux-loss
    gradient is scaled by 1/num_microbatches, so router grads match
    between chunks=1 and chunks=2 on the same global batch."""
    from hetu_galvatron_amd.runtime import GalvatronModel, get_train_iterator

    def router_grad(chunks):
        cfg = make_cfg()
        torch.manual_seed(0)
        plan = HybridParallelPlan.uniform(2, 1, global_bsz=4, chunks=chunks)
        model = GalvatronModel(cfg, plan)
        it = get_train_iterator(cfg, torch.device("cpu"))
        model.forward_backward(next(it))
        for blk in model.stage_model.blocks:
            r = getattr(getattr(blk.inner, "mlp", None), "router", None)
            if r is not None:
                owner = r.weight._galvatron_owner
                seg = next(s for s in owner.segments if s.param is r.weight)
                return owner.flat_grad[seg.offset:seg.offset + seg.numel] \
                    .clone()
        raise AssertionError("no router found")

    g1 = router_grad(1)
    g2 = router_grad(2)
    # aux is mildly nonlinear in the token subset (f, P are per-chunk
    # means), so allow small drift; without the 1/chunks scale the error
    # is ~2x on aux-dominated components
    torch.testing.assert_close(g1, g2, atol=2e-4, rtol=0.05)
    # direct unit check of the injected gradient
    from hetu_galvatron_amd.runtime.moe.router import MoEAuxLossAutoScaler
    old = MoEAuxLossAutoScaler.main_loss_backward_scale
    try:
        MoEAuxLossAutoScaler.main_loss_backward_scale = 0.25
        aux = torch.zeros((), requires_grad=True)
        x = torch.randn(3, requires_grad=True)
        MoEAuxLossAutoScaler.apply(x, aux).sum().backward()
        assert abs(float(aux.grad) - 0.25) < 1e-7
    finally:
        MoEAuxLossAutoScaler.main_loss_backward_scale = old


@pytest.mark.distributed
def test_moe_world8_ep4():
    """ep=4 (one expert per ep rank) on 8 ranks — the widest expert
    sharding tiny-moe supports; dense dp=8, expert grads over edp=2."""
    from tests.utils import run_distributed
    base_losses, state_path = get_baseline()
    # dense dp = 8 (ep shards experts inside the dp domain), so the
    # global batch must be divisible by 8 — different from the gbsz-4
    # baseline; assert finite, decreasing training instead
    plan = HybridParallelPlan.uniform(
        num_layers=2, world_size=8, pp=1, tp=1, dp_type="ddp",
        global_bsz=8, chunks=1, ep=4)
    res = run_distributed(
        _moe_worker, world_size=8,
        args=(plan.to_config_dict(), state_path,
              {"train": {"global_train_batch_size": 8}}))
    for r, losses in enumerate(res):
        assert all(l == l for l in losses)
        assert losses[-1] < losses[0] + 0.5


def _tracker_pp_worker(rank, world):
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from hetu_galvatron_amd.runtime.moe import tracker
    tracker.clear()
    # each "stage" records a DIFFERENT layer's aux (pp key divergence)
    tracker.save_aux_loss("moe_aux", rank, 1.0 + rank)
    out = tracker.reduce_and_get()
    tracker.clear()
    return out


@pytest.mark.distributed
def test_aux_tracker_handles_divergent_keys():
    """reduce_and_get must not deadlock when ranks recorded different
    layer keys (pipeline-split MoE); every key surfaces on every rank."""
    from tests.utils import run_distributed
    res = run_distributed(_tracker_pp_worker, world_size=2)
    for out in res:
        assert out == {"moe_aux/layer_0": 1.0, "moe_aux/layer_1": 2.0}, out
