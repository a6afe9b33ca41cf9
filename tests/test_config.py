"""Config loading / strategy codec tests (reference: tests/test_arguments.py)."""
import json
import os

import pytest

from hetu_galvatron_amd.config import (
    GalvatronConfig, HybridParallelPlan, load_config, apply_overrides,
    MODEL_PRESETS, even_pp_division,
)


def test_default_config():
    cfg = GalvatronConfig()
    assert cfg.model.model_name == "llama-3-8b"
    assert cfg.parallel.mixed_precision == "bf16"
    assert cfg.model.head_dim == 128
    assert cfg.model.kv_heads == 8


def test_load_with_overrides(tmp_path):
    y = tmp_path / "cfg.yaml"
    y.write_text("model:\n  model_name: llama2-7b\ntrain:\n  lr: 0.001\n")
    cfg = load_config(str(y), ["train.train_iters=7", "parallel.pp_deg=2",
                              "model.seq_length=2048"])
    assert cfg.model.hidden_size == 4096
    assert cfg.model.ffn_hidden_size == 11008  # from llama2-7b preset
    assert cfg.train.lr == 0.001
    assert cfg.train.train_iters == 7
    assert cfg.parallel.pp_deg == 2
    assert cfg.model.seq_length == 2048  # override beats preset


def test_override_parsing():
    out = apply_overrides({}, ["a.b=true", "a.c=1.5", "a.d=null", "a.e=[1,2]",
                              "a.f=hello"])
    assert out["a"] == {"b": True, "c": 1.5, "d": None, "e": [1, 2], "f": "hello"}


def test_presets_valid():
    for name in MODEL_PRESETS:
        cfg = load_config(base={"model": {"model_name": name}})
        assert cfg.model.hidden_size % cfg.model.num_attention_heads == 0, name


def test_strategy_json_roundtrip(tmp_path):
    plan = HybridParallelPlan.uniform(num_layers=8, world_size=8, pp=2, tp=2,
                                      dp_type="zero2", checkpoint=True,
                                      chunks=4, global_bsz=32)
    plan.validate(8)
    p = str(tmp_path / "plan.json")
    plan.save(p)
    loaded = HybridParallelPlan.load(p)
    assert loaded.tp_sizes_enc == [2] * 8
    assert loaded.pp_deg == 2
    assert loaded.checkpoint_flags == [1] * 8
    assert loaded.pp_division == [4, 4]
    # json format is the reference-compatible comma-joined contract
    raw = json.load(open(p))
    assert raw["tp_sizes_enc"] == "2,2,2,2,2,2,2,2"
    assert isinstance(raw["pp_deg"], int)


def test_reference_style_plan_parses():
    cfg = {
        "pp_deg": 1,
        "tp_sizes_enc": "1,1,1,1",
        "tp_consecutive_flags": "1,1,1,1",
        "dp_types_enc": "1,1,1,1",
        "use_sp": "0,0,0,0",
        "checkpoint": "1,1,0,0",
        "global_bsz": 16,
        "chunks": 1,
        "pp_division": "4",
        "pipeline_type": "pipedream_flush",
        "default_dp_type": "zero2",
        "vtp": 2,
        "vsp": 1,
        "embed_sdp": 1,
    }
    plan = HybridParallelPlan.from_config_dict(cfg)
    plan.validate(8)
    s0 = plan.layer(0, world_size=8)
    assert s0.dp_type == "zero3" and s0.checkpoint
    s3 = plan.layer(3, world_size=8)
    assert s3.dp_type == "zero3" and not s3.checkpoint
    vs = plan.vocab_strategy(8)
    assert vs.sp == 2 and vs.tp == 1  # vsp=1 -> ulysses on vocab
    assert vs.dp_type == "zero3"


def test_per_layer_mixed_plan():
    plan = HybridParallelPlan(
        pp_deg=1,
        tp_sizes_enc=[4, 2, 1, 1],
        tp_consecutive_flags=[1, 1, 1, 1],
        cp_sizes_enc=[1, 1, 2, 1],
        dp_types_enc=[0, 0, 1, 1],
        use_sp=[0, 1, 0, 0],
        checkpoint_flags=[0, 0, 0, 1],
        global_bsz=8, chunks=1, default_dp_type="ddp",
    )
    plan.validate(8)
    l0 = plan.layer(0, 8)
    assert (l0.tp, l0.dp) == (4, 2)
    l1 = plan.layer(1, 8)
    assert (l1.sp, l1.tp, l1.dp) == (2, 1, 4) and l1.use_ulysses
    l2 = plan.layer(2, 8)
    assert (l2.cp, l2.dp, l2.dp_type) == (2, 4, "zero3")
    assert l2.sdp == 8


def test_even_pp_division():
    assert even_pp_division(32, 4) == [8, 8, 8, 8]
    assert even_pp_division(10, 4) == [3, 3, 2, 2]


def test_validation_errors():
    plan = HybridParallelPlan.uniform(4, 8, tp=2)
    with pytest.raises(ValueError):
        plan.validate(7)  # tp=2 does not divide stage size 7
    bad = HybridParallelPlan.uniform(4, 8, tp=2)
    bad.cp_sizes_enc = [1, 1]
    with pytest.raises(ValueError):
        bad.validate(8)


def test_fp16_mixed_precision_training():
    """fp16 + dynamic loss scaling (reference capability): tiny model
    trains with finite losses and the scaler stays engaged."""
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler, get_train_iterator)

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 4, "train_iters": 3,
                  "lr": 1e-3, "lr_decay_style": "constant",
                  "distributed_backend": "gloo"},
        "parallel": {"mixed_precision": "fp16", "loss_scale_init": 1024.0},
    })
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    assert model.stage_model.blocks[1].flat.param_dtype == torch.float16
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    assert opt.loss_scaler is not None and opt.loss_scaler.scale == 1024.0
    it = get_train_iterator(cfg, torch.device("cpu"))
    losses = []
    for _ in range(3):
        opt.zero_grad()
        st = model.forward_backward(next(it))
        norm = opt.step()
        sched.step()
        assert norm == norm, "no overflow expected at scale 1024 on tiny"
        losses.append(st.loss)
    assert all(l == l and l > 0 for l in losses)
    assert losses[-1] < losses[0] + 0.5  # training, not diverging


def test_fp16_loss_scaler_overflow_skip():
    """An overflowed step is skipped globally: params unchanged, scale
    backed off (reference fp16 semantics)."""
    import math
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler, get_train_iterator)

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 4, "train_iters": 2,
                  "lr": 1e-3, "lr_decay_style": "constant",
                  "distributed_backend": "gloo"},
        "parallel": {"mixed_precision": "fp16"},
    })
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    it = get_train_iterator(cfg, torch.device("cpu"))
    opt.zero_grad()
    model.forward_backward(next(it))
    # poison one grad accumulator -> global inf detection
    blk = model.stage_model.blocks[1].flat
    blk.flat_grad[0] = float("inf")
    before = blk.master.clone()
    s0 = opt.loss_scaler.scale
    norm = opt.step()
    assert math.isnan(norm)
    assert torch.equal(blk.master, before), "skipped step must not update"
    assert opt.loss_scaler.scale == s0 / 2


def test_attention_dropout_trains():
    """attention_dropout was silently ignored; now it routes to the eager
    sdpa path in training (and is a no-op in eval)."""
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler, get_train_iterator)

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama", "attention_dropout": 0.3},
        "train": {"global_train_batch_size": 4, "train_iters": 2,
                  "lr": 1e-3, "lr_decay_style": "constant",
                  "distributed_backend": "gloo"},
    })
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    dec = model.stage_model.blocks[1].inner
    assert dec.attention.attn_dropout == 0.3
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    it = get_train_iterator(cfg, torch.device("cpu"))
    ctx = next(it)
    opt.zero_grad()
    st1 = model.forward_backward(ctx)
    opt.step()
    assert st1.loss == st1.loss and st1.loss > 0
    # dropout actually fires: two forwards of the same batch differ
    model.stage_model.blocks[1].inner.train()
    a = model.forward_backward(ctx).loss
    b = model.forward_backward(ctx).loss
    assert a != b, "attention dropout should randomize the loss"


def test_init_method_std_is_consumed():
    """train-quality init controls (reference init_method_std +
    Megatron scaled output init): std flows to every weight; output
    projections are scaled by 1/sqrt(2L)."""
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel

    stds = {}
    for std in (0.02, 0.1):
        torch.manual_seed(0)
        cfg = load_config(base={"model": {"model_name": "tiny-llama",
                                          "init_method_std": std}})
        m = GalvatronModel(cfg)
        blk = m.stage_model.blocks[1].inner  # first decoder layer
        stds[std] = (blk.attention.linear_qkv.weight.std().item(),
                     blk.attention.linear_proj.weight.std().item())
    import math
    L = 2  # tiny-llama layers
    for std, (qkv_std, proj_std) in stds.items():
        assert abs(qkv_std - std) < std * 0.1, (std, qkv_std)
        want = std / math.sqrt(2 * L)
        assert abs(proj_std - want) < want * 0.1, (std, proj_std)


def test_eod_mask_loss():
    """eod_mask_loss (reference get_batch loss_mask): EOD labels carry no
    loss and no gradient; normalization is over unmasked tokens."""
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.dataloader import build_batch_context

    eod = 7
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "data": {"eod_mask_loss": True, "eod_token_id": eod},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "lr": 1e-3}})
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    g = torch.Generator().manual_seed(3)
    toks = torch.randint(0, cfg.model.vocab_size, (2, 65), generator=g)
    toks[0, 10] = eod
    toks[1, 20] = eod
    ctx = build_batch_context(toks, torch.device("cpu"), eod_token=eod,
                              eod_mask_loss=True)
    n_masked = int((toks[:, 1:] == eod).sum())
    assert ctx["loss_denom"] == 2 * 64 - n_masked
    stats = model.forward_backward(dict(ctx))
    assert stats.token_count == ctx["loss_denom"]
    # unmasked run over the same data differs (mask is consumed)
    torch.manual_seed(0)
    model2 = GalvatronModel(load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "lr": 1e-3}}))
    ctx2 = build_batch_context(toks, torch.device("cpu"))
    stats2 = model2.forward_backward(ctx2)
    assert stats2.token_count == 2 * 64
    assert abs(stats.loss - stats2.loss) > 1e-6


def test_eval_loop_and_split():
    """Validation (forward-only) pass: data.split partitions the dataset,
    train.eval_interval/eval_iters drive a no-grad loss report (the
    reference builds the split iterators; the eval pass exceeds it)."""
    import io
    import contextlib
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime.dataloader import split_range

    assert split_range(1000, "969,30,1", "train") == (0, 969)
    assert split_range(1000, "969,30,1", "valid") == (969, 999)
    assert split_range(1000, "969,30,1", "test") == (999, 1000)
    assert split_range(10, "1,0,0", "valid") == (0, 10)  # degenerate

    from hetu_galvatron_amd.cli.train import main
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        main(["model.model_name=tiny-llama",
              "train.global_train_batch_size=2", "train.train_iters=2",
              "train.lr=1e-3", "train.eval_interval=1",
              "train.eval_iters=2"])
    out = buf.getvalue()
    assert "[eval] iter 1: valid loss" in out
    assert "[eval] iter 2: valid loss" in out


def test_lr_warmup_init_and_load_iteration_fields():
    """lr_warmup_init ramps warmup from a nonzero LR; ckpt.load_iteration
    selects a specific saved iteration (reference args)."""
    from hetu_galvatron_amd.runtime.optimizer.scheduler import (
        OptimizerParamScheduler)

    class FakeOpt:
        param_groups = [{"lr": 0.0}]
        weight_decay = 0.0
    sch = OptimizerParamScheduler(FakeOpt(), max_lr=1.0, warmup_steps=10,
                                  decay_steps=20, decay_style="constant",
                                  warmup_init_lr=0.5)
    sch.step()
    assert abs(sch.get_lr() - (0.5 + 0.5 * 1 / 10)) < 1e-9
    for _ in range(9):
        sch.step()
    assert abs(sch.get_lr() - 1.0) < 1e-9
    from hetu_galvatron_amd.config import load_config
    cfg = load_config(base={"ckpt": {"load_iteration": 3},
                            "train": {"lr_warmup_init": 0.5}})
    assert cfg.ckpt.load_iteration == 3
    assert cfg.train.lr_warmup_init == 0.5


def test_hf_config_adapter_roundtrip(tmp_path):
    """HF config.json -> our model args (reference hf_config_adapter
    resolve_model_config:285) and back (create_hf_config:333)."""
    import json
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.config.model_configs import create_hf_config

    hf = {"model_type": "qwen3", "hidden_size": 256, "num_hidden_layers": 4,
          "num_attention_heads": 8, "num_key_value_heads": 2,
          "intermediate_size": 512, "vocab_size": 1000,
          "max_position_embeddings": 2048, "rms_norm_eps": 1e-6,
          "rope_theta": 1e6, "rope_scaling": {"factor": 4.0},
          "sliding_window": None}
    (tmp_path / "config.json").write_text(json.dumps(hf))
    cfg = load_config(base={"model": {"hf_config_path": str(tmp_path)}})
    m = cfg.model
    assert (m.hidden_size, m.num_hidden_layers, m.kv_heads) == (256, 4, 2)
    assert m.qk_layernorm and m.rope_scaling == 4.0
    assert m.normalization == "rmsnorm" and m.model_type == "llama"
    back = create_hf_config(m)
    for k in ("hidden_size", "num_hidden_layers", "num_attention_heads",
              "vocab_size", "max_position_embeddings"):
        assert back[k] == hf[k], k


def test_training_is_deterministic_run_to_run(tmp_path):
    """Two identical cli.train runs produce byte-identical metric streams
    (seeded init + synthetic data + deterministic kernels on CPU)."""
    import json

    def run(tag):
        from hetu_galvatron_amd.cli.train import main
        d = tmp_path / tag
        main(["model.model_name=tiny-llama",
              "train.global_train_batch_size=2", "train.train_iters=3",
              "train.lr=1e-3", f"logging.tensorboard_dir={d}"])
        lines = [json.loads(l) for l in
                 open(d / "metrics.jsonl").read().splitlines()]
        return [(l.get("loss"), l.get("grad_norm")) for l in lines
                if "loss" in l]

    a = run("a")
    b = run("b")
    assert a and a == b


def test_eval_under_fp16():
    """Forward-only validation under fp16 (no scaler interaction; raw
    per-token sums in the stats)."""
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import GalvatronModel, get_train_iterator
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "parallel": {"mixed_precision": "fp16"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "lr": 1e-3}})
    torch.manual_seed(0)
    m = GalvatronModel(cfg)
    it = get_train_iterator(cfg, torch.device("cpu"), split="valid")
    st = m.evaluate(next(it))
    assert st.loss == st.loss and st.loss > 0
