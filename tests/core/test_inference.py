"""Serving path: KV-cache incremental decode == full-prefix recompute.

Reference role parity: flash-decode inference (optional backend in the
reference); here validated functionally on CPU via the torch fallback of
decode_attention.
"""
import pytest
import torch

from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.ops import decode_attention
from hetu_galvatron_amd.runtime import GalvatronModel
from hetu_galvatron_amd.runtime.inference import GalvatronGenerator, KVCache


def make_model(name="tiny-llama"):
    cfg = load_config(base={
        "model": {"model_name": name},
        "parallel": {"mixed_precision": "fp32"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "distributed_backend": "gloo"},
    })
    torch.manual_seed(0)
    return GalvatronModel(cfg)


def test_decode_attention_matches_dense():
    torch.manual_seed(1)
    b, hq, hkv, d, S = 2, 4, 2, 64, 17
    q = torch.randn(b, hq, d)
    kc = torch.randn(b, 32, hkv, d)
    vc = torch.randn(b, 32, hkv, d)
    o = decode_attention(q, kc, vc, S)
    # dense reference
    k = kc[:, :S].repeat_interleave(hq // hkv, dim=2)
    v = vc[:, :S].repeat_interleave(hq // hkv, dim=2)
    att = torch.einsum("bhd,bshd->bhs", q, k) / d ** 0.5
    want = torch.einsum("bhs,bshd->bhd", att.softmax(-1), v)
    assert torch.allclose(o, want, atol=1e-5)


def test_incremental_decode_matches_recompute():
    model = make_model()
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    torch.manual_seed(2)
    ids = torch.randint(0, model.cfg.model.vocab_size, (2, 9))
    out = gen.generate(ids, max_new_tokens=5, temperature=0.0)
    assert out.shape == (2, 14)
    # recompute: at every step run the FULL prefix through a fresh cache
    seq = ids
    for _ in range(5):
        cache = KVCache(len(gen.layers), 2, 64, model.cfg.model.kv_heads,
                        model.cfg.model.head_dim, seq.device,
                        dtype=torch.float32)
        logits = gen._forward_tokens(seq, cache)
        seq = torch.cat([seq, logits.argmax(-1, keepdim=True)], dim=1)
    assert torch.equal(out, seq)


def test_generate_eos_and_sampling():
    model = make_model()
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    ids = torch.randint(0, model.cfg.model.vocab_size, (2, 4))
    torch.manual_seed(3)
    out = gen.generate(ids, max_new_tokens=4, temperature=0.8, top_k=5)
    assert out.shape[1] <= 8 and out.shape[0] == 2
    # eos stops generation early when every row has emitted it
    out2 = gen.generate(ids, max_new_tokens=8, temperature=0.0,
                        eos_id=int(out[0, -1]))
    assert out2.shape[1] <= 12


def test_generate_cli(capsys):
    from hetu_galvatron_amd.cli.generate import main
    out = main(["model.model_name=tiny-llama",
                "parallel.mixed_precision=fp32",
                "generate.max_new_tokens=3", "generate.prompt_ids=5,6,7"])
    assert out.shape == (1, 6)
    lines = capsys.readouterr().out.strip().splitlines()
    assert lines[-1].startswith("5 6 7 ")


def test_generate_graphed_cpu_fallback():
    """On CPU the capture raises and generate_graphed must fall back to
    the eager loop with identical greedy output."""
    model = make_model()
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    torch.manual_seed(4)
    ids = torch.randint(0, model.cfg.model.vocab_size, (2, 9))
    out_e = gen.generate(ids, max_new_tokens=6, temperature=0.0)
    out_g = gen.generate_graphed(ids, max_new_tokens=6, warmup_steps=2)
    assert torch.equal(out_e, out_g)


def test_generate_cli_with_hf_checkpoint(tmp_path):
    """HF-dir load path: canonical -> HF safetensors dir -> generate CLI;
    greedy tokens match the in-memory generator on the same weights."""
    from hetu_galvatron_amd.cli.generate import main
    from hetu_galvatron_amd.runtime.checkpoint.hf_adapter import (
        canonical_to_hf_llama, save_hf_checkpoint)
    from hetu_galvatron_amd.runtime.checkpoint.state import (
        canonical_state_from_stage)
    model = make_model()
    can = canonical_state_from_stage(model.stage_model)
    hf_dir = tmp_path / "hf"
    save_hf_checkpoint(canonical_to_hf_llama(can, model.cfg.model),
                       str(hf_dir))
    out = main(["model.model_name=tiny-llama",
                "parallel.mixed_precision=fp32",
                f"ckpt.load={hf_dir}",
                "generate.max_new_tokens=4", "generate.prompt_ids=5,6,7"])
    gen = GalvatronGenerator(model, max_batch=1, max_seq=64)
    want = gen.generate(torch.tensor([[5, 6, 7]]), max_new_tokens=4)
    assert torch.equal(out, want)


def test_chunked_prefill_matches_full():
    """Feeding the prompt in two chunks == one-shot prefill (cross-length
    bottom-right-causal attention against the cache)."""
    model = make_model()
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    torch.manual_seed(6)
    ids = torch.randint(0, model.cfg.model.vocab_size, (2, 12))
    c1 = KVCache(len(gen.layers), 2, 64, model.cfg.model.kv_heads,
                 model.cfg.model.head_dim, ids.device, dtype=torch.float32)
    full = gen._forward_tokens(ids, c1)
    c2 = KVCache(len(gen.layers), 2, 64, model.cfg.model.kv_heads,
                 model.cfg.model.head_dim, ids.device, dtype=torch.float32)
    gen._forward_tokens(ids[:, :5], c2)
    chunked = gen._forward_tokens(ids[:, 5:], c2)
    assert torch.allclose(full, chunked, atol=1e-4)
    for li in range(len(gen.layers)):
        assert torch.allclose(c1.k[li][:, :12], c2.k[li][:, :12], atol=1e-5)


def _tp_gen_worker(rank, world, state_path, prompt, want):
    import torch
    from hetu_galvatron_amd.config import HybridParallelPlan
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state
    from hetu_galvatron_amd.runtime.inference import GalvatronTPGenerator

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "parallel": {"mixed_precision": "fp32"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "distributed_backend": "gloo"}})
    initialize_galvatron(cfg, backend="gloo")
    plan = HybridParallelPlan.uniform(num_layers=2, world_size=2, pp=1,
                                      tp=2, dp_type="ddp", global_bsz=2,
                                      vtp=2)
    model = GalvatronModel(cfg, plan)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    gen = GalvatronTPGenerator(model, max_batch=2, max_seq=64)
    out = gen.generate(torch.tensor(prompt), max_new_tokens=5)
    assert out.tolist() == want, (out.tolist(), want)
    return True


@pytest.mark.distributed
def test_tp_decode_matches_single(tmp_path):
    """Megatron-TP decode (world 2) produces the same greedy tokens as
    the single-rank generator on the same weights."""
    from hetu_galvatron_amd.runtime.checkpoint.state import (
        canonical_state_from_stage)
    from tests.utils import run_distributed
    model = make_model()
    state = canonical_state_from_stage(model.stage_model)
    path = str(tmp_path / "state.pt")
    torch.save(state, path)
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    torch.manual_seed(3)
    ids = torch.randint(0, model.cfg.model.vocab_size, (2, 7))
    want = gen.generate(ids, max_new_tokens=5, temperature=0.0).tolist()
    res = run_distributed(_tp_gen_worker, world_size=2,
                          args=(path, ids.tolist(), want))
    assert all(res)


def test_generate_cli_graph_flag():
    from hetu_galvatron_amd.cli.generate import main
    out = main(["model.model_name=tiny-llama",
                "parallel.mixed_precision=fp32",
                "generate.max_new_tokens=3", "generate.prompt_ids=5,6,7",
                "generate.use_graph=1"])
    want = main(["model.model_name=tiny-llama",
                 "parallel.mixed_precision=fp32",
                 "generate.max_new_tokens=3", "generate.prompt_ids=5,6,7"])
    assert out.tolist() == want.tolist()


def _tp_sample_worker(rank, world, state_path):
    import torch
    from hetu_galvatron_amd.config import HybridParallelPlan
    from hetu_galvatron_amd.core.initialize import initialize_galvatron
    from hetu_galvatron_amd.runtime import GalvatronModel
    from hetu_galvatron_amd.runtime.checkpoint.state import load_full_state
    from hetu_galvatron_amd.runtime.inference import GalvatronTPGenerator

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "parallel": {"mixed_precision": "fp32"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "distributed_backend": "gloo"}})
    initialize_galvatron(cfg, backend="gloo")
    plan = HybridParallelPlan.uniform(num_layers=2, world_size=2, pp=1,
                                      tp=2, dp_type="ddp", global_bsz=2,
                                      vtp=2)
    model = GalvatronModel(cfg, plan)
    state = torch.load(state_path, weights_only=True)
    load_full_state(model.stage_model, state, cfg.model)
    gen = GalvatronTPGenerator(model, max_batch=1, max_seq=64)
    torch.manual_seed(100 + rank)  # rank-divergent ambient RNG on purpose
    out = gen.generate(torch.tensor([[5, 6, 7]]), max_new_tokens=5,
                       temperature=0.9, seed=42)
    return out.tolist()


@pytest.mark.distributed
def test_tp_decode_sampling_lockstep(tmp_path):
    """Temperature sampling stays identical across tp ranks (seed-shared
    generator), even with divergent ambient RNG state."""
    from hetu_galvatron_amd.runtime.checkpoint.state import (
        canonical_state_from_stage)
    from tests.utils import run_distributed
    model = make_model()
    state = canonical_state_from_stage(model.stage_model)
    path = str(tmp_path / "state.pt")
    torch.save(state, path)
    res = run_distributed(_tp_sample_worker, world_size=2, args=(path,))
    assert res[0] == res[1]


def test_windowed_decode_matches_dense():
    """decode_attention(window=): only the last `window` positions attend."""
    torch.manual_seed(7)
    b, hq, hkv, d, S, W = 2, 4, 2, 64, 21, 8
    q = torch.randn(b, hq, d)
    kc = torch.randn(b, 32, hkv, d)
    vc = torch.randn(b, 32, hkv, d)
    o = decode_attention(q, kc, vc, S, window=W)
    k = kc[:, S - W:S].repeat_interleave(hq // hkv, dim=2)
    v = vc[:, S - W:S].repeat_interleave(hq // hkv, dim=2)
    att = torch.einsum("bhd,bshd->bhs", q, k) / d ** 0.5
    want = torch.einsum("bhs,bshd->bhd", att.softmax(-1), v)
    assert torch.allclose(o, want, atol=1e-5)


def test_windowed_generator_incremental_matches_recompute():
    """Sliding-window model past its window: incremental decode ==
    full-prefix recompute (both windowed) — the r1 'generator v1' guard
    is gone."""
    model = make_model()
    model.cfg.model.sliding_window = 6
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    assert gen.window == 6
    torch.manual_seed(3)
    ids = torch.randint(0, model.cfg.model.vocab_size, (2, 10))  # > window
    out = gen.generate(ids, max_new_tokens=5, temperature=0.0)
    seq = ids
    for _ in range(5):
        cache = KVCache(len(gen.layers), 2, 64, model.cfg.model.kv_heads,
                        model.cfg.model.head_dim, seq.device,
                        dtype=torch.float32)
        logits = gen._forward_tokens(seq, cache)
        seq = torch.cat([seq, logits.argmax(-1, keepdim=True)], dim=1)
    assert torch.equal(out, seq)


def test_qwen3_decode_matches_recompute():
    """qk_layernorm in the generator's re-implemented layer step: cached
    decode equals full-prefix recompute on a qk-norm model."""
    model = make_model("tiny-qwen3")
    gen = GalvatronGenerator(model, max_batch=2, max_seq=64)
    assert gen.layers[0].attention.q_layernorm is not None
    torch.manual_seed(5)
    ids = torch.randint(0, model.cfg.model.vocab_size, (2, 7))
    out = gen.generate(ids, max_new_tokens=4, temperature=0.0)
    seq = ids
    for _ in range(4):
        cache = KVCache(len(gen.layers), 2, 64, model.cfg.model.kv_heads,
                        model.cfg.model.head_dim, seq.device,
                        dtype=torch.float32)
        logits = gen._forward_tokens(seq, cache)
        seq = torch.cat([seq, logits.argmax(-1, keepdim=True)], dim=1)
    assert torch.equal(out, seq)


def test_rotary_interleaved_decode_matches_recompute():
    """GPT-J interleaved RoPE through the decode path."""
    from hetu_galvatron_amd.config import load_config
    cfg = load_config(base={
        "model": {"model_name": "tiny-llama", "rotary_interleaved": True},
        "parallel": {"mixed_precision": "fp32"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "distributed_backend": "gloo"}})
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    gen = GalvatronGenerator(model, max_batch=1, max_seq=32)
    torch.manual_seed(6)
    ids = torch.randint(0, cfg.model.vocab_size, (1, 5))
    out = gen.generate(ids, max_new_tokens=3, temperature=0.0)
    seq = ids
    for _ in range(3):
        cache = KVCache(len(gen.layers), 1, 32, cfg.model.kv_heads,
                        cfg.model.head_dim, seq.device, dtype=torch.float32)
        logits = gen._forward_tokens(seq, cache)
        seq = torch.cat([seq, logits.argmax(-1, keepdim=True)], dim=1)
    assert torch.equal(out, seq)
