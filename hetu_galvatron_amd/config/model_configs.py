"""Model presets + HF config adapter.

Reference: galvatron/utils/hf_config_adapter.py:285 (resolve_model_config) and
galvatron/models/model_configs/*.yaml.  Presets are inline dicts (no network);
an HF ``config.json``/AutoConfig can also be adapted when a path is given.
"""
from __future__ import annotations

import copy
import json
import os
from typing import Any, Dict

MODEL_PRESETS: Dict[str, Dict[str, Any]] = {
    "gpt2-small": dict(
        model_type="gpt", hidden_size=768, num_hidden_layers=12,
        num_attention_heads=12, num_key_value_heads=None, ffn_hidden_size=3072,
        vocab_size=50257, max_position_embeddings=1024, seq_length=1024,
        hidden_act="gelu", normalization="layernorm", norm_epsilon=1e-5,
        position_embedding_type="learned", add_bias_linear=True,
        add_qkv_bias=True, tie_word_embeddings=True,
        untie_embeddings_and_output_weights=False,
    ),
    "gpt2-xl": dict(
        model_type="gpt", hidden_size=1600, num_hidden_layers=48,
        num_attention_heads=25, num_key_value_heads=None, ffn_hidden_size=6400,
        vocab_size=50257, max_position_embeddings=1024, seq_length=1024,
        hidden_act="gelu", normalization="layernorm", norm_epsilon=1e-5,
        position_embedding_type="learned", add_bias_linear=True,
        add_qkv_bias=True, tie_word_embeddings=True,
        untie_embeddings_and_output_weights=False,
    ),
    # GPT-2 1.3B — BASELINE.json config 2 ("GPT-2 1.3B fixed plan TP=2 PP=2 DP=2")
    "gpt-1.3b": dict(
        model_type="gpt", hidden_size=2048, num_hidden_layers=24,
        num_attention_heads=32, num_key_value_heads=None, ffn_hidden_size=8192,
        vocab_size=50257, max_position_embeddings=2048, seq_length=2048,
        hidden_act="gelu", normalization="layernorm", norm_epsilon=1e-5,
        position_embedding_type="learned", add_bias_linear=True,
        add_qkv_bias=True, tie_word_embeddings=True,
        untie_embeddings_and_output_weights=False,
    ),
    "llama2-7b": dict(
        model_type="llama", hidden_size=4096, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=32, ffn_hidden_size=11008,
        vocab_size=32000, max_position_embeddings=4096, seq_length=4096,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=10000.0,
    ),
    "llama2-70b": dict(
        model_type="llama", hidden_size=8192, num_hidden_layers=80,
        num_attention_heads=64, num_key_value_heads=8, ffn_hidden_size=28672,
        vocab_size=32000, max_position_embeddings=4096, seq_length=4096,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=10000.0,
    ),
    # north-star model (BASELINE.json): Llama-3-8B
    "llama-3-8b": dict(
        model_type="llama", hidden_size=4096, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, ffn_hidden_size=14336,
        vocab_size=128256, max_position_embeddings=8192, seq_length=4096,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=500000.0,
    ),
    "llama-3-70b": dict(
        model_type="llama", hidden_size=8192, num_hidden_layers=80,
        num_attention_heads=64, num_key_value_heads=8, ffn_hidden_size=28672,
        vocab_size=128256, max_position_embeddings=8192, seq_length=8192,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=500000.0,
    ),
    "mistral-7b": dict(
        model_type="llama", hidden_size=4096, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, ffn_hidden_size=14336,
        vocab_size=32000, max_position_embeddings=8192, seq_length=4096,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=10000.0,
        sliding_window=4096,
    ),
    "qwen2.5-7b": dict(
        model_type="llama", hidden_size=3584, num_hidden_layers=28,
        num_attention_heads=28, num_key_value_heads=4, ffn_hidden_size=18944,
        vocab_size=152064, max_position_embeddings=8192, seq_length=4096,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-6,
        position_embedding_type="rope", rope_theta=1000000.0, add_qkv_bias=True,
    ),
    "qwen3-8b": dict(
        model_type="llama", hidden_size=4096, num_hidden_layers=36,
        num_attention_heads=32, num_key_value_heads=8, head_dim=128,
        ffn_hidden_size=12288, vocab_size=151936,
        max_position_embeddings=40960, seq_length=4096,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-6,
        position_embedding_type="rope", rope_theta=1000000.0,
        qk_layernorm=True, tie_word_embeddings=False,
    ),
    "mixtral-8x7b": dict(
        model_type="moe-llama", hidden_size=4096, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, ffn_hidden_size=14336,
        vocab_size=32000, max_position_embeddings=8192, seq_length=4096,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=1000000.0,
        num_experts=8, moe_router_topk=2,
    ),
    # T5-3B (BASELINE config 4) — encoder-decoder; learned positions
    # (T5's bucketized relative bias is a v2 flash-kernel item)
    "t5-3b": dict(
        model_type="t5", hidden_size=1024, num_hidden_layers=24,
        num_decoder_layers=24, num_attention_heads=32,
        num_key_value_heads=None, kv_channels=128, ffn_hidden_size=16384,
        vocab_size=32128, max_position_embeddings=1024, seq_length=512,
        encoder_seq_length=512, hidden_act="geglu",
        normalization="rmsnorm", norm_epsilon=1e-6,
        attention_softmax_scale=1.0,
        position_embedding_type="relative", add_bias_linear=False,
        add_qkv_bias=False,
    ),
    "tiny-t5": dict(
        model_type="t5", hidden_size=128, num_hidden_layers=2,
        num_decoder_layers=2, num_attention_heads=2,
        num_key_value_heads=None, kv_channels=64, ffn_hidden_size=256,
        vocab_size=512, max_position_embeddings=256, seq_length=64,
        encoder_seq_length=96, hidden_act="geglu",
        normalization="rmsnorm", norm_epsilon=1e-6,
        attention_softmax_scale=1.0,
        position_embedding_type="relative", add_bias_linear=False,
        add_qkv_bias=False,
    ),
    # tiny models for tests
    "tiny-llama": dict(
        model_type="llama", hidden_size=128, num_hidden_layers=2,
        num_attention_heads=2, num_key_value_heads=2, kv_channels=64,
        ffn_hidden_size=256,
        vocab_size=512, max_position_embeddings=256, seq_length=128,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=10000.0,
    ),
    "tiny-gpt": dict(
        model_type="gpt", hidden_size=128, num_hidden_layers=2,
        num_attention_heads=2, num_key_value_heads=None, kv_channels=64,
        ffn_hidden_size=512,
        vocab_size=512, max_position_embeddings=256, seq_length=128,
        hidden_act="gelu", normalization="layernorm", norm_epsilon=1e-5,
        position_embedding_type="learned", add_bias_linear=True,
        add_qkv_bias=True, tie_word_embeddings=True,
        untie_embeddings_and_output_weights=False,
    ),
    "tiny-qwen3": dict(
        model_type="llama", hidden_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, ffn_hidden_size=256,
        vocab_size=512, max_position_embeddings=256, seq_length=32,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-6,
        position_embedding_type="rope", rope_theta=10000.0,
        qk_layernorm=True,
    ),
    "tiny-qwen": dict(
        model_type="llama", hidden_size=128, num_hidden_layers=2,
        num_attention_heads=2, num_key_value_heads=1, kv_channels=64,
        ffn_hidden_size=256,
        vocab_size=512, max_position_embeddings=256, seq_length=128,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-6,
        position_embedding_type="rope", rope_theta=10000.0,
        add_qkv_bias=True,
    ),
    "tiny-moe": dict(
        model_type="moe-llama", hidden_size=128, num_hidden_layers=2,
        num_attention_heads=2, num_key_value_heads=2, kv_channels=64,
        ffn_hidden_size=256,
        vocab_size=512, max_position_embeddings=256, seq_length=128,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=10000.0,
        num_experts=4, moe_router_topk=2,
    ),
    # DeepSeek-V2-style MoE: shared expert + sigmoid scores + aux-free
    # bias + group-limited (node-limited) routing
    "deepseek-v2-lite": dict(
        model_type="moe-llama", hidden_size=2048, num_hidden_layers=27,
        num_attention_heads=16, num_key_value_heads=16, kv_channels=128,
        ffn_hidden_size=10944, moe_ffn_hidden_size=1408,
        vocab_size=102400, max_position_embeddings=4096, seq_length=4096,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-6,
        position_embedding_type="rope", rope_theta=10000.0,
        num_experts=64, moe_router_topk=6,
        moe_router_score_function="sigmoid", moe_aux_loss_free=True,
        moe_router_num_groups=8, moe_router_group_topk=3,
        moe_shared_expert_intermediate_size=2816,
    ),
    "tiny-moe-shared": dict(
        model_type="moe-llama", hidden_size=128, num_hidden_layers=2,
        num_attention_heads=2, num_key_value_heads=2, kv_channels=64,
        ffn_hidden_size=256,
        vocab_size=512, max_position_embeddings=256, seq_length=128,
        hidden_act="silu", normalization="rmsnorm", norm_epsilon=1e-5,
        position_embedding_type="rope", rope_theta=10000.0,
        num_experts=4, moe_router_topk=2,
        moe_router_score_function="sigmoid",
        moe_router_num_groups=2, moe_router_group_topk=1,
        moe_shared_expert_intermediate_size=64,
    ),
}

_HF_FIELD_MAP = {
    "hidden_size": "hidden_size",
    "num_hidden_layers": "num_hidden_layers",
    "num_attention_heads": "num_attention_heads",
    "num_key_value_heads": "num_key_value_heads",
    "intermediate_size": "ffn_hidden_size",
    "vocab_size": "vocab_size",
    "max_position_embeddings": "max_position_embeddings",
    "rms_norm_eps": "norm_epsilon",
    "layer_norm_epsilon": "norm_epsilon",
    "rope_theta": "rope_theta",
    "tie_word_embeddings": "tie_word_embeddings",
}


def adapt_hf_config(path: str) -> Dict[str, Any]:
    """Adapt an HF ``config.json`` into our ModelArgs field names
    (reference: hf_config_adapter.py:285-332)."""
    with open(os.path.join(path, "config.json") if os.path.isdir(path) else path) as f:
        hf = json.load(f)
    out: Dict[str, Any] = {}
    for hf_key, our_key in _HF_FIELD_MAP.items():
        if hf_key in hf:
            out[our_key] = hf[hf_key]
    mt = hf.get("model_type", "llama")
    if mt in ("llama", "mistral", "qwen2", "qwen3"):
        out.update(model_type="llama", hidden_act="silu", normalization="rmsnorm",
                   position_embedding_type="rope")
        if mt == "qwen3":
            out["qk_layernorm"] = True
        if hf.get("rope_scaling") and isinstance(hf["rope_scaling"], dict) \
                and hf["rope_scaling"].get("factor"):
            out["rope_scaling"] = float(hf["rope_scaling"]["factor"])
    elif mt == "mixtral":
        out.update(model_type="moe-llama", hidden_act="silu", normalization="rmsnorm",
                   position_embedding_type="rope",
                   num_experts=hf.get("num_local_experts", 8),
                   moe_router_topk=hf.get("num_experts_per_tok", 2))
    elif mt == "gpt2":
        out.update(model_type="gpt", hidden_size=hf.get("n_embd", 768),
                   num_hidden_layers=hf.get("n_layer", 12),
                   num_attention_heads=hf.get("n_head", 12),
                   ffn_hidden_size=4 * hf.get("n_embd", 768),
                   max_position_embeddings=hf.get("n_positions", 1024),
                   hidden_act="gelu", normalization="layernorm",
                   position_embedding_type="learned", add_bias_linear=True,
                   add_qkv_bias=True,
                   untie_embeddings_and_output_weights=False)
    return out


def resolve_model_config(raw: Dict[str, Any]) -> Dict[str, Any]:
    """Fill model architecture fields from a preset or HF config path.

    Explicit fields in ``raw['model']`` win over the preset.
    """
    raw = copy.deepcopy(raw)
    model = raw.get("model", {}) or {}
    name = model.get("model_name")
    hf_path = model.get("hf_config_path")
    base: Dict[str, Any] = {}
    if hf_path:
        base = adapt_hf_config(hf_path)
        model.pop("hf_config_path", None)
    elif name and name in MODEL_PRESETS:
        base = dict(MODEL_PRESETS[name])
    if base:
        merged = dict(base)
        merged.update({k: v for k, v in model.items() if v is not None})
        raw["model"] = merged
    return raw


def create_hf_config(model_args) -> Dict[str, Any]:
    """Reverse adapter: our ModelArgs -> HF-style config dict
    (reference: hf_config_adapter.py:333)."""
    m = model_args
    if m.model_type == "gpt":
        return dict(model_type="gpt2", n_embd=m.hidden_size, n_layer=m.num_hidden_layers,
                    n_head=m.num_attention_heads, n_positions=m.max_position_embeddings,
                    vocab_size=m.vocab_size, layer_norm_epsilon=m.norm_epsilon)
    out = dict(model_type="llama", hidden_size=m.hidden_size,
               num_hidden_layers=m.num_hidden_layers,
               num_attention_heads=m.num_attention_heads,
               num_key_value_heads=m.kv_heads,
               intermediate_size=m.ffn_hidden_size, vocab_size=m.vocab_size,
               max_position_embeddings=m.max_position_embeddings,
               rms_norm_eps=m.norm_epsilon, rope_theta=m.rope_theta,
               tie_word_embeddings=m.tie_word_embeddings)
    if m.model_type == "moe-llama":
        out.update(model_type="mixtral", num_local_experts=m.num_experts,
                   num_experts_per_tok=m.moe_router_topk)
    return out
