"""HuggingFace <-> canonical checkpoint layout conversion.

Reference: galvatron/core/runtime/checkpoint/llama_adapter.py:30-163 /
gpt_adapter.py:18-158 (QKV fusion with GQA interleave, gated-MLP stacking)
and galvatron/tools/checkpoint_convert_h2g.py / _g2h.py.

Canonical layout = this framework's world_size==1 state_dict (state.py).
QKV fusion is KV-group-major: group g rows = [q_{g*rep..}, k_g, v_g]*d so
a ColumnParallelLinear contiguous shard is whole groups.
"""
from __future__ import annotations

import os
from typing import Dict

import torch

from ...config.schema import ModelArgs


def fuse_qkv(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
             margs: ModelArgs) -> torch.Tensor:
    """[hq*d, h], [hkv*d, h], [hkv*d, h] -> group-major fused [(hq+2hkv)*d, h].
    Also fuses 1-D biases (last dim only)."""
    d = margs.head_dim
    hkv = margs.kv_heads
    rep = margs.num_attention_heads // hkv
    qg = q.reshape(hkv, rep * d, *q.shape[1:])
    kg = k.reshape(hkv, d, *k.shape[1:])
    vg = v.reshape(hkv, d, *v.shape[1:])
    return torch.cat([qg, kg, vg], dim=1).reshape(-1, *q.shape[1:])


def split_qkv(fused: torch.Tensor, margs: ModelArgs):
    d = margs.head_dim
    hkv = margs.kv_heads
    rep = margs.num_attention_heads // hkv
    g = fused.reshape(hkv, (rep + 2) * d, *fused.shape[1:])
    q = g[:, :rep * d].reshape(-1, *fused.shape[1:])
    k = g[:, rep * d:(rep + 1) * d].reshape(-1, *fused.shape[1:])
    v = g[:, (rep + 1) * d:].reshape(-1, *fused.shape[1:])
    return q, k, v


def hf_llama_to_canonical(hf: Dict[str, torch.Tensor],
                          margs: ModelArgs) -> Dict[str, torch.Tensor]:
    """HF LlamaForCausalLM keys -> canonical (reference h2g llama path)."""
    out: Dict[str, torch.Tensor] = {}
    out["embedding.word_embeddings.weight"] = hf["model.embed_tokens.weight"]
    for i in range(margs.num_hidden_layers):
        p = f"model.layers.{i}."
        c = f"decoder.{i}."
        out[c + "input_norm.weight"] = hf[p + "input_layernorm.weight"]
        out[c + "post_attn_norm.weight"] = \
            hf[p + "post_attention_layernorm.weight"]
        out[c + "attention.linear_qkv.weight"] = fuse_qkv(
            hf[p + "self_attn.q_proj.weight"],
            hf[p + "self_attn.k_proj.weight"],
            hf[p + "self_attn.v_proj.weight"], margs)
        if p + "self_attn.q_proj.bias" in hf:  # qwen2-style qkv bias
            out[c + "attention.linear_qkv.bias"] = fuse_qkv(
                hf[p + "self_attn.q_proj.bias"],
                hf[p + "self_attn.k_proj.bias"],
                hf[p + "self_attn.v_proj.bias"], margs)
        if p + "self_attn.q_norm.weight" in hf:  # qwen3/gemma2 qk norm
            out[c + "attention.q_layernorm.weight"] = \
                hf[p + "self_attn.q_norm.weight"]
            out[c + "attention.k_layernorm.weight"] = \
                hf[p + "self_attn.k_norm.weight"]
        out[c + "attention.linear_proj.weight"] = \
            hf[p + "self_attn.o_proj.weight"]
        out[c + "mlp.fc1.weight"] = torch.cat(
            [hf[p + "mlp.gate_proj.weight"], hf[p + "mlp.up_proj.weight"]], 0)
        out[c + "mlp.fc2.weight"] = hf[p + "mlp.down_proj.weight"]
    out["final_norm.norm.weight"] = hf["model.norm.weight"]
    if "lm_head.weight" in hf:
        out["lm_head.lm_head.weight"] = hf["lm_head.weight"]
    else:  # tied embeddings
        out["lm_head.lm_head.weight"] = hf["model.embed_tokens.weight"]
    return out


def canonical_to_hf_llama(can: Dict[str, torch.Tensor],
                          margs: ModelArgs) -> Dict[str, torch.Tensor]:
    out: Dict[str, torch.Tensor] = {}
    out["model.embed_tokens.weight"] = can["embedding.word_embeddings.weight"]
    for i in range(margs.num_hidden_layers):
        p = f"model.layers.{i}."
        c = f"decoder.{i}."
        out[p + "input_layernorm.weight"] = can[c + "input_norm.weight"]
        out[p + "post_attention_layernorm.weight"] = \
            can[c + "post_attn_norm.weight"]
        q, k, v = split_qkv(can[c + "attention.linear_qkv.weight"], margs)
        out[p + "self_attn.q_proj.weight"] = q
        out[p + "self_attn.k_proj.weight"] = k
        out[p + "self_attn.v_proj.weight"] = v
        if c + "attention.linear_qkv.bias" in can:  # qwen2-style qkv bias
            qb, kb, vb = split_qkv(can[c + "attention.linear_qkv.bias"],
                                   margs)
            out[p + "self_attn.q_proj.bias"] = qb
            out[p + "self_attn.k_proj.bias"] = kb
            out[p + "self_attn.v_proj.bias"] = vb
        if c + "attention.q_layernorm.weight" in can:
            out[p + "self_attn.q_norm.weight"] = \
                can[c + "attention.q_layernorm.weight"]
            out[p + "self_attn.k_norm.weight"] = \
                can[c + "attention.k_layernorm.weight"]
        out[p + "self_attn.o_proj.weight"] = \
            can[c + "attention.linear_proj.weight"]
        fc1 = can[c + "mlp.fc1.weight"]
        F = fc1.shape[0] // 2
        out[p + "mlp.gate_proj.weight"] = fc1[:F]
        out[p + "mlp.up_proj.weight"] = fc1[F:]
        out[p + "mlp.down_proj.weight"] = can[c + "mlp.fc2.weight"]
    out["model.norm.weight"] = can["final_norm.norm.weight"]
    out["lm_head.weight"] = can["lm_head.lm_head.weight"]
    return out


def hf_mixtral_to_canonical(hf: Dict[str, torch.Tensor],
                            margs: ModelArgs) -> Dict[str, torch.Tensor]:
    """HF MixtralForCausalLM keys -> canonical (reference h2g mixtral
    path, moe_adapter.py:37-349).  HF per-expert w1(gate)/w3(up)/w2(down)
    [F,h]/[F,h]/[h,F] map to grouped w1 [E, h, 2F] ([gate;up] stacked on
    the out dim, transposed) and w2 [E, F, h]."""
    out: Dict[str, torch.Tensor] = {}
    out["embedding.word_embeddings.weight"] = hf["model.embed_tokens.weight"]
    E = margs.num_experts
    for i in range(margs.num_hidden_layers):
        p = f"model.layers.{i}."
        c = f"decoder.{i}."
        out[c + "input_norm.weight"] = hf[p + "input_layernorm.weight"]
        out[c + "post_attn_norm.weight"] = \
            hf[p + "post_attention_layernorm.weight"]
        out[c + "attention.linear_qkv.weight"] = fuse_qkv(
            hf[p + "self_attn.q_proj.weight"],
            hf[p + "self_attn.k_proj.weight"],
            hf[p + "self_attn.v_proj.weight"], margs)
        out[c + "attention.linear_proj.weight"] = \
            hf[p + "self_attn.o_proj.weight"]
        # two HF layouts: legacy per-expert block_sparse_moe.experts.{e}
        # (w1=gate, w3=up, w2=down) and the fused 3D layout
        # (mlp.gate + mlp.experts.gate_up_proj/down_proj)
        if p + "mlp.experts.gate_up_proj" in hf:
            out[c + "mlp.router.weight"] = hf[p + "mlp.gate.weight"].float()
            out[c + "mlp.experts.w1"] = \
                hf[p + "mlp.experts.gate_up_proj"].transpose(1, 2) \
                .contiguous()                          # [E, h, 2F]
            out[c + "mlp.experts.w2"] = \
                hf[p + "mlp.experts.down_proj"].transpose(1, 2).contiguous()
        else:
            out[c + "mlp.router.weight"] = \
                hf[p + "block_sparse_moe.gate.weight"].float()
            w1 = torch.stack([
                torch.cat([hf[f"{p}block_sparse_moe.experts.{e}.w1.weight"],
                           hf[f"{p}block_sparse_moe.experts.{e}.w3.weight"]],
                          dim=0).t()
                for e in range(E)])                   # [E, h, 2F]
            w2 = torch.stack([
                hf[f"{p}block_sparse_moe.experts.{e}.w2.weight"].t()
                for e in range(E)])                   # [E, F, h]
            out[c + "mlp.experts.w1"] = w1.contiguous()
            out[c + "mlp.experts.w2"] = w2.contiguous()
    out["final_norm.norm.weight"] = hf["model.norm.weight"]
    out["lm_head.lm_head.weight"] = hf.get("lm_head.weight",
                                           hf["model.embed_tokens.weight"])
    return out


def canonical_to_hf_mixtral(can: Dict[str, torch.Tensor],
                            margs: ModelArgs) -> Dict[str, torch.Tensor]:
    out: Dict[str, torch.Tensor] = {}
    out["model.embed_tokens.weight"] = can["embedding.word_embeddings.weight"]
    for i in range(margs.num_hidden_layers):
        p = f"model.layers.{i}."
        c = f"decoder.{i}."
        out[p + "input_layernorm.weight"] = can[c + "input_norm.weight"]
        out[p + "post_attention_layernorm.weight"] = \
            can[c + "post_attn_norm.weight"]
        q, k, v = split_qkv(can[c + "attention.linear_qkv.weight"], margs)
        out[p + "self_attn.q_proj.weight"] = q
        out[p + "self_attn.k_proj.weight"] = k
        out[p + "self_attn.v_proj.weight"] = v
        out[p + "self_attn.o_proj.weight"] = \
            can[c + "attention.linear_proj.weight"]
        # emit the fused 3D layout (what current transformers loads)
        out[p + "mlp.gate.weight"] = can[c + "mlp.router.weight"]
        out[p + "mlp.experts.gate_up_proj"] = \
            can[c + "mlp.experts.w1"].transpose(1, 2).contiguous()
        out[p + "mlp.experts.down_proj"] = \
            can[c + "mlp.experts.w2"].transpose(1, 2).contiguous()
    out["model.norm.weight"] = can["final_norm.norm.weight"]
    out["lm_head.weight"] = can["lm_head.lm_head.weight"]
    return out


def hf_gpt2_to_canonical(hf: Dict[str, torch.Tensor],
                         margs: ModelArgs) -> Dict[str, torch.Tensor]:
    """HF GPT2LMHeadModel (Conv1D: stored transposed) -> canonical."""
    out: Dict[str, torch.Tensor] = {}
    out["embedding.word_embeddings.weight"] = hf["transformer.wte.weight"]
    out["embedding.position_embeddings.weight"] = hf["transformer.wpe.weight"]
    for i in range(margs.num_hidden_layers):
        p = f"transformer.h.{i}."
        c = f"decoder.{i}."
        out[c + "input_norm.weight"] = hf[p + "ln_1.weight"]
        out[c + "input_norm.bias"] = hf[p + "ln_1.bias"]
        out[c + "post_attn_norm.weight"] = hf[p + "ln_2.weight"]
        out[c + "post_attn_norm.bias"] = hf[p + "ln_2.bias"]
        w = hf[p + "attn.c_attn.weight"].t().contiguous()  # [3h, h]
        b = hf[p + "attn.c_attn.bias"]
        h = margs.hidden_size
        out[c + "attention.linear_qkv.weight"] = fuse_qkv(
            w[:h], w[h:2 * h], w[2 * h:], margs)
        out[c + "attention.linear_qkv.bias"] = fuse_qkv(
            b[:h], b[h:2 * h], b[2 * h:], margs)
        out[c + "attention.linear_proj.weight"] = \
            hf[p + "attn.c_proj.weight"].t().contiguous()
        out[c + "attention.linear_proj.bias"] = hf[p + "attn.c_proj.bias"]
        out[c + "mlp.fc1.weight"] = hf[p + "mlp.c_fc.weight"].t().contiguous()
        out[c + "mlp.fc1.bias"] = hf[p + "mlp.c_fc.bias"]
        out[c + "mlp.fc2.weight"] = hf[p + "mlp.c_proj.weight"].t().contiguous()
        out[c + "mlp.fc2.bias"] = hf[p + "mlp.c_proj.bias"]
    out["final_norm.norm.weight"] = hf["transformer.ln_f.weight"]
    out["final_norm.norm.bias"] = hf["transformer.ln_f.bias"]
    out["lm_head.lm_head.weight"] = hf["transformer.wte.weight"]
    return out


def load_hf_checkpoint(path: str) -> Dict[str, torch.Tensor]:
    """Load an HF checkpoint dir (safetensors shards or pytorch_model.bin)."""
    state: Dict[str, torch.Tensor] = {}
    st_files = sorted(f for f in os.listdir(path)
                      if f.endswith(".safetensors"))
    if st_files:
        from safetensors.torch import load_file
        for f in st_files:
            state.update(load_file(os.path.join(path, f)))
        return state
    bins = sorted(f for f in os.listdir(path)
                  if f.startswith("pytorch_model") and f.endswith(".bin"))
    for f in bins:
        state.update(torch.load(os.path.join(path, f), map_location="cpu",
                                weights_only=True))
    return state


def save_hf_checkpoint(state: Dict[str, torch.Tensor], path: str) -> None:
    from safetensors.torch import save_file
    os.makedirs(path, exist_ok=True)
    save_file({k: v.contiguous() for k, v in state.items()},
              os.path.join(path, "model.safetensors"))


def hf_t5_to_canonical(hf: Dict[str, torch.Tensor],
                       margs: ModelArgs) -> Dict[str, torch.Tensor]:
    """HF T5ForConditionalGeneration keys -> canonical (t5 family;
    reference-equivalent role of the per-family adapters).

    HF stores the relative_attention_bias table on block 0 only and
    shares it; our per-layer tables each get a copy (relative_bias.py
    layout note).  HF T5LayerNorm has no bias -> canonical layernorm
    biases load as zeros.  Gated (v1.1 wi_0/wi_1) and non-gated (wi)
    DenseReluDense both map onto fc1."""
    out: Dict[str, torch.Tensor] = {}
    emb = hf["shared.weight"]
    out["embedding.word_embeddings.weight"] = emb
    out["encdec_bridge.dec_embedding.word_embeddings.weight"] = emb
    n_enc = margs.num_hidden_layers
    n_dec = margs.num_decoder_layers or n_enc
    h = margs.hidden_size

    def norm(dst, src):
        out[dst + ".weight"] = hf[src + ".weight"]
        out[dst + ".bias"] = hf.get(src + ".bias",
                                    torch.zeros_like(hf[src + ".weight"]))

    def ffn(dst, src):
        if src + ".DenseReluDense.wi_0.weight" in hf:  # gated v1.1
            out[dst + ".fc1.weight"] = torch.cat(
                [hf[src + ".DenseReluDense.wi_0.weight"],
                 hf[src + ".DenseReluDense.wi_1.weight"]], dim=0)
        else:
            out[dst + ".fc1.weight"] = hf[src + ".DenseReluDense.wi.weight"]
        out[dst + ".fc2.weight"] = hf[src + ".DenseReluDense.wo.weight"]

    enc_bias = hf["encoder.block.0.layer.0.SelfAttention"
                  ".relative_attention_bias.weight"]
    dec_bias = hf["decoder.block.0.layer.0.SelfAttention"
                  ".relative_attention_bias.weight"]
    for i in range(n_enc):
        p = f"encoder.block.{i}.layer"
        c = f"encoder.{i}"
        norm(c + ".input_norm", p + ".0.layer_norm")
        norm(c + ".post_attn_norm", p + ".1.layer_norm")
        a = p + ".0.SelfAttention"
        out[c + ".attention.linear_qkv.weight"] = fuse_qkv(
            hf[a + ".q.weight"], hf[a + ".k.weight"], hf[a + ".v.weight"],
            margs)
        out[c + ".attention.linear_proj.weight"] = hf[a + ".o.weight"]
        out[c + ".rel_bias.weight"] = enc_bias.clone()
        ffn(c + ".mlp", p + ".1")
    norm("encdec_bridge.enc_final_norm", "encoder.final_layer_norm")
    for j in range(n_dec):
        p = f"decoder.block.{j}.layer"
        c = f"decoder.{n_enc + j}"
        norm(c + ".input_norm", p + ".0.layer_norm")
        norm(c + ".cross_norm", p + ".1.layer_norm")
        norm(c + ".post_attn_norm", p + ".2.layer_norm")
        a = p + ".0.SelfAttention"
        out[c + ".attention.linear_qkv.weight"] = fuse_qkv(
            hf[a + ".q.weight"], hf[a + ".k.weight"], hf[a + ".v.weight"],
            margs)
        out[c + ".attention.linear_proj.weight"] = hf[a + ".o.weight"]
        x = p + ".1.EncDecAttention"
        out[c + ".cross_attention.linear_q.weight"] = hf[x + ".q.weight"]
        out[c + ".cross_attention.linear_kv.weight"] = torch.cat(
            [hf[x + ".k.weight"], hf[x + ".v.weight"]], dim=0)
        out[c + ".cross_attention.linear_proj.weight"] = hf[x + ".o.weight"]
        out[c + ".rel_bias.weight"] = dec_bias.clone()
        ffn(c + ".mlp", p + ".2")
    norm("final_norm.norm", "decoder.final_layer_norm")
    out["lm_head.lm_head.weight"] = hf.get("lm_head.weight", emb)
    return out


def canonical_to_hf_t5(can: Dict[str, torch.Tensor],
                       margs: ModelArgs) -> Dict[str, torch.Tensor]:
    """Reverse mapping; per-layer bias tables collapse to HF's shared
    block-0 table (layer 0's copy wins — exact when they are tied or an
    HF import was the source)."""
    out: Dict[str, torch.Tensor] = {}
    out["shared.weight"] = can["embedding.word_embeddings.weight"]
    n_enc = margs.num_hidden_layers
    n_dec = margs.num_decoder_layers or n_enc
    gated = margs.hidden_act in ("silu", "swiglu", "geglu")
    for i in range(n_enc):
        p = f"encoder.block.{i}.layer"
        c = f"encoder.{i}"
        out[p + ".0.layer_norm.weight"] = can[c + ".input_norm.weight"]
        out[p + ".1.layer_norm.weight"] = can[c + ".post_attn_norm.weight"]
        q, k, v = split_qkv(can[c + ".attention.linear_qkv.weight"], margs)
        a = p + ".0.SelfAttention"
        out[a + ".q.weight"], out[a + ".k.weight"], out[a + ".v.weight"] = \
            q, k, v
        out[a + ".o.weight"] = can[c + ".attention.linear_proj.weight"]
        fc1 = can[c + ".mlp.fc1.weight"]
        if gated:
            F = fc1.shape[0] // 2
            out[p + ".1.DenseReluDense.wi_0.weight"] = fc1[:F]
            out[p + ".1.DenseReluDense.wi_1.weight"] = fc1[F:]
        else:
            out[p + ".1.DenseReluDense.wi.weight"] = fc1
        out[p + ".1.DenseReluDense.wo.weight"] = can[c + ".mlp.fc2.weight"]
    out["encoder.block.0.layer.0.SelfAttention"
        ".relative_attention_bias.weight"] = can["encoder.0.rel_bias.weight"]
    out["encoder.final_layer_norm.weight"] = \
        can["encdec_bridge.enc_final_norm.weight"]
    for j in range(n_dec):
        p = f"decoder.block.{j}.layer"
        c = f"decoder.{n_enc + j}"
        out[p + ".0.layer_norm.weight"] = can[c + ".input_norm.weight"]
        out[p + ".1.layer_norm.weight"] = can[c + ".cross_norm.weight"]
        out[p + ".2.layer_norm.weight"] = can[c + ".post_attn_norm.weight"]
        q, k, v = split_qkv(can[c + ".attention.linear_qkv.weight"], margs)
        a = p + ".0.SelfAttention"
        out[a + ".q.weight"], out[a + ".k.weight"], out[a + ".v.weight"] = \
            q, k, v
        out[a + ".o.weight"] = can[c + ".attention.linear_proj.weight"]
        x = p + ".1.EncDecAttention"
        out[x + ".q.weight"] = can[c + ".cross_attention.linear_q.weight"]
        kv = can[c + ".cross_attention.linear_kv.weight"]
        H = kv.shape[0] // 2
        out[x + ".k.weight"], out[x + ".v.weight"] = kv[:H], kv[H:]
        out[x + ".o.weight"] = can[c + ".cross_attention.linear_proj.weight"]
        fc1 = can[c + ".mlp.fc1.weight"]
        if gated:
            F = fc1.shape[0] // 2
            out[p + ".2.DenseReluDense.wi_0.weight"] = fc1[:F]
            out[p + ".2.DenseReluDense.wi_1.weight"] = fc1[F:]
        else:
            out[p + ".2.DenseReluDense.wi.weight"] = fc1
        out[p + ".2.DenseReluDense.wo.weight"] = can[c + ".mlp.fc2.weight"]
    out["decoder.block.0.layer.0.SelfAttention"
        ".relative_attention_bias.weight"] = \
        can[f"decoder.{n_enc}.rel_bias.weight"]
    out["decoder.final_layer_norm.weight"] = can["final_norm.norm.weight"]
    out["lm_head.weight"] = can["lm_head.lm_head.weight"]
    return out


def hf_to_canonical(hf: Dict[str, torch.Tensor], margs: ModelArgs
                    ) -> Dict[str, torch.Tensor]:
    if margs.model_type == "gpt":
        return hf_gpt2_to_canonical(hf, margs)
    if margs.model_type == "t5":
        return hf_t5_to_canonical(hf, margs)
    if margs.model_type.startswith("moe") and margs.num_experts > 0:
        return hf_mixtral_to_canonical(hf, margs)
    return hf_llama_to_canonical(hf, margs)
