"""Canonical full-model state <-> per-rank TP-sharded block params.

Reference: galvatron/core/runtime/checkpoint/llama_adapter.py:30-234 (QKV
interleave, vocab shard slicing, column/row splits) — rebuilt around OUR
canonical layout, which is simply the world_size=1 model's state_dict:

  embedding.word_embeddings.weight           [V, h]
  embedding.position_embeddings.weight       [P, h]        (gpt)
  decoder.{i}.input_norm.{weight,bias}
  decoder.{i}.attention.linear_qkv.{weight,bias}   KV-group-major interleave
  decoder.{i}.attention.linear_proj.{weight,bias}
  decoder.{i}.post_attn_norm.{weight,bias}
  decoder.{i}.mlp.fc1.{weight,bias}          gated: [gate(F); up(F)] stacked
  decoder.{i}.mlp.fc2.{weight,bias}
  final_norm.norm.{weight,bias}
  lm_head.lm_head.weight                     [V, h]
  encoder.{i}.* / encdec_bridge.*            (t5: encoder stack, bridge's
                                             enc_final_norm + dec_embedding;
                                             decoder indices continue after
                                             the encoder's)
  *.rel_bias.weight                          [buckets, H] replicated

Sharding rules per layer strategy (megatron tp=t, rank r; ulysses/cp/dp
replicate):
  vocab weights        : row slice V/t
  linear_qkv.weight    : contiguous row slice = whole KV groups (canonical is
                         group-major, so slice boundaries align)
  fc1 (gated)          : [gate_r ; up_r] slices re-stacked per rank
  column linears       : row slice;  row linears: column slice
  norms / biases-of-row-linears / pos-emb: replicated
"""
from __future__ import annotations

from typing import Dict, Optional

import torch

from ...config.schema import ModelArgs
from ..models.builder import StageModel
from ..tensor_parallel.mappings import group_rank, group_size


def _block_prefix(block) -> Optional[str]:
    if block.kind == "embedding":
        return "embedding"
    if block.kind == "decoder":
        return f"decoder.{block.inner.layer_idx}"
    if block.kind == "encoder":
        return f"encoder.{block.inner.layer_idx}"
    if block.kind == "encdec_bridge":
        return "encdec_bridge"
    if block.kind == "final_norm":
        return "final_norm"
    if block.kind == "lm_head":
        return "lm_head"
    raise ValueError(f"unknown block kind {block.kind!r}")


def canonical_state_from_stage(sm: StageModel) -> Dict[str, torch.Tensor]:
    """world_size==1 only: dump the canonical full state."""
    assert sm.world_size == 1, "canonical dump requires a 1-rank model"
    out: Dict[str, torch.Tensor] = {}
    for blk in sm.blocks:
        prefix = _block_prefix(blk)
        for name, p in blk.inner.state_dict().items():
            out[f"{prefix}.{name}"] = p.detach().clone()
    return out


def shard_for_rank(name: str, full: torch.Tensor, strategy, tp_rank: int,
                   tp_size: int, margs: ModelArgs) -> torch.Tensor:
    """Slice one canonical tensor for a megatron-TP rank."""
    t, r = tp_size, tp_rank
    if t == 1 or strategy.use_ulysses:
        return full
    if ".shared." in name or ".router." in name or "rel_bias" in name:
        return full  # shared expert / router / relative bias: replicated
    base = name.split(".")[-2] + "." + name.split(".")[-1]
    if "word_embeddings.weight" in name or "lm_head.weight" in name:
        V = full.shape[0]
        return full[r * V // t:(r + 1) * V // t]
    if "linear_qkv." in name:
        hkv = margs.kv_heads
        rows_per_group = full.shape[0] // hkv
        g0, g1 = r * hkv // t, (r + 1) * hkv // t
        return full[g0 * rows_per_group:g1 * rows_per_group]
    if "linear_proj.weight" in name or "fc2.weight" in name:
        C = full.shape[1]
        return full[:, r * C // t:(r + 1) * C // t]
    if "linear_q." in name:  # cross-attention query: plain column-parallel
        R = full.shape[0]
        return full[r * R // t:(r + 1) * R // t]
    if "linear_kv." in name:  # cross-attention fused [k; v]: slice halves
        H = full.shape[0] // 2
        kk, vv = full[:H], full[H:]
        return torch.cat([kk[r * H // t:(r + 1) * H // t],
                          vv[r * H // t:(r + 1) * H // t]], dim=0)
    if "fc1." in name:
        gated = margs.hidden_act in ("silu", "swiglu", "geglu")
        if gated:
            F = full.shape[0] // 2
            gate, up = full[:F], full[F:]
            return torch.cat([gate[r * F // t:(r + 1) * F // t],
                              up[r * F // t:(r + 1) * F // t]], dim=0)
        R = full.shape[0]
        return full[r * R // t:(r + 1) * R // t]
    # norms, row-linear biases, position embeddings: replicated
    return full


def _etp_ffn_slice(t: torch.Tensor, kind: str, tp_rank: int, tp_size: int,
                   margs: ModelArgs):
    """Slice an (ep-resolved) expert ffn weight for an etp rank.
    kind: 'w1' [.., h, out1] / 'fc1' [out1, h] gated [gate(F); up(F)]:
    each half sliced F/t; 'w2' [.., F, h] rows / 'fc2' [h, F] cols."""
    if tp_size <= 1:
        return t
    gated = margs.hidden_act in ("silu", "swiglu", "geglu")
    F = margs.moe_ffn_hidden_size or margs.ffn_hidden_size
    fl = F // tp_size
    r = tp_rank
    if kind == "w1":
        if gated:
            return torch.cat([t[..., r * fl:(r + 1) * fl],
                              t[..., F + r * fl:F + (r + 1) * fl]], dim=-1)
        return t[..., r * fl:(r + 1) * fl]
    if kind == "w2":
        return t[..., r * fl:(r + 1) * fl, :]
    if kind == "fc1":
        if gated:
            return torch.cat([t[r * fl:(r + 1) * fl],
                              t[F + r * fl:F + (r + 1) * fl]], dim=0)
        return t[r * fl:(r + 1) * fl]
    assert kind == "fc2"
    return t[:, r * fl:(r + 1) * fl]


def _expert_resolve(name: str, full_key: str, state: Dict[str, torch.Tensor],
                    ep_rank: int, ep: int, num_experts: int,
                    tp_rank: int = 0, tp_size: int = 1,
                    margs: Optional[ModelArgs] = None):
    """Map an EP/ETP-local expert param name to its canonical tensor slice.
    Grouped weights (experts.w1/w2 [E, ...]): slice dim 0 by ep rank, then
    ffn dim by etp rank.  Sequential (experts.fc1.{e}.weight): renumber
    local->global index, then ffn slice."""
    if ".experts." not in name:
        return state.get(full_key)
    parts = name.split(".")
    i = parts.index("experts")
    e_local = num_experts // ep
    if parts[i + 1] in ("w1", "w2"):
        full = state.get(full_key)
        if full is None:
            return None
        if ep > 1:
            full = full[ep_rank * e_local:(ep_rank + 1) * e_local]
        return _etp_ffn_slice(full, parts[i + 1], tp_rank, tp_size, margs)
    # sequential: experts.fc1.{e}.weight
    g = int(parts[i + 2]) + ep_rank * e_local
    parts[i + 2] = str(g)
    key = full_key.rsplit(name, 1)[0] + ".".join(parts)
    full = state.get(key)
    if full is None:
        return None
    return _etp_ffn_slice(full, parts[i + 1], tp_rank, tp_size, margs)


def load_full_state(sm: StageModel, state: Dict[str, torch.Tensor],
                    margs: ModelArgs) -> None:
    """Load a canonical full state into this rank's sharded blocks
    (works for any world size / plan; reference: param_init_fn shard loading
    parallel.py:87-97)."""
    for blk in sm.blocks:
        prefix = _block_prefix(blk)
        s = blk.groups.strategy
        tp_size = group_size(blk.groups.tp_group) if not s.use_ulysses else 1
        tp_rank = group_rank(blk.groups.tp_group) if not s.use_ulysses else 0
        epg = getattr(blk.groups, "ep_group", None)
        ep = epg.size if epg is not None else 1
        ep_rank = group_rank(epg) if epg is not None else 0
        if blk.flat is not None:
            blk.flat.gather_params()
        if getattr(blk, "flat_expert", None) is not None:
            blk.flat_expert.gather_params()
        with torch.no_grad():
            own = blk.inner.state_dict()
            for name, p in own.items():
                full_key = f"{prefix}.{name}"
                if ".experts." in name:
                    full = _expert_resolve(name, full_key, state, ep_rank,
                                           ep, margs.num_experts,
                                           tp_rank, tp_size, margs)
                    if full is None:
                        continue
                    if full.shape != p.shape:
                        raise ValueError(
                            f"{full_key}: expert shard {tuple(full.shape)} "
                            f"vs param {tuple(p.shape)} (ep={ep})")
                    p.copy_(full.to(p.dtype).to(p.device))
                    continue
                if full_key not in state:
                    continue
                shard = shard_for_rank(full_key, state[full_key], s, tp_rank,
                                       tp_size, margs)
                if shard.shape != p.shape:
                    raise ValueError(
                        f"{full_key}: shard {tuple(shard.shape)} vs param "
                        f"{tuple(p.shape)} (tp={tp_size} r={tp_rank})")
                p.copy_(shard.to(p.dtype).to(p.device))
        if blk.flat is not None:
            blk.flat.refresh_from_params()
        if getattr(blk, "flat_expert", None) is not None:
            blk.flat_expert.refresh_from_params()
