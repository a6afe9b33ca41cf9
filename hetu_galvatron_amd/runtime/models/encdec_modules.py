"""Encoder-decoder (T5-style) layer modules.

BASELINE config 4 ("T5-3B encoder-decoder auto-search") — beyond the
reference's model zoo (gpt/llama/moe only).  Arch:

  [embedding(enc ids)] + [encoder]*Ne + [encdec_bridge] +
  [decoder_x]*Nd + [final_norm] + [lm_head]

The bridge applies the encoder final norm, all-gathers the encoder output
to the FULL sequence (replicated within the dp group: cross-attention
then needs no per-layer redistribution however the decoder layers are
laid out), stashes it in the batch context, and emits the decoder
embedding.  pp>1: the engine rides the memory along the p2p boundary
(engine.py _fwd_step).  Scope: tp / ulysses / ring-CP / ulysses x CP /
pp all compose on both stacks (relative bias follows each layout:
tp head slices, post-a2a chunks, cp-local zigzag rows).

Position encoding: T5 bucketized relative-position bias
(transformer/relative_bias.py) on an eager fp32-softmax attention path;
a bias input on the native flash kernel is the remaining perf item.
"""
from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...config.schema import ModelArgs
from ..redistribute import natural_rows
from ..tensor_parallel import group_size
from ..tensor_parallel.layers import normal_init, scaled_init
from ..tensor_parallel.mappings import gather_from_sequence_parallel_region
from ..transformer import MLP, SelfAttention, build_norm
from ..transformer.attention import CrossAttention
from ..transformer.relative_bias import RelativePositionBias
from .modules import GalvatronEmbedding, _my_rank, _tag_tp_replicated


def _build_rel_bias(margs: ModelArgs, groups, bidirectional: bool, dtype):
    """T5 bucketized relative bias (per layer; see relative_bias.py for the
    HF layer-0-shared layout note)."""
    if margs.position_embedding_type != "relative":
        return None
    rb = RelativePositionBias(
        margs.relative_attention_num_buckets,
        margs.relative_attention_max_distance,
        margs.num_attention_heads, bidirectional=bidirectional, dtype=dtype,
        init_std=margs.init_method_std)
    if group_size(groups.tp_group) > 1:
        # full table on every tp rank; per-rank grads cover disjoint head
        # columns, the tp grad all-reduce assembles the full gradient
        _tag_tp_replicated(rb.weight)
    return rb


def _rank_bias(rel_bias, attention, groups, hidden):
    """Bias for the heads of the q entering SelfAttention.forward.
    megatron-SP: input arrives seq-sharded S/tp, heads pre-sharded ->
    slice this tp rank's chunk of a full-seq bias.  ulysses: heads are
    full before the a2a -> full-head bias (DistributedAttention slices
    the post-a2a chunk itself)."""
    if rel_bias is None:
        return None
    s = groups.strategy
    c = groups.coord_of(_my_rank())
    if s.use_ulysses:
        # rows after the a2a = the full cp-local zigzag pair; heads are
        # chunked inside DistributedAttention
        S = hidden.shape[0] * s.tp_sp * s.cp
        full = rel_bias(S, S, hidden.device)
        if s.cp > 1:
            rows = natural_rows(S, s.cp, 1, c.cp_idx, 0, hidden.device)
            full = full[:, rows, :]
        return full
    tp = group_size(groups.tp_group)
    S = hidden.shape[0] * tp * s.cp
    hl = attention.heads_local
    full = rel_bias(S, S, hidden.device, c.tp_idx * hl, (c.tp_idx + 1) * hl)
    if s.cp > 1:
        # ring path: rows = this cp rank's packed zigzag q rows — the
        # FULL pair (megatron-SP re-gathers the tp seq split before
        # attention); the kv (column) axis stays natural/global
        # (attention_impl slices it per ring step)
        rows = natural_rows(S, s.cp, 1, c.cp_idx, 0, hidden.device)
        full = full[:, rows, :]
    return full


class GalvatronEncoderLayer(nn.Module):
    """Pre-norm bidirectional self-attention + MLP."""

    def __init__(self, margs: ModelArgs, groups, layer_idx: int = 0,
                 dtype=None):
        super().__init__()
        self.margs = margs
        self.groups = groups
        self.layer_idx = layer_idx
        s = groups.strategy
        self.strategy = s
        seq_par = not s.use_ulysses
        self.input_norm = build_norm(margs.normalization, margs.hidden_size,
                                     margs.norm_epsilon, dtype)
        self.post_attn_norm = build_norm(margs.normalization,
                                         margs.hidden_size,
                                         margs.norm_epsilon, dtype)
        self.attention = SelfAttention(
            margs, groups.tp_group, groups.sp_group, groups.cp_group,
            use_ulysses=s.use_ulysses, sequence_parallel=seq_par,
            dtype=dtype, causal=False)
        self.mlp = MLP(margs.hidden_size, margs.ffn_hidden_size,
                       None if s.use_ulysses else groups.tp_group,
                       hidden_act="silu"
                       if margs.hidden_act in ("silu", "swiglu")
                       else margs.hidden_act,
                       add_bias=margs.add_bias_linear,
                       sequence_parallel=seq_par, dtype=dtype,
                       init_method=normal_init(margs.init_method_std),
                       output_init_method=scaled_init(
                           margs.init_method_std, margs.num_hidden_layers))
        if seq_par and group_size(groups.tp_group) > 1:
            _tag_tp_replicated(self.input_norm.weight,
                               getattr(self.input_norm, "bias", None),
                               self.post_attn_norm.weight,
                               getattr(self.post_attn_norm, "bias", None),
                               self.attention.linear_proj.bias,
                               self.mlp.fc2.bias)
        self.rel_bias = _build_rel_bias(margs, groups, bidirectional=True,
                                        dtype=dtype)
        self.dropout_p = margs.hidden_dropout

    def forward(self, hidden: torch.Tensor, ctx: Dict) -> torch.Tensor:
        residual = hidden
        x = self.input_norm(hidden)
        bias = _rank_bias(self.rel_bias, self.attention, self.groups, hidden)
        x = self.attention(x, None, None, attn_bias=bias)
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        hidden = residual + x
        residual = hidden
        x = self.post_attn_norm(hidden)
        x = self.mlp(x)
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        return residual + x


class GalvatronEncDecBridge(nn.Module):
    """Encoder final norm -> full-seq memory in ctx -> decoder embedding."""

    def __init__(self, margs: ModelArgs, groups, dtype=None):
        super().__init__()
        self.margs = margs
        self.groups = groups
        self.strategy = groups.strategy
        self.enc_final_norm = build_norm(margs.normalization,
                                         margs.hidden_size,
                                         margs.norm_epsilon, dtype)
        self.dec_embedding = GalvatronEmbedding(margs, groups, dtype=dtype,
                                                ids_key="input_ids")

    def forward(self, enc_hidden: torch.Tensor, ctx: Dict) -> torch.Tensor:
        mem = self.enc_final_norm(enc_hidden)
        s = self.strategy
        # encoder output arrives in this block's layout (seq / tsp*cp after
        # LayerBlock redistribution); gather to the full sequence
        group = self.groups.tsp_cp_group
        if group is not None and group_size(group) > 1:
            mem = gather_from_sequence_parallel_region(mem, group)
            if s.cp > 1:
                # gathered order is group-rank packed (each rank's zigzag
                # pair); permute back to natural sequence order
                import torch
                S = mem.shape[0]
                sl = S // group_size(group)
                tsp = self.groups.seq_shard_degree // s.cp
                perm = torch.empty(S, dtype=torch.long, device=mem.device)
                for pi, gr in enumerate(group.ranks):
                    c = self.groups.coord_of(gr)
                    rows = natural_rows(S, s.cp, max(tsp, 1), c.cp_idx,
                                        c.tp_idx, mem.device)
                    perm[rows] = torch.arange(pi * sl, (pi + 1) * sl,
                                              device=mem.device)
                mem = mem[perm]
        ctx["encoder_memory"] = mem            # [S_enc, b_loc, h]
        return self.dec_embedding(ctx)


class GalvatronDecoderLayerX(nn.Module):
    """Pre-norm causal self-attention + cross-attention + MLP."""

    def __init__(self, margs: ModelArgs, groups, layer_idx: int = 0,
                 dtype=None):
        super().__init__()
        self.margs = margs
        self.groups = groups
        self.layer_idx = layer_idx
        s = groups.strategy
        self.strategy = s
        seq_par = not s.use_ulysses
        self.input_norm = build_norm(margs.normalization, margs.hidden_size,
                                     margs.norm_epsilon, dtype)
        self.cross_norm = build_norm(margs.normalization, margs.hidden_size,
                                     margs.norm_epsilon, dtype)
        self.post_attn_norm = build_norm(margs.normalization,
                                         margs.hidden_size,
                                         margs.norm_epsilon, dtype)
        self.attention = SelfAttention(
            margs, groups.tp_group, groups.sp_group, groups.cp_group,
            use_ulysses=s.use_ulysses, sequence_parallel=seq_par, dtype=dtype,
            causal=True)
        self.cross_attention = CrossAttention(
            margs, groups.tp_group, sequence_parallel=seq_par, dtype=dtype,
            sp_group=groups.sp_group, use_ulysses=s.use_ulysses)
        self.mlp = MLP(margs.hidden_size, margs.ffn_hidden_size,
                       None if s.use_ulysses else groups.tp_group,
                       hidden_act="silu"
                       if margs.hidden_act in ("silu", "swiglu")
                       else margs.hidden_act,
                       add_bias=margs.add_bias_linear,
                       sequence_parallel=seq_par, dtype=dtype,
                       init_method=normal_init(margs.init_method_std),
                       output_init_method=scaled_init(
                           margs.init_method_std, margs.num_hidden_layers))
        if seq_par and group_size(groups.tp_group) > 1:
            _tag_tp_replicated(self.input_norm.weight,
                               getattr(self.input_norm, "bias", None),
                               self.cross_norm.weight,
                               getattr(self.cross_norm, "bias", None),
                               self.post_attn_norm.weight,
                               getattr(self.post_attn_norm, "bias", None),
                               self.attention.linear_proj.bias,
                               self.cross_attention.linear_proj.bias,
                               self.mlp.fc2.bias)
        self.rel_bias = _build_rel_bias(margs, groups, bidirectional=False,
                                        dtype=dtype)
        self.dropout_p = margs.hidden_dropout

    def forward(self, hidden: torch.Tensor, ctx: Dict) -> torch.Tensor:
        residual = hidden
        x = self.input_norm(hidden)
        bias = _rank_bias(self.rel_bias, self.attention, self.groups, hidden)
        x = self.attention(x, None, None, attn_bias=bias)
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        hidden = residual + x

        residual = hidden
        x = self.cross_norm(hidden)
        x = self.cross_attention(x, ctx["encoder_memory"])
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        hidden = residual + x

        residual = hidden
        x = self.post_attn_norm(hidden)
        x = self.mlp(x)
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        return residual + x
