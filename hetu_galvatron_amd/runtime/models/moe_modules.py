"""MoE layer modules (mixtral-style).

Reference: galvatron/core/runtime/models/moe_modules.py:19-155
(GalvatronMoEAttention / Router / MLP / DecoderLayer).  Here one module:
pre-norm attention + pre-norm routed-MLP with residuals; the MoE MLP =
TopKRouter -> dispatcher (alltoall|allgather over the layer's EP group)
-> Grouped/Sequential experts -> optional shared expert; router aux loss
injected through MoEAuxLossAutoScaler.

Expert tensor parallelism (etp == the layer's megatron tp, reference
comm_groups.py:322-345): the seq-sharded activation is all-gathered over
tp (every etp rank routes the identical full token set), experts hold
ffn/etp column/row shards producing PARTIAL outputs, and the combined
result is reduce-scattered back to the sequence shard — one gather +
one reduce-scatter per MoE layer, exactly the dense CPL/RPL traffic.
The shared expert stays on the local sequence shard (applied after the
reduce-scatter; its replicated params are tp-summed via `tp_replicated`).
"""
from __future__ import annotations

from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...config.schema import ModelArgs
from ..moe.dispatcher import AllGatherDispatcher, AlltoAllDispatcher
from ..moe.experts import GroupedMLP, SequentialMLP, SharedExpertMLP
from ..moe.router import TopKRouter, attach_aux_loss
from ..tensor_parallel.mappings import (
    gather_from_sequence_parallel_region, group_size,
    reduce_scatter_to_sequence_parallel_region)
from ..transformer import RotaryEmbedding
from ..transformer.attention import SelfAttention
from ..transformer.norm import build_norm
from .modules import _my_rank, _tag_tp_replicated


class GalvatronMoEMLP(nn.Module):
    def __init__(self, margs: ModelArgs, groups, dtype=None,
                 layer_idx: int = 0):
        super().__init__()
        self.margs = margs
        self.layer_idx = layer_idx
        self.topk = margs.moe_router_topk
        self.router = TopKRouter(margs, dtype=dtype)
        ep_group = groups.ep_group
        ep = ep_group.size if ep_group is not None else 1
        assert margs.num_experts % max(ep, 1) == 0
        if margs.moe_token_dispatcher_type == "allgather":
            self.dispatcher = AllGatherDispatcher(ep_group, margs.num_experts)
        else:
            self.dispatcher = AlltoAllDispatcher(
                ep_group, margs.num_experts,
                capacity_factor=margs.moe_expert_capacity_factor,
                pad_to_capacity=margs.moe_pad_expert_input_to_capacity)
        n_local = margs.num_experts // max(ep, 1)
        ffn = margs.moe_ffn_hidden_size or margs.ffn_hidden_size
        gated = margs.hidden_act in ("silu", "swiglu", "geglu")
        # expert-TP: ffn shard per etp rank, partial outputs reduced by the
        # post-combine reduce-scatter
        self.tp_group = None if groups.strategy.use_ulysses else groups.tp_group
        self.etp = group_size(self.tp_group) if self.tp_group is not None else 1
        assert ffn % self.etp == 0, f"etp={self.etp} must divide moe ffn {ffn}"
        if self.etp > 1:
            # routing-path router grads are PARTIAL per etp rank (combine
            # weights partial expert outputs); the tp-group sum assembles
            # them — aux is scaled 1/etp at attach so its full per-rank
            # grad sums back to 1x
            for prm in self.router.parameters():
                _tag_tp_replicated(prm)
        ffn_local = ffn // self.etp
        if margs.moe_grouped_gemm:
            self.experts = GroupedMLP(n_local, margs.hidden_size, ffn_local,
                                      dtype=dtype, gated=gated,
                                      act=margs.hidden_act,
                                      init_std=margs.init_method_std)
        else:
            self.experts = SequentialMLP(n_local, margs.hidden_size, ffn_local,
                                         dtype=dtype, gated=gated,
                                         act=margs.hidden_act,
                                         init_std=margs.init_method_std)
        if margs.moe_shared_expert_intermediate_size:
            self.shared = SharedExpertMLP(
                margs.hidden_size, margs.moe_shared_expert_intermediate_size,
                dtype=dtype, init_std=margs.init_method_std)
            if self.etp > 1:
                for prm in self.shared.parameters():
                    _tag_tp_replicated(prm)
        else:
            self.shared = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: [s_local, b, h] SBH (seq/tp shard under megatron-SP)."""
        x_in = x
        if self.etp > 1:
            x = gather_from_sequence_parallel_region(x, self.tp_group)
        s, b, h = x.shape
        flat = x.reshape(-1, h)
        probs, idx, aux = self.router(flat, seq_len=s)
        if self.training:
            from .. import moe  # noqa: F401  (package anchor)
            from ..moe import tracker
            tracker.save_aux_loss("moe_aux", self.layer_idx, aux)
        # shared expert FIRST: its GEMMs enqueue on the compute stream
        # before the dispatch all-to-all's stream dependency, so on RCCL
        # they overlap the a2a (the reference overlaps the same way via
        # SharedExpertMLP + the flex dispatcher, moe/mlp.py:215)
        shared_out = (self.shared(x_in.reshape(-1, x_in.shape[-1]))
                      if self.shared is not None else None)
        expert_in, tokens_per_expert = self.dispatcher.dispatch(
            flat, probs, idx)
        expert_out = self.experts(expert_in, tokens_per_expert)
        merged = self.dispatcher.combine(expert_out, flat.shape[0], self.topk)
        if self.training and aux.requires_grad:
            merged = attach_aux_loss(merged, aux / self.etp)
        out = merged.reshape(s, b, h)
        if self.etp > 1:
            # partial ffn outputs: the reduce-scatter both sums the etp
            # partials and restores the sequence shard
            out = reduce_scatter_to_sequence_parallel_region(out, self.tp_group)
        if shared_out is not None:
            out = out + shared_out.reshape(out.shape)
        return out


class GalvatronMoEDecoderLayer(nn.Module):
    """Pre-norm attention + pre-norm routed MLP
    (reference moe_modules.py:131)."""

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
            use_ulysses=s.use_ulysses, sequence_parallel=seq_par, dtype=dtype)
        self.mlp = GalvatronMoEMLP(margs, groups, dtype=dtype,
                                   layer_idx=layer_idx)
        if seq_par and group_size(groups.tp_group) > 1:
            _tag_tp_replicated(self.input_norm.weight,
                               getattr(self.input_norm, "bias", None),
                               self.post_attn_norm.weight,
                               getattr(self.post_attn_norm, "bias", None),
                               self.attention.linear_proj.bias)
        if margs.position_embedding_type == "rope":
            self.rotary = RotaryEmbedding(margs.head_dim, margs.rope_theta,
                                          scaling=margs.rope_scaling)
        else:
            self.rotary = None
        self.dropout_p = margs.hidden_dropout

    def _rope_tables(self, S: int, device):
        if self.rotary is None:
            return None, None
        s = self.strategy
        c = self.groups.coord_of(_my_rank())
        if s.use_ulysses:
            return self.rotary.get_for_rank(S, device, sp_rank=c.tp_idx,
                                            sp_size=s.tp_sp, cp_rank=c.cp_idx,
                                            cp_size=s.cp)
        return self.rotary.get_for_rank(S, device, sp_rank=0, sp_size=1,
                                        cp_rank=c.cp_idx, cp_size=s.cp)

    def forward(self, hidden: torch.Tensor, ctx: Dict) -> torch.Tensor:
        S = ctx["seq_len"]
        cos, sin = self._rope_tables(S, hidden.device)
        residual = hidden
        x = self.input_norm(hidden)
        x = self.attention(x, cos, sin)
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        hidden = residual + x
        residual = hidden
        x = self.post_attn_norm(hidden)
        x = self.mlp(x)
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        return residual + x
