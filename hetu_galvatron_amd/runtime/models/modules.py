"""Galvatron layer modules: per-layer TP/SP/CP-aware building blocks.

Reference: galvatron/core/runtime/models/modules.py:35-347.
Every module slices the FULL microbatch tensors from the batch context by
its own layout (batch by dp_idx, seq zigzag-by-cp + contiguous-by-tsp), so
per-layer dp/tp/sp/cp degrees may all differ — the LayerBlock wrapper
redistributes hidden activations between layouts.

Batch context (ctx) keys: input_ids [B, S], labels [B, S], batch_size B,
seq_len S.
"""
from __future__ import annotations

from typing import Dict

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from ...config.schema import ModelArgs
from ...core.comm_groups import LayerCommGroups
from ..redistribute import natural_rows
from ..tensor_parallel import (
    ColumnParallelLinear, VocabParallelEmbedding, group_rank, group_size,
    vocab_parallel_cross_entropy,
)
from ..tensor_parallel.layers import normal_init, scaled_init
from ..transformer import MLP, RotaryEmbedding, SelfAttention, build_norm


def _my_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def _tag_tp_replicated(*params) -> None:
    """Mark params whose grads need the extra TP-group all-reduce under
    Megatron-SP (replaces the reference's patched FSDP SP-layernorm hook,
    sp_grad_reduce.py:48-132)."""
    for p in params:
        if p is not None:
            p.tp_replicated = True


class GalvatronEmbedding(nn.Module):
    """Vocab-parallel embedding (+ optional learned positions) producing this
    rank's SBH activation shard (reference: modules.py:35-102)."""

    def __init__(self, margs: ModelArgs, groups: LayerCommGroups, dtype=None,
                 ids_key: str = "input_ids"):
        super().__init__()
        self.margs = margs
        self.groups = groups
        self.ids_key = ids_key
        s = groups.strategy
        self.strategy = s
        if s.use_ulysses:
            # vocab-sp: weight replicated over the sp group, seq sharded
            self.word_embeddings = VocabParallelEmbedding(
                margs.vocab_size, margs.hidden_size, None, dtype=dtype,
                init_method=normal_init(margs.init_method_std))
        else:
            self.word_embeddings = VocabParallelEmbedding(
                margs.vocab_size, margs.hidden_size, groups.tp_group,
                sequence_parallel=True, dtype=dtype,
                init_method=normal_init(margs.init_method_std))
        if margs.position_embedding_type == "learned":
            self.position_embeddings = nn.Embedding(
                margs.max_position_embeddings, margs.hidden_size,
                **({"dtype": dtype} if dtype else {}))
            nn.init.normal_(self.position_embeddings.weight, 0.0,
                            margs.init_method_std)
        else:
            self.position_embeddings = None
        self.dropout_p = margs.hidden_dropout

    def forward(self, ctx: Dict) -> torch.Tensor:
        ids = ctx[self.ids_key]
        B, S = ids.shape
        c = self.groups.coord_of(_my_rank())
        s = self.strategy
        b_loc = B // s.dp
        ids = ids[c.dp_idx * b_loc:(c.dp_idx + 1) * b_loc]
        if s.use_ulysses:
            rows = natural_rows(S, s.cp, s.tp_sp, c.cp_idx, c.tp_idx, ids.device)
        else:
            rows = natural_rows(S, s.cp, 1, c.cp_idx, 0, ids.device)
        ids = ids[:, rows]
        h = self.word_embeddings(ids)  # [s_rows(/tp), b_loc, h]
        if self.position_embeddings is not None:
            pos = self.position_embeddings(rows)  # [rows, h]
            if not s.use_ulysses and group_size(self.groups.tp_group) > 1:
                # word embedding was reduce-scattered: add the matching slice
                tp = group_size(self.groups.tp_group)
                r = group_rank(self.groups.tp_group)
                sl = pos.shape[0] // tp
                pos = pos[r * sl:(r + 1) * sl]
            h = h + pos.unsqueeze(1)
        if self.dropout_p > 0 and self.training:
            h = F.dropout(h, self.dropout_p)
        return h


class GalvatronDecoderLayer(nn.Module):
    """Pre-norm attention + pre-norm MLP with residuals
    (reference: modules.py:103-242)."""

    def __init__(self, margs: ModelArgs, groups: LayerCommGroups,
                 layer_idx: int = 0, dtype=None):
        super().__init__()
        self.margs = margs
        self.groups = groups
        self.layer_idx = layer_idx
        s = groups.strategy
        self.strategy = s
        seq_par = not s.use_ulysses  # megatron-SP always-on with TP
        self.input_norm = build_norm(margs.normalization, margs.hidden_size,
                                     margs.norm_epsilon, dtype)
        self.post_attn_norm = build_norm(margs.normalization, margs.hidden_size,
                                         margs.norm_epsilon, dtype)
        self.attention = SelfAttention(
            margs, groups.tp_group, groups.sp_group, groups.cp_group,
            use_ulysses=s.use_ulysses, sequence_parallel=seq_par, dtype=dtype)
        self.mlp = MLP(margs.hidden_size, margs.ffn_hidden_size,
                       None if s.use_ulysses else groups.tp_group,
                       hidden_act="silu" if margs.hidden_act in ("silu", "swiglu")
                       else margs.hidden_act,
                       add_bias=margs.add_bias_linear,
                       sequence_parallel=seq_par, dtype=dtype,
                       init_method=normal_init(margs.init_method_std),
                       output_init_method=scaled_init(
                           margs.init_method_std, margs.num_hidden_layers))
        if margs.position_embedding_type == "rope":
            self.rotary = RotaryEmbedding(margs.head_dim, margs.rope_theta,
                                          scaling=margs.rope_scaling)
        else:
            self.rotary = None
        if seq_par and group_size(groups.tp_group) > 1:
            _tag_tp_replicated(self.input_norm.weight,
                               getattr(self.input_norm, "bias", None),
                               self.post_attn_norm.weight,
                               getattr(self.post_attn_norm, "bias", None),
                               self.attention.linear_proj.bias,
                               self.mlp.fc2.bias)
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
        # megatron: RoPE applies after the SP allgather -> cp-local full seq
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
        if (self.dropout_p == 0 or not self.training) \
                and hasattr(self.post_attn_norm, "forward_fused_add") \
                and x.shape == residual.shape:
            # fused residual-add + norm (one kernel each way)
            x, residual = self.post_attn_norm.forward_fused_add(x, residual)
        else:
            hidden = residual + x
            residual = hidden
            x = self.post_attn_norm(hidden)
        x = self.mlp(x)
        if self.dropout_p > 0 and self.training:
            x = F.dropout(x, self.dropout_p)
        return residual + x


class GalvatronFinalNorm(nn.Module):
    """Final pre-head norm (reference: modules.py:243-258)."""

    def __init__(self, margs: ModelArgs, groups: LayerCommGroups, dtype=None):
        super().__init__()
        self.norm = build_norm(margs.normalization, margs.hidden_size,
                               margs.norm_epsilon, dtype)
        self.groups = groups
        s = groups.strategy
        if not s.use_ulysses and group_size(groups.tp_group) > 1:
            _tag_tp_replicated(self.norm.weight, getattr(self.norm, "bias", None))

    def forward(self, hidden: torch.Tensor, ctx: Dict) -> torch.Tensor:
        return self.norm(hidden)


class GalvatronCausalLMHead(nn.Module):
    """Vocab-parallel LM head computing the per-token CE loss in forward
    (reference: modules.py:259-347).  Returns per-token losses [rows, b_loc]
    fp32; the engine normalizes by the GLOBAL token count so distributed
    grad-sums reproduce the single-GPU gradient."""

    def __init__(self, margs: ModelArgs, groups: LayerCommGroups, dtype=None):
        super().__init__()
        self.margs = margs
        self.groups = groups
        s = groups.strategy
        self.strategy = s
        if s.use_ulysses:
            self.lm_head = ColumnParallelLinear(
                margs.hidden_size, margs.vocab_size, None, bias=False,
                dtype=dtype,
                init_method=normal_init(margs.init_method_std))
        else:
            self.lm_head = ColumnParallelLinear(
                margs.hidden_size, margs.vocab_size, groups.tp_group,
                bias=False, sequence_parallel=True, dtype=dtype,
                init_method=normal_init(margs.init_method_std))

    def tie_to(self, embedding: GalvatronEmbedding) -> None:
        """Share the vocab-sharded weight with the embedding (same layout)."""
        assert self.lm_head.weight.shape == embedding.word_embeddings.weight.shape, \
            "tied embedding requires identical vocab sharding (vtp == emb tp)"
        self.lm_head.weight = embedding.word_embeddings.weight

    def forward(self, hidden: torch.Tensor, ctx: Dict) -> torch.Tensor:
        labels = ctx["labels"]
        B, S = labels.shape
        s = self.strategy
        c = self.groups.coord_of(_my_rank())
        b_loc = B // s.dp
        labels = labels[c.dp_idx * b_loc:(c.dp_idx + 1) * b_loc]
        logits = self.lm_head(hidden)  # megatron: [S/cp, b, V/vtp]; ulysses: [rows, b, V]
        if s.use_ulysses:
            rows = natural_rows(S, s.cp, s.tp_sp, c.cp_idx, c.tp_idx,
                                labels.device)
        else:
            rows = natural_rows(S, s.cp, 1, c.cp_idx, 0, labels.device)
        tgt = labels[:, rows].transpose(0, 1).contiguous()  # [rows, b_loc]
        group = None if s.use_ulysses else self.groups.tp_group.group
        loss = vocab_parallel_cross_entropy(
            logits.float() if logits.dtype != torch.bfloat16 else logits,
            tgt, group)
        if "loss_mask" in ctx:  # eod_mask_loss (reference get_batch)
            mask = ctx["loss_mask"][c.dp_idx * b_loc:(c.dp_idx + 1) * b_loc]
            mask = mask[:, rows].transpose(0, 1)
            loss = loss * mask.to(loss.dtype)
            ctx["_loss_count_local"] = float(mask.sum())
        return loss  # [rows, b_loc] fp32
