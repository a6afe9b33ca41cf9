"""Self-attention with per-layer TP / Megatron-SP / Ulysses / zigzag-CP.

Reference: galvatron/core/runtime/transformer/attention.py:111-1017.
Dispatch (attention.py:515-732 in the reference, CP-only gap fixed):
    tp>1 (megatron)      : fused QKV CPL (SP-allgather inside) -> heads
                           sharded; inner attention over full cp-local seq
    ulysses sp>1         : heads full, seq sharded; DistributedAttention a2a
    cp>1                 : inner attention = ZigzagRingAttention (composable
                           with either of the above)
GQA interleaved-group QKV layout: the fused weight is ordered by KV group
[q*(hq/hkv), k, v] so ColumnParallelLinear's contiguous shard = whole groups
(checkpoint adapters interleave accordingly; reference llama_adapter.py).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from ..tensor_parallel import (
    ColumnParallelLinear,
    RowParallelLinear,
    group_size,
)
from ..tensor_parallel.layers import normal_init, scaled_init
from .attention_impl import (
    DistributedAttention, ZigzagRingAttention, eager_bias_attention,
    flash_bias_attention, local_attention,
)
from .rope import apply_rope_qk


def _dropout_attention(q, k, v, causal, scale, p):
    """[b,s,h,d] eager sdpa with attention dropout (training only)."""
    import torch.nn.functional as F
    hq, hkv = q.shape[2], k.shape[2]
    qh = q.permute(0, 2, 1, 3)
    kh = k.permute(0, 2, 1, 3)
    vh = v.permute(0, 2, 1, 3)
    if hq != hkv:
        kh = kh.repeat_interleave(hq // hkv, dim=1)
        vh = vh.repeat_interleave(hq // hkv, dim=1)
    o = F.scaled_dot_product_attention(qh, kh, vh, is_causal=causal,
                                       dropout_p=p, scale=scale)
    return o.permute(0, 2, 1, 3)


class SelfAttention(nn.Module):
    def __init__(self, model_args, tp_group, sp_group, cp_group,
                 use_ulysses: bool = False, sequence_parallel: bool = True,
                 dtype=None, causal: bool = True):
        super().__init__()
        self.causal = causal
        m = model_args
        self.hidden_size = m.hidden_size
        self.num_heads = m.num_attention_heads
        self.num_kv_heads = m.kv_heads
        self.head_dim = m.head_dim
        self.use_ulysses = use_ulysses
        self.tp_group = tp_group
        self.sp_group = sp_group
        self.cp_group = cp_group
        tp = 1 if use_ulysses else group_size(tp_group)
        self.tp = tp
        assert self.num_heads % tp == 0, "tp must divide num heads"
        assert self.num_kv_heads % tp == 0 or tp % self.num_kv_heads == 0, \
            "tp must divide kv heads (kv replication unsupported yet)"
        assert self.num_kv_heads % tp == 0, \
            f"tp={tp} > kv_heads={self.num_kv_heads} needs KV replication"
        self.num_groups_local = self.num_kv_heads // tp
        self.q_per_group = self.num_heads // self.num_kv_heads
        self.heads_local = self.num_heads // tp
        qkv_out = (self.num_heads + 2 * self.num_kv_heads) * self.head_dim
        lin_group = tp_group if not use_ulysses else None
        std = getattr(m, "init_method_std", 0.02)
        L = m.num_hidden_layers
        self.linear_qkv = ColumnParallelLinear(
            self.hidden_size, qkv_out, lin_group,
            bias=m.add_qkv_bias or m.add_bias_linear,
            sequence_parallel=sequence_parallel and not use_ulysses, dtype=dtype,
            init_method=normal_init(std))
        self.linear_proj = RowParallelLinear(
            self.num_heads * self.head_dim, self.hidden_size, lin_group,
            bias=m.add_bias_linear,
            sequence_parallel=sequence_parallel and not use_ulysses, dtype=dtype,
            init_method=scaled_init(std, L))
        self.softmax_scale = m.attention_softmax_scale \
            if getattr(m, "attention_softmax_scale", None) is not None \
            else 1.0 / math.sqrt(self.head_dim)
        self.window = getattr(m, "sliding_window", None)
        self.rope_interleaved = bool(getattr(m, "rotary_interleaved", False))
        # qk_layernorm (reference attention.py:917-921, Qwen3/Llama4/
        # Gemma2): per-head-dim norm on q and k after the QKV split,
        # before RoPE; params are tp-replicated (every tp rank normalizes
        # its own heads with the same [head_dim] weight)
        self.q_layernorm = self.k_layernorm = None
        if getattr(m, "qk_layernorm", False):
            from .norm import LayerNorm, RMSNorm
            ncls = RMSNorm if m.normalization == "rmsnorm" else LayerNorm
            self.q_layernorm = ncls(self.head_dim, eps=m.norm_epsilon,
                                    dtype=dtype)
            self.k_layernorm = ncls(self.head_dim, eps=m.norm_epsilon,
                                    dtype=dtype)
            for p_ in list(self.q_layernorm.parameters()) +                     list(self.k_layernorm.parameters()):
                p_.tp_replicated = True
        cp = group_size(cp_group) if cp_group is not None else 1
        if self.window is not None:
            assert cp == 1, "sliding window + ring-CP is a v2 item"
        # attention dropout (reference: flash_attn dropout arg) — eager
        # sdpa path when active; composes with ulysses (full-seq inner)
        # but not with ring-CP (dropout inside per-block softmax breaks
        # the LSE merge) or a sliding window
        self.attn_dropout = float(getattr(m, "attention_dropout", 0.0) or 0.0)
        if self.attn_dropout > 0:
            assert cp == 1, "attention_dropout + ring-CP is unsupported"
            assert self.window is None, \
                "attention_dropout + sliding window is unsupported"
        inner = ZigzagRingAttention(cp_group) if cp > 1 else None
        if use_ulysses and group_size(sp_group) > 1:
            sp = group_size(sp_group)
            assert self.num_heads % sp == 0, \
                f"ulysses sp={sp} must divide num_heads={self.num_heads}"
            self.core_attention = DistributedAttention(
                sp_group, inner_attention=inner)
        elif inner is not None:
            self.core_attention = inner
        else:
            self.core_attention = None  # plain local flash

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                attn_bias: torch.Tensor = None) -> torch.Tensor:
        """x: [s_local, b, h] SBH; cos/sin: tables matching the seq layout at
        RoPE time (megatron-sp: cp-local full seq; ulysses: the local slice).
        attn_bias: additive scores bias (t5 relative bias) — eager fp32
        softmax path; megatron-tp gets [h_local, sq, skv], ulysses the
        full-head table (sliced post-a2a), ring-CP the cp-local rows
        over the global kv axis."""
        qkv = self.linear_qkv(x)  # [s, b, (hq_l + 2*hkv_l)*d]
        s, b = qkv.shape[0], qkv.shape[1]
        qkv = qkv.view(s, b, self.num_groups_local, self.q_per_group + 2,
                       self.head_dim)
        q = qkv[:, :, :, : self.q_per_group].reshape(s, b, -1, self.head_dim)
        k = qkv[:, :, :, self.q_per_group].reshape(s, b, -1, self.head_dim)
        v = qkv[:, :, :, self.q_per_group + 1].reshape(s, b, -1, self.head_dim)
        if self.q_layernorm is not None:
            q = self.q_layernorm(q)
            k = self.k_layernorm(k)
        q, k = apply_rope_qk(q.contiguous(), k.contiguous(), cos, sin,
                             interleaved=self.rope_interleaved)
        if self.attn_dropout > 0 and self.training and attn_bias is None:
            drop = self.attn_dropout
            if isinstance(self.core_attention, DistributedAttention):
                qb = q.permute(1, 0, 2, 3).contiguous()
                kb = k.permute(1, 0, 2, 3).contiguous()
                vb = v.permute(1, 0, 2, 3).contiguous()
                inner = lambda qq, kk, vv, causal=True, softmax_scale=None, \
                    window=None: _dropout_attention(
                        qq, kk, vv, causal, softmax_scale, drop)
                saved = self.core_attention.inner_attention
                self.core_attention.inner_attention = inner
                try:
                    o = self.core_attention(qb, kb, vb, causal=self.causal,
                                            softmax_scale=self.softmax_scale)
                finally:
                    self.core_attention.inner_attention = saved
                o = o.permute(1, 0, 2, 3).reshape(s, b, -1)
                return self.linear_proj(o)
            assert self.core_attention is None, \
                "attention_dropout composes with ulysses/local only"
            o = _dropout_attention(
                q.permute(1, 0, 2, 3).contiguous(),
                k.permute(1, 0, 2, 3).contiguous(),
                v.permute(1, 0, 2, 3).contiguous(),
                self.causal, self.softmax_scale, drop)
            return self.linear_proj(o.permute(1, 0, 2, 3).reshape(s, b, -1))
        if self.core_attention is None and attn_bias is None:
            # plain local flash takes the native [s,b,h,d] layout directly
            # (sbhd kernels) — no permute+contiguous copies either way
            o = local_attention(q, k, v.contiguous(), causal=self.causal,
                                softmax_scale=self.softmax_scale,
                                window=self.window, sbhd=True)
            return self.linear_proj(o.reshape(s, b, -1))
        # [s,b,h,d] -> [b,s,h,d] for the distributed attention paths
        q = q.permute(1, 0, 2, 3).contiguous()
        k = k.permute(1, 0, 2, 3).contiguous()
        v = v.permute(1, 0, 2, 3).contiguous()
        if attn_bias is not None and isinstance(
                self.core_attention, (DistributedAttention,
                                      ZigzagRingAttention)):
            # ulysses: full-head bias, the a2a re-shards heads and
            # DistributedAttention slices the matching chunk.
            # ring-CP: bias rows are this rank's packed zigzag q rows over
            # the natural global kv axis (the layer builds it that way)
            o = self.core_attention(q, k, v, causal=self.causal,
                                    softmax_scale=self.softmax_scale,
                                    attn_bias=attn_bias)
        elif attn_bias is not None:
            o = flash_bias_attention(q, k, v, attn_bias, self.causal,
                                     self.softmax_scale)
        elif self.core_attention is not None:
            # ulysses a2a keeps the global seq intact: window semantics
            # apply unchanged to the inner full-seq attention
            if self.window is not None and isinstance(
                    self.core_attention, DistributedAttention):
                assert self.core_attention.inner_attention is None
                o = self.core_attention(q, k, v, causal=self.causal,
                                        softmax_scale=self.softmax_scale,
                                        window=self.window)
            else:
                o = self.core_attention(q, k, v, causal=self.causal,
                                        softmax_scale=self.softmax_scale)
        else:
            o = local_attention(q, k, v, causal=self.causal,
                                softmax_scale=self.softmax_scale,
                                window=self.window)
        o = o.permute(1, 0, 2, 3).reshape(s, b, -1)  # back to SBH
        return self.linear_proj(o)


class CrossAttention(nn.Module):
    """Decoder->encoder cross attention (reference: attention.py:929
    CrossAttention).  Queries come from the (possibly seq-sharded) decoder
    hidden state; keys/values from the FULL-sequence encoder memory the
    EncDecBridge replicated within the dp group, so no redistribution is
    needed however the decoder layer is laid out.

    megatron tp: heads shard both q and kv via the parallel linears.
    ulysses sp: linears replicated; q a2a's decoder-seq->heads (like
    DistributedAttention), kv computed on the full memory and sliced to
    this rank's post-a2a head chunk (its weight-grad slices re-assemble
    through the ulysses sdp all-reduce, which spans the sp group).
    """

    def __init__(self, model_args, tp_group, sequence_parallel: bool = True,
                 dtype=None, sp_group=None, use_ulysses: bool = False):
        super().__init__()
        m = model_args
        self.num_heads = m.num_attention_heads
        self.num_kv_heads = m.kv_heads
        self.head_dim = m.head_dim
        self.tp_group = tp_group
        self.sp_group = sp_group
        self.use_ulysses = use_ulysses
        lin_group = None if use_ulysses else tp_group
        seq_par = sequence_parallel and not use_ulysses
        tp = 1 if use_ulysses else group_size(tp_group)
        assert self.num_heads % tp == 0 and self.num_kv_heads % tp == 0
        if use_ulysses and sp_group is not None:
            sp = group_size(sp_group)
            assert self.num_heads % sp == 0, "sp must divide heads"
        std = getattr(m, "init_method_std", 0.02)
        self.linear_q = ColumnParallelLinear(
            m.hidden_size, self.num_heads * self.head_dim, lin_group,
            bias=m.add_qkv_bias or m.add_bias_linear,
            sequence_parallel=seq_par, dtype=dtype,
            init_method=normal_init(std))
        # kv over the full-seq memory: no sequence-parallel gather
        self.linear_kv = ColumnParallelLinear(
            m.hidden_size, 2 * self.num_kv_heads * self.head_dim, lin_group,
            bias=m.add_qkv_bias or m.add_bias_linear,
            sequence_parallel=False, dtype=dtype,
            init_method=normal_init(std))
        self.linear_proj = RowParallelLinear(
            self.num_heads * self.head_dim, m.hidden_size, lin_group,
            bias=m.add_bias_linear, sequence_parallel=seq_par,
            dtype=dtype,
            init_method=scaled_init(std, m.num_hidden_layers))
        self.softmax_scale = m.attention_softmax_scale \
            if getattr(m, "attention_softmax_scale", None) is not None \
            else 1.0 / math.sqrt(self.head_dim)
        self.heads_local = self.num_heads // tp
        self.kv_heads_local = self.num_kv_heads // tp

    def forward(self, x: torch.Tensor, memory: torch.Tensor) -> torch.Tensor:
        """x: [s_dec(_shard), b, h]; memory: [s_enc, b, h] full sequence."""
        from ..tensor_parallel.mappings import all_to_all, group_rank
        q = self.linear_q(x)            # [s_dec, b, hq_l*d]
        kv = self.linear_kv(memory)     # [s_enc, b, 2*hkv_l*d]
        s, bsz = q.shape[0], q.shape[1]
        se = kv.shape[0]
        q = q.view(s, bsz, self.heads_local, self.head_dim)
        kv = kv.view(se, bsz, 2, self.kv_heads_local, self.head_dim)
        k, v = kv[:, :, 0], kv[:, :, 1]
        q = q.permute(1, 0, 2, 3).contiguous()
        k = k.permute(1, 0, 2, 3).contiguous()
        v = v.permute(1, 0, 2, 3).contiguous()
        sp = group_size(self.sp_group) if (self.use_ulysses and
                                           self.sp_group is not None) else 1
        if sp > 1:
            hkv = k.shape[2]
            if hkv < sp:
                rep = sp // hkv
                k = k.repeat_interleave(rep, dim=2)
                v = v.repeat_interleave(rep, dim=2)
            q = all_to_all(q, self.sp_group, scatter_dim=2, gather_dim=1)
            # memory kv is already full-seq: take this rank's head chunk
            r = group_rank(self.sp_group)
            hl = k.shape[2] // sp
            k = k[:, :, r * hl:(r + 1) * hl].contiguous()
            v = v[:, :, r * hl:(r + 1) * hl].contiguous()
        o = local_attention(q, k, v, causal=False,
                            softmax_scale=self.softmax_scale)
        if sp > 1:
            o = all_to_all(o, self.sp_group, scatter_dim=1, gather_dim=2)
        o = o.permute(1, 0, 2, 3).reshape(s, bsz, -1)
        return self.linear_proj(o)
