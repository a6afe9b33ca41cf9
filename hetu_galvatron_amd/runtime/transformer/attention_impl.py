"""Attention implementations: local flash, Ulysses SP, zigzag-ring CP.

Reference: galvatron/core/runtime/transformer/attention_impl.py:29-910.
Fixes the reference's CP mis-wiring (ring attention was reachable
only through Ulysses): here ring CP is a first-class dispatch, composable
with Ulysses (sequence first split over SP, then zigzag over CP).

The flash building block is the CDNA4 MFMA kernel (ops/csrc/flash_attn.hip)
returning LSE; the LSE-merge math runs in fp32 on-device.  Ring KV exchange
uses batch_isend_irecv over the CP group — on the MI355X node each hop maps
to one xGMI link, so the per-step [b, s_local, hkv, d] block transfers at
link rate and is overlapped with the flash compute of the current block.
"""
from __future__ import annotations

import math
from typing import Callable, List, Optional, Tuple

import torch
import torch.distributed as dist

from ...ops import (flash_attention, flash_attention_bwd_only,
                    flash_attention_fwd_only, flash_bias_attention)
from ..tensor_parallel.mappings import all_to_all, group_rank, group_size


def local_attention(q, k, v, causal: bool = True,
                    softmax_scale: Optional[float] = None,
                    window: Optional[int] = None,
                    sbhd: bool = False) -> torch.Tensor:
    """Plain local flash attention, autograd-capable. [b, s, h, d]
    (or [s, b, h, d] with sbhd=True — the runtime's native layout, no
    permute copies).  window: mistral sliding window (no-op when
    window >= kv length)."""
    return flash_attention(q, k, v, causal=causal, softmax_scale=softmax_scale,
                           window=window, sbhd=sbhd)


# ---------------------------------------------------------------------------
# Ulysses (DeepSpeed-style) sequence parallelism: heads<->sequence all-to-all
# Reference: attention_impl.py:139-405 (single_all_to_all, DistributedAttention)


def eager_bias_attention(q, k, v, bias, causal: bool, scale: float
                         ) -> torch.Tensor:
    """fp32 softmax attention with an additive score bias [h, sq, skv]
    (t5 relative-position bias; native flash bias input is a v2 kernel
    item).  q/k/v: [b, s, h, d]."""
    b, sq, h, d = q.shape
    skv = k.shape[1]
    if scale is None:
        scale = d ** -0.5
    if k.shape[2] != h:
        rep = h // k.shape[2]
        k = k.repeat_interleave(rep, dim=2)
        v = v.repeat_interleave(rep, dim=2)
    att = torch.einsum("bqhd,bkhd->bhqk", q.float(), k.float()) * scale
    att = att + bias.unsqueeze(0).float()
    if causal:
        mask = torch.ones(sq, skv, dtype=torch.bool, device=q.device) \
            .triu(diagonal=1 + skv - sq)
        att = att.masked_fill(mask, float("-inf"))
    o = torch.einsum("bhqk,bkhd->bqhd", att.softmax(-1), v.float())
    return o.to(q.dtype)


class DistributedAttention(torch.nn.Module):
    """a2a q,k,v (scatter heads, gather seq) -> inner attention -> a2a out.

    inner_attention: callable (q, k, v) -> o on full-sequence tensors; by
    default local flash, but a ZigzagRingAttention can be composed inside
    (CP within each Ulysses shard group).
    """

    def __init__(self, sp_group, inner_attention: Optional[Callable] = None):
        super().__init__()
        self.sp_group = sp_group
        self.inner_attention = inner_attention

    def forward(self, q, k, v, causal=True, softmax_scale=None,
                attn_bias=None, window=None):
        # q: [b, s_local, hq, d]; k/v: [b, s_local, hkv, d].
        # attn_bias [H_full, S, S] (t5 relative bias): sliced to this
        # rank's post-a2a head chunk; incompatible with a ring inner.
        sp = group_size(self.sp_group)
        if sp > 1:
            hkv = k.shape[2]
            if hkv < sp:
                # GQA with fewer KV heads than the a2a fan-out: repeat KV
                # (reference: attention_impl.py:335-345)
                rep = sp // hkv
                k = k.repeat_interleave(rep, dim=2)
                v = v.repeat_interleave(rep, dim=2)
            q = all_to_all(q, self.sp_group, scatter_dim=2, gather_dim=1)
            k = all_to_all(k, self.sp_group, scatter_dim=2, gather_dim=1)
            v = all_to_all(v, self.sp_group, scatter_dim=2, gather_dim=1)
        if attn_bias is not None:
            # slice this rank's post-a2a head chunk; with a ring inner the
            # rows are the cp-local zigzag pair and the ring handles the
            # per-step kv column slices itself
            r = group_rank(self.sp_group)
            hl = q.shape[2]
            bias_l = attn_bias[r * hl:(r + 1) * hl]
            if self.inner_attention is not None:
                o = self.inner_attention(q, k, v, causal=causal,
                                         softmax_scale=softmax_scale,
                                         attn_bias=bias_l)
            else:
                o = flash_bias_attention(q, k, v, bias_l, causal,
                                         softmax_scale)
        elif self.inner_attention is not None:
            o = self.inner_attention(q, k, v, causal=causal, softmax_scale=softmax_scale)
        else:
            o = local_attention(q, k, v, causal=causal,
                                softmax_scale=softmax_scale, window=window)
        if sp > 1:
            o = all_to_all(o, self.sp_group, scatter_dim=1, gather_dim=2)
        return o


# ---------------------------------------------------------------------------
# Zigzag ring context parallelism
# Reference: attention_impl.py:408-910 (RingComm, zigzag fwd/bwd, LSE merge)


class RingComm:
    """Single-step ring exchange on a CP group via batch_isend_irecv
    (reference: attention_impl.py:481-563)."""

    def __init__(self, group):
        self.group = group
        self.rank = group_rank(group)
        self.size = group_size(group)
        ranks = list(group.ranks) if hasattr(group, "ranks") else None
        self._pg = group.group if hasattr(group, "group") else group
        if ranks is None:
            ranks = dist.get_process_group_ranks(self._pg)
        self.ranks = ranks
        self.next_rank = ranks[(self.rank + 1) % self.size]
        self.prev_rank = ranks[(self.rank - 1) % self.size]
        self._ops: List[dist.P2POp] = []
        self._reqs = None

    def send_recv(self, send: torch.Tensor, recv: Optional[torch.Tensor] = None):
        if recv is None:
            recv = torch.empty_like(send)
        self._ops.append(dist.P2POp(dist.isend, send.contiguous(), self.next_rank,
                                    group=self._pg))
        self._ops.append(dist.P2POp(dist.irecv, recv, self.prev_rank,
                                    group=self._pg))
        return recv

    def commit(self):
        """Launch the batched ops; returns the request list so callers can
        keep several batches in flight (kv rotation overlapping dkv)."""
        self._reqs = dist.batch_isend_irecv(self._ops)
        self._ops = []
        return self._reqs

    def wait(self, reqs=None):
        reqs = self._reqs if reqs is None else reqs
        if reqs:
            for r in reqs:
                r.wait()
        if reqs is self._reqs:
            self._reqs = None


def _merge_attn_out(o: Optional[torch.Tensor], lse: Optional[torch.Tensor],
                    o_i: torch.Tensor, lse_i: torch.Tensor,
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Online LSE merge (reference: attention_impl.py:437-478).
    o: [b, s, h, d]; lse: [b, h, s] (fp32)."""
    if o is None:
        return o_i.float(), lse_i
    new_lse = torch.logaddexp(lse, lse_i)
    w_old = torch.exp(lse - new_lse).transpose(1, 2).unsqueeze(-1)  # [b,s,h,1]
    w_new = torch.exp(lse_i - new_lse).transpose(1, 2).unsqueeze(-1)
    o = o * w_old + o_i.float() * w_new
    return o, new_lse


def _zigzag_cols(S: int, n: int, j: int, device) -> torch.Tensor:
    """Natural col indices of rank j's zigzag kv chunks (j, 2n-1-j)."""
    cs = S // (2 * n)
    return torch.cat([torch.arange(j * cs, (j + 1) * cs, device=device),
                      torch.arange((2 * n - 1 - j) * cs,
                                   (2 * n - j) * cs, device=device)])


def zigzag_ring_flash_attn_fwd(q, k, v, comm: RingComm, softmax_scale: float,
                               causal: bool = True, bias=None
                               ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Forward ring pass over zigzag-sharded sequence.

    Chunk math (causal): rank r holds q/kv chunks (r, 2n-1-r) of 2n.  KV
    from rank j:
      j == r : local causal attention
      j <  r : BOTH q chunks attend the FIRST kv half, un-masked
      j >  r : only the SECOND q half attends the FULL kv, un-masked
    bias [h, s_local, S_global]: t5 relative bias rows for THIS rank's
    packed zigzag q rows over the natural global kv axis; per-step col
    slices follow the chunk math.  causal=False (t5 encoder): every step
    attends rank j's full kv un-masked.
    (reference: attention_impl.py:564-653)
    """
    n, r = comm.size, comm.rank
    b, s_local, hq, d = q.shape
    S = s_local * n
    half = s_local // 2
    cs = S // (2 * n)
    o_acc: Optional[torch.Tensor] = None
    lse_acc: Optional[torch.Tensor] = None
    o2_acc: Optional[torch.Tensor] = None  # second-half-only accumulator
    lse2_acc: Optional[torch.Tensor] = None
    k_cur, v_cur = k, v
    for step in range(n):
        if step + 1 < n:
            k_nxt = comm.send_recv(k_cur)
            v_nxt = comm.send_recv(v_cur)
            comm.commit()
        j = (r - step) % n

        def bcols(jj):
            return None if bias is None else \
                bias[:, :, _zigzag_cols(S, n, jj, q.device)]

        if not causal:
            o_i, lse_i = flash_attention_fwd_only(
                q, k_cur, v_cur, causal=False, softmax_scale=softmax_scale,
                bias=bcols(j))
            o_acc, lse_acc = _merge_attn_out(o_acc, lse_acc, o_i, lse_i)
        elif step == 0:
            o_i, lse_i = flash_attention_fwd_only(q, k_cur, v_cur, causal=True,
                                                  softmax_scale=softmax_scale,
                                                  bias=bcols(r))
            o_acc, lse_acc = _merge_attn_out(o_acc, lse_acc, o_i, lse_i)
        elif j < r:
            bb = None if bias is None else bias[:, :, j * cs:(j + 1) * cs]
            o_i, lse_i = flash_attention_fwd_only(
                q, k_cur[:, :half], v_cur[:, :half], causal=False,
                softmax_scale=softmax_scale, bias=bb)
            o_acc, lse_acc = _merge_attn_out(o_acc, lse_acc, o_i, lse_i)
        else:
            bb = None if bias is None else bcols(j)[:, half:]
            o_i, lse_i = flash_attention_fwd_only(
                q[:, half:], k_cur, v_cur, causal=False,
                softmax_scale=softmax_scale, bias=bb)
            o2_acc, lse2_acc = _merge_attn_out(o2_acc, lse2_acc, o_i, lse_i)
        if step + 1 < n:
            comm.wait()
            k_cur, v_cur = k_nxt, v_nxt
    if o2_acc is not None:
        o_half, lse_half = _merge_attn_out(
            o_acc[:, half:].contiguous(), lse_acc[:, :, half:].contiguous(),
            o2_acc.to(q.dtype), lse2_acc)
        o_acc = torch.cat([o_acc[:, :half], o_half], dim=1)
        lse_acc = torch.cat([lse_acc[:, :, :half], lse_half], dim=2)
    return o_acc.to(q.dtype), lse_acc


def zigzag_ring_flash_attn_bwd(do, q, k, v, o, lse, comm: RingComm,
                               softmax_scale: float, causal: bool = True,
                               bias=None):
    """Backward ring pass: dq accumulates locally; (k, v, dk, dv) rotate
    together n steps and arrive home fully accumulated.  With a bias,
    also accumulates dbias [h, s_local, S_global] on this rank's q rows
    (the bias table's cross-cp grad sum rides the sdp reduction, which
    spans cp) (reference: attention_impl.py:654-782)."""
    n, r = comm.size, comm.rank
    b, s_local, hq, d = q.shape
    S = s_local * n
    half = s_local // 2
    cs = S // (2 * n)
    dq = torch.zeros_like(q, dtype=torch.float32)
    dbias = None if bias is None else torch.zeros_like(bias,
                                                       dtype=torch.float32)
    k_cur, v_cur = k, v
    dk_cur = torch.zeros_like(k, dtype=torch.float32)
    dv_cur = torch.zeros_like(v, dtype=torch.float32)
    o_half = o[:, half:].contiguous()
    do_half = do[:, half:].contiguous()
    lse_half = lse[:, :, half:].contiguous()
    q_half = q[:, half:].contiguous()
    # comm/compute overlap (the forward already pre-posts its sends): the
    # next step's (k, v) rotation is in flight during this step's flash
    # kernels, and the accumulated (dk, dv) rotation posted after step s
    # completes during step s+1's compute — nothing blocks inline except
    # the final homecoming rotation.
    kv_reqs = dkv_reqs = None
    k_nxt = v_nxt = dk_nxt = dv_nxt = None
    if n > 1:
        k_nxt = comm.send_recv(k_cur)
        v_nxt = comm.send_recv(v_cur)
        kv_reqs = comm.commit()
    for step in range(n):
        j = (r - step) % n
        cols = _zigzag_cols(S, n, j, q.device) if bias is not None else None
        # -- compute this step's flash backward (kernels launch async) ----
        if not causal:
            mode = "full"
            res = flash_attention_bwd_only(
                do, q, k_cur, v_cur, o, lse, causal=False,
                softmax_scale=softmax_scale,
                bias=None if bias is None else bias[:, :, cols])
        elif step == 0:
            mode = "full"
            res = flash_attention_bwd_only(
                do, q, k_cur, v_cur, o, lse, causal=True,
                softmax_scale=softmax_scale,
                bias=None if bias is None else bias[:, :, cols])
        elif j < r:
            mode = "kv_half"
            res = flash_attention_bwd_only(
                do, q, k_cur[:, :half].contiguous(), v_cur[:, :half].contiguous(),
                o, lse, causal=False, softmax_scale=softmax_scale,
                bias=None if bias is None else
                bias[:, :, j * cs:(j + 1) * cs])
        else:
            mode = "q_half"
            res = flash_attention_bwd_only(
                do_half, q_half, k_cur, v_cur, o_half, lse_half, causal=False,
                softmax_scale=softmax_scale,
                bias=None if bias is None else bias[:, half:][:, :, cols])
        # -- the dkv posted after step-1 completed during the kernels above
        if step > 0:
            comm.wait(dkv_reqs)
            dk_cur, dv_cur = dk_nxt, dv_nxt
        # -- accumulate -----------------------------------------------------
        if mode == "full":
            dq += res[0].float()
            dk_cur += res[1].float()
            dv_cur += res[2].float()
            if bias is not None:
                dbias[:, :, cols] += res[3]
        elif mode == "kv_half":
            dq += res[0].float()
            dk_cur[:, :half] += res[1].float()
            dv_cur[:, :half] += res[2].float()
            if bias is not None:
                dbias[:, :, j * cs:(j + 1) * cs] += res[3]
        else:
            dq[:, half:] += res[0].float()
            dk_cur += res[1].float()
            dv_cur += res[2].float()
            if bias is not None:
                db = dbias[:, half:]
                db[:, :, cols] += res[3]
        # -- rotate: accumulated dkv now; kv for the step after next --------
        dk_nxt = comm.send_recv(dk_cur)
        dv_nxt = comm.send_recv(dv_cur)
        dkv_reqs = comm.commit()
        if step < n - 1:
            comm.wait(kv_reqs)
            k_cur, v_cur = k_nxt, v_nxt
            if step < n - 2:
                k_nxt = comm.send_recv(k_cur)
                v_nxt = comm.send_recv(v_cur)
                kv_reqs = comm.commit()
    comm.wait(dkv_reqs)  # final homecoming rotation
    dk_cur, dv_cur = dk_nxt, dv_nxt
    out = (dq.to(q.dtype), dk_cur.to(k.dtype), dv_cur.to(v.dtype))
    return out + ((dbias,) if bias is not None else ())


class ZigzagRingFlashAttnFunc(torch.autograd.Function):
    """reference: attention_impl.py:785 ZigZagRingFlashAttnFunc."""

    @staticmethod
    def forward(ctx, q, k, v, bias, cp_group, softmax_scale, causal):
        scale = softmax_scale if softmax_scale is not None else 1.0 / math.sqrt(q.shape[-1])
        comm = RingComm(cp_group)
        bias_c = bias.contiguous() if bias is not None else None
        o, lse = zigzag_ring_flash_attn_fwd(q.contiguous(), k.contiguous(),
                                            v.contiguous(), comm, scale,
                                            causal=causal, bias=bias_c)
        ctx.save_for_backward(q, k, v, o, lse,
                              *([] if bias is None else [bias_c]))
        ctx.cp_group = cp_group
        ctx.scale = scale
        ctx.causal = causal
        ctx.has_bias = bias is not None
        return o

    @staticmethod
    def backward(ctx, do):
        if ctx.has_bias:
            q, k, v, o, lse, bias = ctx.saved_tensors
        else:
            q, k, v, o, lse = ctx.saved_tensors
            bias = None
        comm = RingComm(ctx.cp_group)
        res = zigzag_ring_flash_attn_bwd(
            do.contiguous(), q, k, v, o, lse, comm, ctx.scale,
            causal=ctx.causal, bias=bias)
        dbias = res[3].to(bias.dtype) if bias is not None else None
        return res[0], res[1], res[2], dbias, None, None, None


class ZigzagRingAttention(torch.nn.Module):
    """First-class CP dispatch (fixing the reference's CP-only gap).
    Expects zigzag-sharded [b, s_local, h, d] inputs."""

    def __init__(self, cp_group):
        super().__init__()
        self.cp_group = cp_group

    def forward(self, q, k, v, causal=True, softmax_scale=None,
                attn_bias=None):
        """attn_bias [h, s_local, S_global]: this rank's packed zigzag q
        rows over the natural global kv axis (t5 relative bias)."""
        if group_size(self.cp_group) == 1:
            if attn_bias is not None:
                return flash_bias_attention(q, k, v, attn_bias, causal,
                                            softmax_scale)
            return local_attention(q, k, v, causal=causal, softmax_scale=softmax_scale)
        return ZigzagRingFlashAttnFunc.apply(q, k, v, attn_bias,
                                             self.cp_group, softmax_scale,
                                             causal)
