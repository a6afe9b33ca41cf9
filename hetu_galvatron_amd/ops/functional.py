"""Autograd-wrapped ops dispatching native HIP (GPU) / torch reference (CPU).

Every op here replaces an external CUDA dependency of the reference:
  rms_norm / layer_norm  <- flash_attn.ops.rms_norm / dropout_layer_norm
                            (reference: core/runtime/transformer/norm.py:3-30)
  swiglu / bias-gelu     <- torch.jit fused kernels
                            (reference: transformer/fused_kernels.py:143-226)
  apply_rope             <- flash_attn rotary_emb ext
                            (reference: transformer/fused_kernels.py:227-257)
  flash_attention        <- flash_attn_varlen CUDA
                            (reference: transformer/attention_impl.py:18-108)
"""
from __future__ import annotations

import math
from typing import Optional

import torch

from . import reference_ops as ref
from ._ext import get_ext, native_available, use_native


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        if use_native(x):
            y, invrms = get_ext(False).rmsnorm_fwd(x, weight, eps)
        else:
            y, invrms = ref.rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, invrms)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, invrms = ctx.saved_tensors
        if use_native(x):
            dx, dw = get_ext(False).rmsnorm_bwd(dy.contiguous(), x, weight, invrms)
        else:
            dx, dw = ref.rmsnorm_bwd(dy, x, weight, invrms)
        return dx, dw.to(weight.dtype), None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    return _RMSNorm.apply(x.contiguous(), weight, eps)


class _AddRMSNorm(torch.autograd.Function):
    """Fused residual-add + RMSNorm: (y, sum) = (rmsnorm(x+res)*w, x+res).
    One kernel each way — removes the standalone residual-add in forward
    AND the grad-join add in backward (the reference fuses the same way
    via flash-attn's DropoutAddLayerNorm, norm.py:3-30)."""

    @staticmethod
    def forward(ctx, x, res, weight, eps):
        y, s, invrms = get_ext(False).add_rmsnorm_fwd(x, res, weight, eps)
        ctx.save_for_backward(s, weight, invrms)
        return y, s

    @staticmethod
    def backward(ctx, dy, dsum):
        s, weight, invrms = ctx.saved_tensors
        dx, dw = get_ext(False).add_rmsnorm_bwd(
            dy.contiguous(), dsum.contiguous(), s, weight, invrms)
        return dx, dx, dw.to(weight.dtype), None


def fused_add_rms_norm(x, res, weight, eps: float = 1e-5):
    """Returns (normed, sum).  Native bf16 path only; callers fall back to
    `rms_norm(x + res)` elsewhere."""
    return _AddRMSNorm.apply(x.contiguous(), res.contiguous(), weight, eps)


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        if use_native(x):
            y, mean, invstd = get_ext(False).layernorm_fwd(x, weight, bias, eps)
        else:
            y, mean, invstd = ref.layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, invstd = ctx.saved_tensors
        if use_native(x):
            dx, dw, db = get_ext(False).layernorm_bwd(dy.contiguous(), x, weight, mean, invstd)
        else:
            dx, dw, db = ref.layernorm_bwd(dy, x, weight, mean, invstd)
        return dx, dw.to(weight.dtype), db.to(weight.dtype), None


def layer_norm(x, weight, bias, eps: float = 1e-5) -> torch.Tensor:
    return _LayerNorm.apply(x.contiguous(), weight, bias, eps)


class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        if use_native(x):
            return get_ext(False).swiglu_fwd(x)
        return ref.swiglu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        if use_native(x):
            return get_ext(False).swiglu_bwd(dy.contiguous(), x)
        return ref.swiglu_bwd(dy, x)


def swiglu(x: torch.Tensor) -> torch.Tensor:
    """x[..., 2F] = [gate, up] -> silu(gate)*up."""
    return _SwiGLU.apply(x.contiguous())


class _RoPE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin):
        ctx.save_for_backward(cos, sin)
        if use_native(x):
            return get_ext(False).rope_fwd(x, cos, sin, False)
        return ref.rope_apply(x, cos, sin, conj=False)

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        dy = dy.contiguous()
        if use_native(dy):
            return get_ext(False).rope_fwd(dy, cos, sin, True), None, None
        return ref.rope_apply(dy, cos, sin, conj=True), None, None


def apply_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """x: [s, b, h, d]; cos/sin: [s, d/2] (NEOX half-rotation)."""
    return _RoPE.apply(x.contiguous(), cos, sin)


class _FlashAttention(torch.autograd.Function):
    """Flash attention with LSE output.

    Layout [b, s, h, d] bf16/fp32; GQA supported (hq multiple of hkv);
    causal masking bottom-right aligned.  Native path: CDNA4 MFMA kernel
    (ops/csrc/flash_attn.hip).
    """

    @staticmethod
    def forward(ctx, q, k, v, causal, softmax_scale, window, sbhd):
        scale = softmax_scale if softmax_scale is not None else 1.0 / math.sqrt(q.shape[-1])
        skv = k.shape[0] if sbhd else k.shape[1]
        # an effective window (window < kv length) routes to the eager
        # reference path (a window argument on the native kernel is a v2
        # item); window >= kv length is a no-op -> native
        eff_window = window if (window is not None and
                                window < skv) else None
        if use_native(q) \
                and q.dtype == torch.bfloat16 and q.shape[-1] in (64, 128) \
                and (eff_window is None or causal):
            o, lse = get_ext(False).flash_attn_fwd(q, k, v, causal, scale,
                                                   None, sbhd,
                                                   eff_window or 0)
        elif sbhd:
            ob, lse = ref.attention_fwd(
                q.permute(1, 0, 2, 3), k.permute(1, 0, 2, 3),
                v.permute(1, 0, 2, 3), causal, scale, window=eff_window)
            o = ob.permute(1, 0, 2, 3).contiguous()
        else:
            o, lse = ref.attention_fwd(q, k, v, causal, scale,
                                       window=eff_window)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.scale = scale
        ctx.window = eff_window
        ctx.sbhd = sbhd
        return o, lse

    @staticmethod
    def backward(ctx, do, dlse):
        q, k, v, o, lse = ctx.saved_tensors
        do = do.contiguous()
        if use_native(q) \
                and q.dtype == torch.bfloat16 and q.shape[-1] in (64, 128) \
                and (ctx.window is None or ctx.causal):
            dq, dk, dv = get_ext(False).flash_attn_bwd(
                do, q, k, v, o, lse, ctx.causal, ctx.scale, None, ctx.sbhd,
                ctx.window or 0)
        elif ctx.sbhd:
            dq, dk, dv = ref.attention_bwd(
                do.permute(1, 0, 2, 3), q.permute(1, 0, 2, 3),
                k.permute(1, 0, 2, 3), v.permute(1, 0, 2, 3),
                o.permute(1, 0, 2, 3), lse, ctx.causal, ctx.scale,
                window=ctx.window)
            dq = dq.permute(1, 0, 2, 3).contiguous()
            dk = dk.permute(1, 0, 2, 3).contiguous()
            dv = dv.permute(1, 0, 2, 3).contiguous()
        else:
            dq, dk, dv = ref.attention_bwd(do, q, k, v, o, lse, ctx.causal,
                                           ctx.scale, window=ctx.window)
        return dq, dk, dv, None, None, None, None


def flash_attention(q, k, v, causal: bool = True,
                    softmax_scale: Optional[float] = None,
                    return_lse: bool = False, window: Optional[int] = None,
                    sbhd: bool = False):
    """q: [b,s,hq,d]; k,v: [b,s,hkv,d] -> o [b,s,hq,d] (+ lse [b,hq,s]).
    sbhd=True: tensors are [s,b,h,d] (the runtime's native activation
    layout) — saves the permute+contiguous copies around the kernel.
    window: mistral-style sliding window (causal only)."""
    o, lse = _FlashAttention.apply(q.contiguous(), k.contiguous(), v.contiguous(),
                                   causal, softmax_scale, window, sbhd)
    return (o, lse) if return_lse else o


class _FlashBiasAttention(torch.autograd.Function):
    """Biased flash attention (t5 relative bias [hq, sq, skv]) with grads
    for q/k/v AND the bias (dbias batch-summed, flows back into the bias
    table's construction graph).  Native CDNA4 path when available."""

    @staticmethod
    def forward(ctx, q, k, v, bias, causal, scale):
        o, lse = flash_attention_fwd_only(q, k, v, causal, scale, bias=bias)
        ctx.save_for_backward(q, k, v, o, lse, bias)
        ctx.causal = causal
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse, bias = ctx.saved_tensors
        dq, dk, dv, dbias = flash_attention_bwd_only(
            do.contiguous(), q, k, v, o, lse, ctx.causal, ctx.scale,
            bias=bias)
        return dq, dk, dv, dbias.to(bias.dtype), None, None


def flash_bias_attention(q, k, v, bias, causal: bool = True,
                         softmax_scale: Optional[float] = None):
    """q: [b,s,hq,d]; bias [hq, sq, skv] additive pre-softmax scores."""
    scale = softmax_scale if softmax_scale is not None \
        else 1.0 / math.sqrt(q.shape[-1])
    return _FlashBiasAttention.apply(q.contiguous(), k.contiguous(),
                                     v.contiguous(), bias, causal, scale)


def flash_attention_fwd_only(q, k, v, causal=True, softmax_scale=None,
                             bias=None, window=None):
    """No-autograd forward returning (o, lse) — building block for ring CP
    and the generator prefill.  bias [hq, sq, skv] (t5 relative bias) and
    mistral sliding `window` both take the native CDNA4 kernel path."""
    scale = softmax_scale if softmax_scale is not None else 1.0 / math.sqrt(q.shape[-1])
    eff_window = window if (window is not None and
                            window < k.shape[1]) else None
    if use_native(q) and q.dtype == torch.bfloat16 \
            and q.shape[-1] in (64, 128) \
            and (bias is None or bias.shape[-1] % 4 == 0) \
            and (eff_window is None or (causal and bias is None)):
        bias_n = None if bias is None else bias.to(torch.bfloat16).contiguous()
        return get_ext(False).flash_attn_fwd(q.contiguous(), k.contiguous(),
                                             v.contiguous(), causal, scale,
                                             bias_n, False, eff_window or 0)
    return ref.attention_fwd(q, k, v, causal, scale, bias,
                             window=eff_window)


def flash_attention_bwd_only(do, q, k, v, o, lse, causal=True,
                             softmax_scale=None, bias=None):
    """Without bias: (dq, dk, dv); with bias also dbias (batch-summed,
    fp32 — the native kernel accumulates it atomically in the dK phase)."""
    scale = softmax_scale if softmax_scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if use_native(q) and q.dtype == torch.bfloat16 \
            and q.shape[-1] in (64, 128) \
            and (bias is None or bias.shape[-1] % 4 == 0):
        bias_n = None if bias is None else bias.to(torch.bfloat16).contiguous()
        out = get_ext(False).flash_attn_bwd(do.contiguous(), q, k, v, o, lse,
                                            causal, scale, bias_n)
        return tuple(out)
    return ref.attention_bwd(do, q, k, v, o, lse, causal, scale, bias)


def _gg_tiles(counts, device):
    """Host-built [n_tiles, 3] int32 (expert, row0, rows) + row_off [E+1]."""
    tiles = []
    off = 0
    row_off = [0]
    for e, m in enumerate(counts):
        m = int(m)
        r = 0
        while r < m:
            tiles.append((e, off + r, min(128, m - r)))
            r += 128
        off += m
        row_off.append(off)
    td = torch.tensor(tiles if tiles else [(0, 0, 0)], dtype=torch.int32)
    ro = torch.tensor(row_off, dtype=torch.int32)
    return td.to(device, non_blocking=True), ro.to(device, non_blocking=True)


class _GroupedGemm(torch.autograd.Function):
    """C_e = A_e @ W[e] over expert-sorted rows (ops/csrc/grouped_gemm.hip).

    Replaces the reference's external grouped_gemm CUDA dep
    (moe/grouped_gemm_util.py). counts is a host list (the MoE dispatcher
    already synchronized it for the all-to-all splits).
    """

    @staticmethod
    def forward(ctx, a, w, counts):
        tiles, row_off = _gg_tiles(counts, a.device)
        c = get_ext(False).grouped_gemm(a, w, tiles, w.shape[2], False)
        ctx.save_for_backward(a, w, tiles, row_off)
        return c

    @staticmethod
    def backward(ctx, dc):
        a, w, tiles, row_off = ctx.saved_tensors
        dc = dc.contiguous()
        ext = get_ext(False)
        # dA = dC @ W^T ; dW = A^T dC
        da = ext.grouped_gemm(dc, w, tiles, w.shape[1], True)
        dw = ext.grouped_gemm_dw(a, dc, row_off, w.shape[0])
        return da, dw.to(w.dtype), None


def grouped_gemm(a: torch.Tensor, w: torch.Tensor, counts) -> torch.Tensor:
    """a [M, K] bf16 expert-sorted; w [E, K, N]; counts: per-expert rows."""
    return _GroupedGemm.apply(a.contiguous(), w, counts)


def grouped_gemm_available(a, w) -> bool:
    """fwd needs K%32, N%128; backward dW needs K%128 and N%128; dA needs
    the reverse — so both dims of every W must be multiples of 128."""
    return (a.is_cuda and a.dtype == torch.bfloat16
            and w.shape[1] % 128 == 0 and w.shape[2] % 128 == 0
            and use_native(a))


class _MoEPermute(torch.autograd.Function):
    """Expert-sorted row gather (ops/csrc/moe_permute.hip); backward
    scatter-adds repeated source rows through an fp32 accumulator."""

    @staticmethod
    def forward(ctx, x, rows):
        ctx.save_for_backward(rows)
        ctx.n = x.shape[0]
        return get_ext(False).moe_permute(x, rows)

    @staticmethod
    def backward(ctx, dy):
        (rows,) = ctx.saved_tensors
        return get_ext(False).moe_permute_bwd(dy.contiguous(), rows,
                                              ctx.n), None


def moe_permute(x: torch.Tensor, rows: torch.Tensor) -> torch.Tensor:
    if x.is_cuda and native_available() and x.shape[-1] % 8 == 0 \
            and x.dtype in (torch.bfloat16, torch.float32):
        return _MoEPermute.apply(x.contiguous(), rows.contiguous())
    return x[rows]


class _MoEUnpermute(torch.autograd.Function):
    """Probability-weighted top-k merge back to token order."""

    @staticmethod
    def forward(ctx, back, probs, order, n, k):
        inv = torch.argsort(order)
        t_of = torch.div(order, k, rounding_mode="floor")
        ctx.save_for_backward(back, probs, t_of)
        return get_ext(False).moe_unpermute(back, probs, inv, n, k)

    @staticmethod
    def backward(ctx, dout):
        back, probs, t_of = ctx.saved_tensors
        dback, dprobs = get_ext(False).moe_unpermute_bwd(
            dout.contiguous(), back, probs, t_of)
        return dback, dprobs.to(probs.dtype), None, None, None


def moe_unpermute(back: torch.Tensor, probs: torch.Tensor,
                  order: torch.Tensor, n: int, k: int) -> torch.Tensor:
    """back [m,h] expert-sorted outputs; probs [m] sorted routing weights;
    order: argsort of the flat (token*k) expert assignment."""
    if back.is_cuda and native_available() and back.shape[-1] % 8 == 0 \
            and back.dtype in (torch.bfloat16, torch.float32):
        return _MoEUnpermute.apply(back.contiguous(), probs.contiguous(),
                                   order.contiguous(), n, k)
    h = back.shape[-1]
    out = back.new_zeros(n * k, h)
    out[order] = back * probs.unsqueeze(-1).to(back.dtype)
    return out.reshape(n, k, h).sum(1)


@torch.no_grad()
def decode_attention(q: torch.Tensor, k_cache: torch.Tensor,
                     v_cache: torch.Tensor, cur_len: int,
                     softmax_scale=None, window=None) -> torch.Tensor:
    """Single-token decode attention against a KV cache (serving path).

    q: [b, hq, d]; k_cache/v_cache: [b, max_s, hkv, d]; attends to
    positions [0, cur_len), or the last `window` of them (mistral
    sliding-window decode).  Native CDNA4 kernel on GPU bf16
    (memory-bound KV streaming, decode_attn_kernel in elementwise.hip);
    plain torch reference elsewhere.  Reference role: the optional
    flash-decode path (nvidia_chunked_flash_attn, attention.py:398-514).
    """
    scale = softmax_scale if softmax_scale is not None \
        else q.shape[-1] ** -0.5
    start = max(0, cur_len - window) if window else 0
    if q.is_cuda and native_available() and q.dtype == torch.bfloat16 \
            and q.shape[-1] in (64, 128):
        return get_ext().decode_attn(q.contiguous(), k_cache, v_cache,
                                     int(cur_len), float(scale), int(start))
    b, hq, d = q.shape
    hkv = k_cache.shape[2]
    k = k_cache[:, start:cur_len].float()
    v = v_cache[:, start:cur_len].float()
    if hq != hkv:
        rep = hq // hkv
        k = k.repeat_interleave(rep, dim=2)
        v = v.repeat_interleave(rep, dim=2)
    att = torch.einsum("bhd,bshd->bhs", q.float(), k) * scale
    w = att.softmax(-1)
    o = torch.einsum("bhs,bshd->bhd", w, v)
    return o.to(q.dtype)
