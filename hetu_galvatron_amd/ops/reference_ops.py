"""Plain-PyTorch fp32 reference implementations.

Dual role: (a) CPU execution path for GPU-less test runs, (b) the numerics
baseline every HIP kernel is validated against (tests/ops/*, run with
@pytest.mark.gpu on the MI355X box).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def rmsnorm_fwd(x: torch.Tensor, weight: torch.Tensor, eps: float) -> Tuple[torch.Tensor, torch.Tensor]:
    xf = x.float()
    invrms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    y = (xf * invrms) * weight.float()
    return y.to(x.dtype), invrms.squeeze(-1)


def rmsnorm_bwd(dy: torch.Tensor, x: torch.Tensor, weight: torch.Tensor,
                invrms: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    xf = x.float()
    dyf = dy.float()
    wf = weight.float()
    r = invrms.unsqueeze(-1)
    xhat = xf * r
    dxhat = dyf * wf
    H = x.shape[-1]
    dx = r * (dxhat - xhat * (dxhat * xhat).mean(-1, keepdim=True))
    dw = (dyf * xhat).reshape(-1, H).sum(0)
    return dx.to(x.dtype), dw


def layernorm_fwd(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
                  eps: float) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    xf = x.float()
    mean = xf.mean(-1, keepdim=True)
    var = xf.var(-1, unbiased=False, keepdim=True)
    invstd = torch.rsqrt(var + eps)
    y = (xf - mean) * invstd * weight.float() + bias.float()
    return y.to(x.dtype), mean.squeeze(-1), invstd.squeeze(-1)


def layernorm_bwd(dy, x, weight, mean, invstd):
    xf = x.float()
    dyf = dy.float()
    wf = weight.float()
    m = mean.unsqueeze(-1)
    r = invstd.unsqueeze(-1)
    xhat = (xf - m) * r
    dxhat = dyf * wf
    H = x.shape[-1]
    dx = r * (dxhat - dxhat.mean(-1, keepdim=True)
              - xhat * (dxhat * xhat).mean(-1, keepdim=True))
    dw = (dyf * xhat).reshape(-1, H).sum(0)
    db = dyf.reshape(-1, H).sum(0)
    return dx.to(x.dtype), dw, db


def swiglu_fwd(x: torch.Tensor) -> torch.Tensor:
    """x[..., 2F] = [gate, up] -> silu(gate) * up."""
    gate, up = x.chunk(2, dim=-1)
    gf = gate.float()
    return (F.silu(gf) * up.float()).to(x.dtype)


def swiglu_bwd(dy: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    gate, up = x.chunk(2, dim=-1)
    gf, uf, dyf = gate.float(), up.float(), dy.float()
    sig = torch.sigmoid(gf)
    silu = gf * sig
    dgate = dyf * uf * sig * (1 + gf * (1 - sig))
    dup = dyf * silu
    return torch.cat([dgate, dup], dim=-1).to(x.dtype)


def geglu_fwd(x: torch.Tensor) -> torch.Tensor:
    gate, up = x.chunk(2, dim=-1)
    return (F.gelu(gate.float(), approximate="tanh") * up.float()).to(x.dtype)


def geglu_bwd(dy: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    gate, up = x.chunk(2, dim=-1)
    gf, uf, dyf = gate.float().requires_grad_(True), up.float(), dy.float()
    with torch.enable_grad():
        g = F.gelu(gf, approximate="tanh")
    (dgate,) = torch.autograd.grad(g, gf, dyf * uf)
    dup = dyf * F.gelu(gate.float(), approximate="tanh")
    return torch.cat([dgate, dup], dim=-1).to(x.dtype)


def rope_freqs(seq_len: int, dim: int, theta: float = 10000.0,
               device=None, dtype=torch.float32,
               pos_offset: int = 0,
               interp: float = 1.0) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables [seq, dim/2] (host-precomputed per the CDNA4 elementwise
    guideline: no on-device trig in the hot path).  interp > 1 is linear
    sequence-length interpolation (reference rotary_pos_embedding.py
    rotary_seq_len_interpolation_factor: positions divided by the factor)."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, dim, 2, device=device).float() / dim))
    t = torch.arange(pos_offset, pos_offset + seq_len, device=device).float()
    if interp and interp != 1.0:
        t = t / interp
    freqs = torch.outer(t, inv_freq)
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def rope_apply(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               conj: bool = False) -> torch.Tensor:
    """Apply rotary embedding, NEOX (non-interleaved half-rotation) style.

    x: [s, b, h, d]; cos/sin: [s, d/2].  conj=True applies the inverse
    rotation (the backward pass).
    """
    d2 = x.shape[-1] // 2
    x1 = x[..., :d2].float()
    x2 = x[..., d2:].float()
    c = cos[:, None, None, :].float()
    s = sin[:, None, None, :].float()
    if conj:
        s = -s
    y1 = x1 * c - x2 * s
    y2 = x2 * c + x1 * s
    return torch.cat([y1, y2], dim=-1).to(x.dtype)


def rope_apply_interleaved(x: torch.Tensor, cos: torch.Tensor,
                           sin: torch.Tensor) -> torch.Tensor:
    """Interleaved (GPT-J style) rotary embedding: channel pairs
    (2i, 2i+1) rotate together (reference rotary_interleaved arg).
    x: [s, b, h, d]; cos/sin: [s, d/2]."""
    x1 = x[..., 0::2].float()
    x2 = x[..., 1::2].float()
    c = cos[:, None, None, :].float()
    s_ = sin[:, None, None, :].float()
    y = torch.empty_like(x, dtype=torch.float32)
    y[..., 0::2] = x1 * c - x2 * s_
    y[..., 1::2] = x2 * c + x1 * s_
    return y.to(x.dtype)


def rope_apply_neox_batched(x: torch.Tensor, cos: torch.Tensor,
                            sin: torch.Tensor) -> torch.Tensor:
    """NEOX half-rotation with PER-BATCH tables (multimodal RoPE).

    x: [s, b, h, d]; cos/sin: [s, b, d/2].
    """
    d2 = x.shape[-1] // 2
    x1 = x[..., :d2].float()
    x2 = x[..., d2:].float()
    c = cos[:, :, None, :].float()
    s = sin[:, :, None, :].float()
    y1 = x1 * c - x2 * s
    y2 = x2 * c + x1 * s
    return torch.cat([y1, y2], dim=-1).to(x.dtype)


def attention_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  causal: bool = True, softmax_scale: Optional[float] = None,
                  bias: Optional[torch.Tensor] = None,
                  window: Optional[int] = None,
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Reference attention WITH log-sum-exp output (needed for ring-CP merges).

    q: [b, s, hq, d]; k,v: [b, s, hkv, d] (GQA: hq % hkv == 0).
    bias: optional [hq, sq, skv] additive scores bias (t5 relative bias).
    Returns o [b, s, hq, d], lse [b, hq, s] (natural log).
    """
    b, sq, hq, d = q.shape
    skv = k.shape[1]
    hkv = k.shape[2]
    scale = softmax_scale if softmax_scale is not None else 1.0 / math.sqrt(d)
    rep = hq // hkv
    kx = k.repeat_interleave(rep, dim=2) if rep > 1 else k
    vx = v.repeat_interleave(rep, dim=2) if rep > 1 else v
    qf = q.permute(0, 2, 1, 3).float()   # [b,h,s,d]
    kf = kx.permute(0, 2, 1, 3).float()
    vf = vx.permute(0, 2, 1, 3).float()
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale  # [b,h,sq,skv]
    if bias is not None:
        scores = scores + bias.unsqueeze(0).float()
    if causal:
        # causal with bottom-right alignment when sq != skv
        i = torch.arange(sq, device=q.device)[:, None]
        j = torch.arange(skv, device=q.device)[None, :]
        mask = j > (i + (skv - sq))
        if window is not None:
            # mistral sliding window: key positions below the window edge
            mask = mask | (j <= (i + (skv - sq) - window))
        scores = scores.masked_fill(mask, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)  # [b,h,sq]
    p = torch.exp(scores - lse.unsqueeze(-1))
    p = torch.nan_to_num(p)  # fully-masked rows
    o = torch.matmul(p, vf)  # [b,h,sq,d]
    return o.permute(0, 2, 1, 3).to(q.dtype), lse


def attention_bwd(do, q, k, v, o, lse, causal=True, softmax_scale=None,
                  bias=None, window=None):
    """Reference backward, flash-style against the PASSED o/lse.

    Using the caller's (global) lse/o matters for ring-CP: each per-block
    backward must normalize with the GLOBAL softmax so the block gradients
    are true partials that sum across ring steps (the native kernel does
    exactly this; an autograd recompute would silently use block-LOCAL
    softmax).  Falls back to autograd recompute when o/lse is None.
    With a bias, also returns dbias [hq, sq, skv] (summed over batch).
    """
    if o is None or lse is None:
        q32 = q.detach().float().requires_grad_(True)
        k32 = k.detach().float().requires_grad_(True)
        v32 = v.detach().float().requires_grad_(True)
        b32 = bias.detach().float().requires_grad_(True) \
            if bias is not None else None
        with torch.enable_grad():
            o2, _ = attention_fwd(q32, k32, v32, causal, softmax_scale, b32,
                                  window)
        if b32 is None:
            gq, gk, gv = torch.autograd.grad(o2, (q32, k32, v32), do.float())
            return gq.to(q.dtype), gk.to(k.dtype), gv.to(v.dtype)
        gq, gk, gv, gb = torch.autograd.grad(o2, (q32, k32, v32, b32),
                                             do.float())
        return gq.to(q.dtype), gk.to(k.dtype), gv.to(v.dtype), gb

    b, sq, hq, d = q.shape
    skv, hkv = k.shape[1], k.shape[2]
    scale = softmax_scale if softmax_scale is not None else 1.0 / math.sqrt(d)
    rep = hq // hkv
    kx = k.repeat_interleave(rep, dim=2) if rep > 1 else k
    vx = v.repeat_interleave(rep, dim=2) if rep > 1 else v
    qf = q.permute(0, 2, 1, 3).float()
    kf = kx.permute(0, 2, 1, 3).float()
    vf = vx.permute(0, 2, 1, 3).float()
    dof = do.permute(0, 2, 1, 3).float()
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale  # [b,h,sq,skv]
    if bias is not None:
        s = s + bias.unsqueeze(0).float()
    if causal:
        i = torch.arange(sq, device=q.device)[:, None]
        j = torch.arange(skv, device=q.device)[None, :]
        mask = j > (i + (skv - sq))
        if window is not None:
            mask = mask | (j <= (i + (skv - sq) - window))
        s = s.masked_fill(mask, float("-inf"))
    p = torch.exp(s - lse.float().unsqueeze(-1))
    p = torch.nan_to_num(p)
    dv_h = torch.matmul(p.transpose(-1, -2), dof)       # [b,h,skv,d]
    dp = torch.matmul(dof, vf.transpose(-1, -2))        # [b,h,sq,skv]
    di = (do.float() * o.float()).sum(-1).permute(0, 2, 1)  # [b,h,sq]
    dsr = p * (dp - di.unsqueeze(-1))                   # dL/ds_raw
    dq = torch.matmul(dsr, kf) * scale
    dk_h = torch.matmul(dsr.transpose(-1, -2), qf) * scale
    if rep > 1:
        dk_h = dk_h.view(b, hkv, rep, skv, d).sum(2)
        dv_h = dv_h.view(b, hkv, rep, skv, d).sum(2)
    gq = dq.permute(0, 2, 1, 3).to(q.dtype)
    gk = dk_h.permute(0, 2, 1, 3).to(k.dtype)
    gv = dv_h.permute(0, 2, 1, 3).to(v.dtype)
    if bias is None:
        return gq, gk, gv
    return gq, gk, gv, dsr.sum(0)


def vocab_ce_stats(logits: torch.Tensor, target: torch.Tensor,
                   vocab_start: int, vocab_end: int):
    """Local-shard stats for vocab-parallel cross entropy.

    logits: [n, v_local] (any float dtype, reduced in fp32);
    target: [n] GLOBAL vocab ids.
    Returns (local_max[n], sumexp_given_gmax needs gmax) - split into 2 calls.
    """
    return logits.float().max(dim=-1).values


def vocab_ce_fwd_local(logits, target, gmax, vocab_start, vocab_end):
    lf = logits.float()
    sumexp = torch.exp(lf - gmax.unsqueeze(-1)).sum(-1)
    in_shard = (target >= vocab_start) & (target < vocab_end)
    t_local = (target - vocab_start).clamp(0, logits.shape[-1] - 1)
    tlogit = lf.gather(-1, t_local.unsqueeze(-1)).squeeze(-1)
    tlogit = torch.where(in_shard, tlogit, torch.zeros_like(tlogit))
    return sumexp, tlogit


def vocab_ce_bwd_local(logits, target, gmax, gsumexp, grad_out,
                       vocab_start, vocab_end):
    lf = logits.float()
    p = torch.exp(lf - gmax.unsqueeze(-1)) / gsumexp.unsqueeze(-1)
    in_shard = (target >= vocab_start) & (target < vocab_end)
    t_local = (target - vocab_start).clamp(0, logits.shape[-1] - 1)
    onehot = torch.zeros_like(p)
    onehot.scatter_(-1, t_local.unsqueeze(-1), in_shard.float().unsqueeze(-1))
    return ((p - onehot) * grad_out.unsqueeze(-1)).to(logits.dtype)


def adamw_step(params, grads, exp_avgs, exp_avg_sqs, masters, step: int,
               lr: float, beta1: float, beta2: float, eps: float, wd: float):
    """Multi-tensor AdamW reference: fp32 master update + bf16 param copy."""
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    for p, g, m, v, mp in zip(params, grads, exp_avgs, exp_avg_sqs, masters):
        gf = g.float()
        m.mul_(beta1).add_(gf, alpha=1 - beta1)
        v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
        denom = (v / bc2).sqrt().add_(eps)
        upd = (m / bc1) / denom + wd * mp
        mp.add_(upd, alpha=-lr)
        p.copy_(mp.to(p.dtype))
