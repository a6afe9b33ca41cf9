#!/usr/bin/env python3
"""Standalone per-kernel microbenchmarks (GPU box).

python tools/bench_kernels.py [flash|norms|elementwise|adam|all] [--iters N]

Prints per-kernel wall time + achieved TF/s (compute) or GB/s (memory).
Used with rocprofv3 PMC runs to drive the guide's diagnostic loop.
"""
import argparse
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def ext():
    from hetu_galvatron_amd.ops._ext import get_ext
    return get_ext(False)


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_flash(iters):
    e = ext()
    # bench-shape slice: one microbatch of the llama-3-8b bench
    b, s, hq, hkv, d = 1, 4096, 32, 8, 128
    q = torch.randn(b, s, hq, d, device="cuda").bfloat16()
    k = torch.randn(b, s, hkv, d, device="cuda").bfloat16()
    v = torch.randn(b, s, hkv, d, device="cuda").bfloat16()
    scale = 1.0 / math.sqrt(d)
    t = timeit(lambda: e.flash_attn_fwd(q, k, v, True, scale), iters)
    flops = 4 * b * hq * (s * s / 2) * d  # QK^T + PV, causal
    print(f"flash_fwd  b{b} s{s} hq{hq} d{d}: {t*1e3:8.2f} ms  "
          f"{flops/t/1e12:7.1f} TF/s")
    o, lse = e.flash_attn_fwd(q, k, v, True, scale)
    do = torch.randn_like(o)
    t = timeit(lambda: e.flash_attn_bwd(do, q, k, v, o, lse, True, scale),
               iters)
    print(f"flash_bwd  b{b} s{s} hq{hq} d{d}: {t*1e3:8.2f} ms  "
          f"{2.5*flops/t/1e12:7.1f} TF/s")


def bench_norms(iters):
    e = ext()
    n, H = 4096, 4096
    x = torch.randn(n, H, device="cuda").bfloat16()
    w = torch.randn(H, device="cuda").bfloat16()
    t = timeit(lambda: e.rmsnorm_fwd(x, w, 1e-5), iters)
    gb = 2 * n * H * 2 / 1e9
    print(f"rmsnorm_fwd [{n}x{H}]: {t*1e6:8.1f} us  {gb/t:7.0f} GB/s")
    y, inv = e.rmsnorm_fwd(x, w, 1e-5)
    dy = torch.randn_like(x)
    t = timeit(lambda: e.rmsnorm_bwd(dy, x, w, inv), iters)
    gb = 3 * n * H * 2 / 1e9
    print(f"rmsnorm_bwd [{n}x{H}]: {t*1e6:8.1f} us  {gb/t:7.0f} GB/s")


def bench_elementwise(iters):
    e = ext()
    n, F = 4096, 14336
    x = torch.randn(n, 2 * F, device="cuda").bfloat16()
    t = timeit(lambda: e.swiglu_fwd(x), iters)
    gb = 3 * n * F * 2 / 1e9
    print(f"swiglu_fwd [{n}x2x{F}]: {t*1e6:8.1f} us  {gb/t:7.0f} GB/s")
    s, bb, h, d = 4096, 1, 32, 128
    xr = torch.randn(s, bb, h, d, device="cuda").bfloat16()
    cos = torch.randn(s, d // 2, device="cuda").float()
    sin = torch.randn(s, d // 2, device="cuda").float()
    t = timeit(lambda: e.rope_fwd(xr, cos, sin, False), iters)
    gb = 2 * s * bb * h * d * 2 / 1e9
    print(f"rope_fwd [{s}x{bb}x{h}x{d}]: {t*1e6:8.1f} us  {gb/t:7.0f} GB/s")


def bench_adam(iters):
    e = ext()
    n = 500_000_000
    master = torch.randn(n, device="cuda").float()
    g = torch.randn(n, device="cuda").bfloat16()
    m = torch.zeros(n, device="cuda").float()
    v = torch.zeros(n, device="cuda").float()
    out = torch.zeros(n, device="cuda").bfloat16()
    t = timeit(lambda: e.fused_adamw([master], [g], [m], [v], [out], 2,
                                     1e-4, 0.9, 0.95, 1e-8, 0.01), iters)
    gb = n * (4 * 3 * 2 + 2 + 2) / 1e9  # m/v/master rw, g read, out write
    print(f"fused_adamw [{n/1e6:.0f}M]: {t*1e3:8.2f} ms  {gb/t:7.0f} GB/s")
    flat = torch.zeros(n, device="cuda").float()
    t = timeit(lambda: e.grad_accum(flat, g, 0), iters)
    gb = n * (4 * 2 + 2) / 1e9
    print(f"grad_accum [{n/1e6:.0f}M]: {t*1e3:8.2f} ms  {gb/t:7.0f} GB/s")


def bench_bias(iters):
    """t5-3b attention layer shapes: native biased flash vs the eager fp32
    path it replaces (VERDICT r1 #5 done-criterion: >=5x)."""
    import torch.autograd as ag
    from hetu_galvatron_amd.ops import flash_bias_attention
    from hetu_galvatron_amd.runtime.transformer.attention_impl import (
        eager_bias_attention)
    b, s, h, d = 4, 512, 32, 128
    q = torch.randn(b, s, h, d, device="cuda", requires_grad=True).bfloat16()
    k = torch.randn(b, s, h, d, device="cuda").bfloat16()
    v = torch.randn(b, s, h, d, device="cuda").bfloat16()
    bias = torch.randn(h, s, s, device="cuda").bfloat16().requires_grad_(True)
    scale = 1.0

    def run(fn):
        o = fn()
        o.float().sum().backward()

    qn = q.detach().requires_grad_(True)
    t_native = timeit(lambda: run(lambda: flash_bias_attention(
        qn, k, v, bias, True, scale)), iters)
    t_eager = timeit(lambda: run(lambda: eager_bias_attention(
        qn, k, v, bias, True, scale)), iters)
    print(f"bias_attn t5-3b layer b{b} s{s} h{h} d{d} fwd+bwd: "
          f"native {t_native*1e3:7.2f} ms  eager {t_eager*1e3:7.2f} ms  "
          f"speedup {t_eager/t_native:5.1f}x")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("what", nargs="?", default="all")
    ap.add_argument("--iters", type=int, default=10)
    args = ap.parse_args()
    torch.cuda.set_device(0)
    fns = {"flash": bench_flash, "norms": bench_norms,
           "elementwise": bench_elementwise, "adam": bench_adam,
           "bias": bench_bias}
    if args.what == "all":
        for f in fns.values():
            f(args.iters)
    else:
        fns[args.what](args.iters)


if __name__ == "__main__":
    main()
