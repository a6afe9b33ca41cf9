"""Serving-path microbenchmark: decode_attn kernel bandwidth + end-to-end
KV-cache generation latency on Llama-3-8B (random init, bf16).

Run on a GPU box:  python tools/bench_decode.py
Writes gpurun_out/decode_bench.json (commit a copy under profiles/).
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.ops._ext import get_ext
from hetu_galvatron_amd.runtime import GalvatronModel
from hetu_galvatron_amd.runtime.inference import GalvatronGenerator

results = {}

# ---- kernel: decode_attn bandwidth (KV streaming bound) -------------------
ext = get_ext()
for b, hq, hkv, d, ctx_len in [(1, 32, 8, 128, 4096), (1, 32, 8, 128, 32768),
                               (8, 32, 8, 128, 4096), (32, 32, 8, 128, 4096)]:
    q = torch.randn(b, hq, d, device="cuda", dtype=torch.bfloat16)
    kc = torch.randn(b, ctx_len, hkv, d, device="cuda", dtype=torch.bfloat16)
    vc = torch.randn(b, ctx_len, hkv, d, device="cuda", dtype=torch.bfloat16)
    for _ in range(5):
        ext.decode_attn(q, kc, vc, ctx_len, d ** -0.5)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 50
    for _ in range(iters):
        ext.decode_attn(q, kc, vc, ctx_len, d ** -0.5)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    bytes_moved = 2 * b * ctx_len * hkv * d * 2  # K + V bf16
    results[f"decode_attn_b{b}_ctx{ctx_len}"] = {
        "us": round(us, 1), "gb_per_s": round(bytes_moved / us / 1e3, 1)}
    print(results[f"decode_attn_b{b}_ctx{ctx_len}"])

if "--kernels-only" in sys.argv:
    os.makedirs("gpurun_out", exist_ok=True)
    json.dump(results, open("gpurun_out/decode_bench.json", "w"), indent=1)
    print("WROTE gpurun_out/decode_bench.json (kernels only)")
    sys.exit(0)

# ---- end-to-end: Llama-3-8B generation ------------------------------------
cfg = load_config(base={
    "model": {"model_name": "llama-3-8b"},
    "train": {"global_train_batch_size": 8, "train_iters": 1},
})
torch.manual_seed(0)
t0 = time.perf_counter()
model = GalvatronModel(cfg)
print(f"model built in {time.perf_counter() - t0:.1f}s")

for batch in (1, 8):
    gen = GalvatronGenerator(model, max_batch=batch, max_seq=4096)
    ids = torch.randint(0, cfg.model.vocab_size, (batch, 512), device="cuda")
    out = gen.generate(ids, max_new_tokens=8, temperature=0.0)  # warmup
    torch.cuda.synchronize()
    n_new = 64
    t0 = time.perf_counter()
    out = gen.generate(ids, max_new_tokens=n_new, temperature=0.0)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert out.shape == (batch, 512 + n_new)
    results[f"generate_8b_b{batch}"] = {
        "prompt": 512, "new_tokens": n_new,
        "ms_per_token": round(dt / n_new * 1e3, 2),
        "decode_tokens_per_s": round(batch * n_new / dt, 1)}
    print(results[f"generate_8b_b{batch}"])
    # hipGraph-captured decode step (one replay per token)
    t0 = time.perf_counter()
    outg = gen.generate_graphed(ids, max_new_tokens=n_new, warmup_steps=3)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert outg.shape == (batch, 512 + n_new)
    results[f"generate_8b_b{batch}_hipgraph"] = {
        "prompt": 512, "new_tokens": n_new,
        "ms_per_token": round(dt / n_new * 1e3, 2),
        "decode_tokens_per_s": round(batch * n_new / dt, 1)}
    print(results[f"generate_8b_b{batch}_hipgraph"])

os.makedirs("gpurun_out", exist_ok=True)
json.dump(results, open("gpurun_out/decode_bench.json", "w"), indent=1)
print("WROTE gpurun_out/decode_bench.json")
