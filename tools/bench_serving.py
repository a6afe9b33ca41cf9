"""Serving throughput benchmark (run on a GPU box; round-2 measurement).

  python tools/bench_serving.py [--model llama-3-8b] [--slots 8]

Measures continuous-batching decode throughput (tokens/s across active
slots) and per-request latency at mixed prompt lengths.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.runtime import GalvatronModel
from hetu_galvatron_amd.runtime.serving import ContinuousBatchingEngine

ap = argparse.ArgumentParser()
ap.add_argument("--model", default="llama-3-8b")
ap.add_argument("--slots", type=int, default=8)
ap.add_argument("--new-tokens", type=int, default=64)
ap.add_argument("--max-seq", type=int, default=2048)
args = ap.parse_args()

cfg = load_config(base={"model": {"model_name": args.model},
                        "train": {"global_train_batch_size": 1,
                                  "train_iters": 1}})
torch.manual_seed(0)
t0 = time.perf_counter()
model = GalvatronModel(cfg)
print(f"model built in {time.perf_counter() - t0:.1f}s", file=sys.stderr)
eng = ContinuousBatchingEngine(model, max_slots=args.slots,
                               max_seq=args.max_seq)
rng = torch.Generator().manual_seed(1)
for i in range(args.slots):
    p = torch.randint(0, cfg.model.vocab_size, (128 + 64 * i,),
                      generator=rng).to(model.stage_model.blocks[0]
                                        .inner.word_embeddings.weight.device)
    eng.add_request(p, args.new_tokens)
torch.cuda.synchronize()
t0 = time.perf_counter()
total = 0
while eng.n_active:
    total += len(eng.step())
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print({"decode_tokens": total, "seconds": round(dt, 2),
       "tokens_per_s": round(total / dt, 1),
       "slots": args.slots, "new_tokens": args.new_tokens})
