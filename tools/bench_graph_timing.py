"""Launch-overhead measurement: eager vs hipGraph-captured decode on a
tiny 2-layer model (isolates per-token host/launch cost; kernel time is
negligible at this size).  Second graphed call reuses the capture."""
import os, sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from hetu_galvatron_amd.config import load_config
from hetu_galvatron_amd.runtime import GalvatronModel
from hetu_galvatron_amd.runtime.inference import GalvatronGenerator
cfg = load_config(base={"model": {"model_name": "tiny-llama"},
                        "train": {"global_train_batch_size": 1, "train_iters": 1}})
torch.manual_seed(0)
model = GalvatronModel(cfg)
gen = GalvatronGenerator(model, max_batch=1, max_seq=512)
ids = torch.randint(0, cfg.model.vocab_size, (1, 8), device="cuda")
gen.generate(ids, max_new_tokens=8)  # warm
torch.cuda.synchronize(); t0 = time.perf_counter()
gen.generate(ids, max_new_tokens=128)
torch.cuda.synchronize(); eager = (time.perf_counter() - t0) / 128 * 1e3
gen.generate_graphed(ids, max_new_tokens=16, warmup_steps=3)  # capture
torch.cuda.synchronize(); t0 = time.perf_counter()
out = gen.generate_graphed(ids, max_new_tokens=128, warmup_steps=3)
torch.cuda.synchronize(); graphed = (time.perf_counter() - t0) / 128 * 1e3
print(f"TINY-LLAMA 2-layer decode: eager {eager:.3f} ms/tok, "
      f"graphed-steady {graphed:.3f} ms/tok ({eager/graphed:.1f}x)")
