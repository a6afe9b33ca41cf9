"""HTTP serving entry: FastAPI over the continuous-batching engine.

  python -m hetu_galvatron_amd.cli.serve model.model_name=llama-3-8b \\
      ckpt.load=/path/to/ckpt serve.port=8000 serve.max_slots=8

POST /generate  {"prompt_ids": [..], "max_new_tokens": 64}
    -> {"request_id": r, "tokens": [...]}   (blocks until complete)
GET  /health    -> {"status": "ok", "active": n, "free_slots": k}

A background decode loop drains the slot pool continuously; concurrent
requests share decode steps (continuous batching).  Token-id interface —
pair with `runtime/datasets/tokenizer.py` wrappers client-side when a
tokenizer file is available (no network here).
"""
from __future__ import annotations

import sys
import threading

import torch


def build_app(engine, lock: threading.Lock):
    from fastapi import Body, FastAPI, HTTPException

    app = FastAPI(title="hetu_galvatron_amd serving")

    done = {}          # rid -> threading.Event

    def decode_loop():
        while True:
            with lock:
                if engine.n_active:
                    engine.step()
                    for rid, ev in list(done.items()):
                        if rid not in engine.slot_of:
                            ev.set()
            threading.Event().wait(0.001)

    threading.Thread(target=decode_loop, daemon=True).start()

    @app.get("/health")
    def health():
        return {"status": "ok", "active": engine.n_active,
                "free_slots": len(engine.free)}

    @app.post("/generate")
    def generate(payload: dict = Body(...)):
        ids = torch.tensor(payload["prompt_ids"], dtype=torch.long,
                           device=engine.gen._dev)
        max_new = int(payload.get("max_new_tokens", 32))
        temperature = float(payload.get("temperature", 0.0))
        seed = int(payload.get("seed", 0))
        with lock:
            if not engine.free:
                raise HTTPException(503, "no free slots")
            eos = payload.get("eos_id")
            rid = engine.add_request(ids, max_new, temperature=temperature,
                                     seed=seed,
                                     eos_id=int(eos) if eos is not None
                                     else None)
            ev = threading.Event()
            if rid not in engine.slot_of:   # finished at prefill
                ev.set()
            else:
                done[rid] = ev
        ev.wait()
        done.pop(rid, None)
        return {"request_id": rid, "tokens": engine.collect(rid)}

    return app


def main(argv=None):
    from ..config.loader import config_from_cli
    from ..runtime import GalvatronModel
    from ..runtime.serving import ContinuousBatchingEngine

    argv = list(sys.argv[1:] if argv is None else argv)
    opts = {"port": 8000, "host": "0.0.0.0", "max_slots": 8,
            "max_seq": 4096}
    rest = []
    for a in argv:
        if a.startswith("serve.") and "=" in a:
            k, v = a[len("serve."):].split("=", 1)
            opts[k] = type(opts[k])(v)
        else:
            rest.append(a)
    cfg = config_from_cli(rest)
    device = torch.device("cuda", 0) if torch.cuda.is_available() \
        else torch.device("cpu")
    torch.manual_seed(cfg.train.seed)
    model = GalvatronModel(cfg, device=device)
    if cfg.ckpt.load:
        import os
        if os.path.exists(os.path.join(cfg.ckpt.load,
                                       "latest_checkpointed_iteration.txt")):
            from ..runtime.checkpoint.distributed import (
                load_distributed_checkpoint)
            load_distributed_checkpoint(model, None, None, cfg)
        else:
            from ..runtime.checkpoint.hf_adapter import (hf_to_canonical,
                                                         load_hf_checkpoint)
            from ..runtime.checkpoint.state import load_full_state
            state = hf_to_canonical(load_hf_checkpoint(cfg.ckpt.load),
                                    cfg.model)
            load_full_state(model.stage_model, state, cfg.model)
    engine = ContinuousBatchingEngine(model, max_slots=opts["max_slots"],
                                      max_seq=opts["max_seq"])
    app = build_app(engine, threading.Lock())
    import uvicorn
    uvicorn.run(app, host=opts["host"], port=opts["port"])


if __name__ == "__main__":
    main()
