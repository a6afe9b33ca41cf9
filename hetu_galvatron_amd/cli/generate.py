"""Serving entry point: KV-cache autoregressive generation on one GPU.

  python -m hetu_galvatron_amd.cli.generate model.model_name=llama-3-8b \\
      generate.max_new_tokens=64 generate.prompt_ids=1,2,3 \\
      [parallel.load=/path/to/checkpoint_dir]

Reference role: the inference side the reference delegates to its
optional flash-decode attention backend; here a first-class CLI over
runtime/inference.py.  Loads a Galvatron distributed checkpoint or an HF
directory via the checkpoint adapters when `parallel.load` is set,
otherwise serves random-init weights (smoke/benchmark mode).
"""
from __future__ import annotations

import sys

import torch


def main(argv=None):
    from ..config.loader import config_from_cli
    from ..runtime import GalvatronModel
    from ..runtime.inference import GalvatronGenerator

    argv = list(sys.argv[1:] if argv is None else argv)
    # generate.* keys are CLI-local (not part of the schema)
    gen_args = {"max_new_tokens": 32, "temperature": 0.0, "top_k": 0,
                "prompt_ids": "", "batch": 1, "prompt_len": 8,
                "max_seq": 4096, "eos_id": -1, "use_graph": 0}
    rest = []
    for a in argv:
        if a.startswith("generate.") and "=" in a:
            k, v = a[len("generate."):].split("=", 1)
            if k not in gen_args:
                raise SystemExit(f"unknown generate option: {k}")
            gen_args[k] = type(gen_args[k])(v) if k != "prompt_ids" else v
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
        else:  # HF directory
            from ..runtime.checkpoint.hf_adapter import (hf_to_canonical,
                                                         load_hf_checkpoint)
            from ..runtime.checkpoint.state import load_full_state
            state = hf_to_canonical(load_hf_checkpoint(cfg.ckpt.load),
                                    cfg.model)
            load_full_state(model.stage_model, state, cfg.model)
    if gen_args["prompt_ids"]:
        ids = torch.tensor(
            [[int(t) for t in gen_args["prompt_ids"].split(",")]],
            device=device)
    else:
        ids = torch.randint(0, cfg.model.vocab_size,
                            (gen_args["batch"], gen_args["prompt_len"]),
                            device=device)
    gen = GalvatronGenerator(model, max_batch=ids.shape[0],
                             max_seq=gen_args["max_seq"])
    if gen_args["use_graph"]:
        # hipGraph-captured decode step (greedy); falls back to eager off-GPU
        out = gen.generate_graphed(
            ids, max_new_tokens=gen_args["max_new_tokens"])
    else:
        out = gen.generate(
            ids, max_new_tokens=gen_args["max_new_tokens"],
            temperature=gen_args["temperature"], top_k=gen_args["top_k"],
            eos_id=None if gen_args["eos_id"] < 0 else gen_args["eos_id"])
    for row in out.tolist():
        print(" ".join(str(t) for t in row))
    return out


if __name__ == "__main__":
    main()
