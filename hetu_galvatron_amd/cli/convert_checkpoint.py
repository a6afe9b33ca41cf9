"""HF <-> canonical checkpoint converters (reference: galvatron/tools/
checkpoint_convert_h2g.py / _g2h.py).

  python -m hetu_galvatron_amd.cli.convert_checkpoint h2g \
      --hf-dir /path/llama3 --out canonical.pt model.model_name=llama-3-8b
  python -m hetu_galvatron_amd.cli.convert_checkpoint g2h \
      --canonical canonical.pt --out-dir /path/out ...
"""
from __future__ import annotations

import argparse
import sys

import torch


def main(argv=None):
    from ..config.loader import load_config
    from ..runtime.checkpoint.hf_adapter import (
        canonical_to_hf_llama, canonical_to_hf_mixtral, canonical_to_hf_t5,
        hf_to_canonical, load_hf_checkpoint, save_hf_checkpoint)

    args = list(sys.argv[1:] if argv is None else argv)
    mode = args.pop(0)
    ap = argparse.ArgumentParser()
    ap.add_argument("--hf-dir")
    ap.add_argument("--canonical")
    ap.add_argument("--out")
    ap.add_argument("--out-dir")
    ns, overrides = ap.parse_known_args(args)
    cfg = load_config(overrides=overrides)
    if mode == "h2g":
        hf = load_hf_checkpoint(ns.hf_dir)
        can = hf_to_canonical(hf, cfg.model)
        torch.save(can, ns.out)
        print(f"wrote canonical checkpoint: {ns.out} ({len(can)} tensors)")
    elif mode == "g2h":
        can = torch.load(ns.canonical, map_location="cpu", weights_only=True)
        m = cfg.model
        if m.model_type == "t5":
            hf = canonical_to_hf_t5(can, m)
        elif m.model_type.startswith("moe") and m.num_experts > 0:
            hf = canonical_to_hf_mixtral(can, m)
        else:
            hf = canonical_to_hf_llama(can, m)
        save_hf_checkpoint(hf, ns.out_dir)
        print(f"wrote HF checkpoint dir: {ns.out_dir}")
    else:
        raise SystemExit(f"unknown mode {mode} (h2g|g2h)")


if __name__ == "__main__":
    main()
