"""Model-profiling entry (reference: galvatron/models/gpt/profiler.py:7).

  python -m hetu_galvatron_amd.cli.profile_model [cfg.yaml] \\
      model.model_name=llama-3-8b profile.profile_type=computation
Runs the memory+computation sweeps (subprocess torchrun) and writes the
parsed model profile the search engine consumes.
"""
from __future__ import annotations


def main(argv=None):
    from ..config.loader import config_from_cli
    from ..profiler.model import ModelProfiler, T5ModelProfiler

    cfg = config_from_cli(argv)
    mp = T5ModelProfiler(cfg) if cfg.model.model_type == "t5" \
        else ModelProfiler(cfg)
    if cfg.profile.profile_type in ("computation", "all"):
        mp.launch_computation_profiling()
    if cfg.profile.profile_type in ("memory", "all"):
        import torch
        nproc = max(min(torch.cuda.device_count(), 8), 1)
        mp.launch_memory_profiling(nproc=nproc)
        mp.process_memory_data()
    print("[profile_model] done")


if __name__ == "__main__":
    main()
