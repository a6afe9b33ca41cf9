"""Hardware-profiling entry (reference: profile_hardware/profile_hardware.py:7).

  python -m hetu_galvatron_amd.cli.profile_hardware [cfg.yaml] \\
      profile_hardware.num_gpus_per_node=8 [--scripts-only]
"""
from __future__ import annotations

import sys


def main(argv=None):
    from ..config.loader import config_from_cli
    from ..profiler.hardware import HardwareProfiler

    args = list(sys.argv[1:] if argv is None else argv)
    scripts_only = "--scripts-only" in args
    if scripts_only:
        args.remove("--scripts-only")
    algo_sweep = "--nccl-algo-sweep" in args
    if algo_sweep:
        args.remove("--nccl-algo-sweep")
    cfg = config_from_cli(args)
    hp = HardwareProfiler(cfg)
    paths = hp.generate_scripts()
    print(f"[profile_hardware] wrote {paths}")
    if not scripts_only:
        hp.profile_bandwidth()
        if algo_sweep:
            hp.profile_nccl_algo_sweep()


if __name__ == "__main__":
    main()
