"""Hardware-profiler orchestrator: RCCL/xGMI bandwidth sweep launcher.

Reference: galvatron/core/profiler/hardware_profiler.py:39-229 (generates
torchrun scripts for allreduce/p2p/sp/overlap sweeps and writes the
bandwidth JSONs).  Here the worker is a python module
(profiler/hardware_bench.py); this class both WRITES the equivalent shell
scripts (parity artifact) and can run the sweeps directly via subprocess.
"""
from __future__ import annotations

import os
import subprocess
import sys
from typing import List, Optional

from ..config import GalvatronConfig


class HardwareProfiler:
    def __init__(self, cfg: GalvatronConfig):
        self.cfg = cfg
        self.args = cfg.profile_hardware
        self.out_dir = self.args.hardware_config_dir

    def _launch_cmd(self, op: str, extra: Optional[List[str]] = None) -> List[str]:
        a = self.args
        world = a.num_nodes * a.num_gpus_per_node
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--nnodes", str(a.num_nodes),
               "--nproc-per-node", str(a.num_gpus_per_node),
               "--master-addr", a.master_addr,
               "--master-port", str(a.master_port),
               "-m", "hetu_galvatron_amd.profiler.hardware_bench",
               "--op", op, "--output-dir", self.out_dir,
               "--num-nodes", str(a.num_nodes),
               "--warmup-iters", str(a.warmup_iters),
               "--measure-iters", str(a.measure_iters),
               "--start-mb", str(a.start_mb), "--end-mb", str(a.end_mb)]
        return cmd + (extra or [])

    def generate_scripts(self, script_dir: str = "scripts") -> List[str]:
        """Write profile_{op}.sh launcher scripts (reference generate_script)."""
        os.makedirs(script_dir, exist_ok=True)
        paths = []
        for op in ("allreduce", "p2p", "sp_time", "overlap"):
            path = os.path.join(script_dir, f"profile_{op}.sh")
            with open(path, "w") as f:
                f.write("#!/bin/bash\n" + " ".join(self._launch_cmd(op)) + "\n")
            os.chmod(path, 0o755)
            paths.append(path)
        return paths

    def profile_bandwidth(self, ops: Optional[List[str]] = None,
                          env: Optional[dict] = None) -> None:
        """Run the sweeps in-process via subprocess torchrun."""
        for op in ops or ("allreduce", "p2p", "sp_time", "overlap"):
            e = dict(os.environ)
            e.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
            if env:
                e.update(env)
            subprocess.run(self._launch_cmd(op), check=True, env=e)

    def profile_nccl_algo_sweep(self, op: str = "allreduce",
                                algos: Optional[List[str]] = None) -> None:
        """Re-run a sweep once per RCCL algorithm (NCCL_ALGO) — xGMI is
        point-to-point (7 links/GPU), so ring vs tree vs direct behave
        very differently from NVSwitch and the cost model should absorb
        whichever the profiler finds fastest (SURVEY §5 comm-backend
        notes).  Each pass writes into its own subdirectory
        `<out_dir>/algo_<name>/`; compare the JSONs and copy the winner
        up into hardware_configs/."""
        base = self.out_dir
        for algo in algos or ("Ring", "Tree"):
            self.out_dir = os.path.join(base, f"algo_{algo.lower()}")
            os.makedirs(self.out_dir, exist_ok=True)
            try:
                self.profile_bandwidth([op], env={"NCCL_ALGO": algo})
            finally:
                self.out_dir = base
