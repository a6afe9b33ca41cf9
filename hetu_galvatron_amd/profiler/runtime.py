"""In-training runtime profiler: iteration timing + peak-memory snapshots.

Reference: galvatron/core/profiler/runtime_profiler.py:24-370 (memory
snapshots at Before-Fwd / After-Fwd / After-Bwd / After-step, CUDA-event
iteration timing, save_profiled_memory / computation JSON writers).
hipEventRecord works unchanged through torch.cuda on ROCm.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


def _mb(x: int) -> float:
    return x / (1024.0 * 1024.0)


class RuntimeProfiler:
    def __init__(self, enabled: bool = True, device=None, rank: int = 0,
                 warmup_iters: int = 2):
        self.enabled = enabled and torch.cuda.is_available()
        self.device = device
        self.rank = rank
        self.warmup = warmup_iters
        self.iteration = 0
        self.mem_snaps: Dict[str, Dict[str, float]] = {}
        self.iter_times_ms: List[float] = []
        self.fwd_times_ms: List[float] = []
        self._ev_start: Optional[torch.cuda.Event] = None
        self._ev_end: Optional[torch.cuda.Event] = None
        self._fwd_pairs: List = []

    # -- memory ------------------------------------------------------------
    def snap(self, tag: str) -> None:
        """Record allocated/reserved/peak at a named point of the step."""
        if not self.enabled:
            return
        d = {
            "allocated_mb": _mb(torch.cuda.memory_allocated(self.device)),
            "reserved_mb": _mb(torch.cuda.memory_reserved(self.device)),
            "peak_mb": _mb(torch.cuda.max_memory_allocated(self.device)),
        }
        self.mem_snaps[f"iter{self.iteration}_{tag}"] = d

    def reset_peak(self) -> None:
        if self.enabled:
            torch.cuda.reset_peak_memory_stats(self.device)

    def profile_memory(self, stage: str) -> None:
        """stage in {Before-Fwd, After-Fwd, After-Bwd, After-step}
        (reference runtime_profiler.py:105-194)."""
        if stage == "Before-Fwd":
            self.reset_peak()
        self.snap(stage)

    def memory_summary(self) -> Dict[str, float]:
        """model-states / activation split from the 4-point snapshots
        (reference post_profile_memory:134-194).  Uses the LAST iteration
        that actually recorded a Before-Fwd snap (after the training loop
        self.iteration already points one past the end)."""
        it = self.iteration
        while it > 0 and f"iter{it}_Before-Fwd" not in self.mem_snaps:
            it -= 1
        g = lambda tag, k="allocated_mb": self.mem_snaps.get(
            f"iter{it}_{tag}", {}).get(k, 0.0)
        before = g("Before-Fwd")
        after_fwd = g("After-Fwd")
        peak = self.mem_snaps.get(f"iter{it}_After-Bwd", {}).get("peak_mb", 0.0)
        return {
            "model_states_mb": before,
            "activation_mb": max(after_fwd - before, 0.0),
            "peak_activation_mb": max(peak - before, 0.0),
            "peak_mb": peak,
        }

    # -- time --------------------------------------------------------------
    def time_start(self) -> None:
        if not self.enabled:
            return
        self._ev_start = torch.cuda.Event(enable_timing=True)
        self._ev_end = torch.cuda.Event(enable_timing=True)
        self._ev_start.record()

    def fwd_start(self) -> None:
        """Bracket one forward chunk (engine calls these around each
        microbatch forward; reference profile_forward methodology — the
        computation profile's fct is FORWARD time, bct derives from it)."""
        if not self.enabled:
            return
        a = torch.cuda.Event(enable_timing=True)
        b = torch.cuda.Event(enable_timing=True)
        a.record()
        self._fwd_pairs.append((a, b))

    def fwd_end(self) -> None:
        if self.enabled and self._fwd_pairs:
            self._fwd_pairs[-1][1].record()

    def time_end(self) -> Optional[float]:
        if not self.enabled or self._ev_start is None:
            self.iteration += 1
            return None
        self._ev_end.record()
        torch.cuda.synchronize(self.device)
        ms = self._ev_start.elapsed_time(self._ev_end)
        if self._fwd_pairs:
            fwd = sum(a.elapsed_time(b) for a, b in self._fwd_pairs)
            self._fwd_pairs = []
            if self.iteration >= self.warmup:
                self.fwd_times_ms.append(fwd)
        if self.iteration >= self.warmup:
            self.iter_times_ms.append(ms)
        self.iteration += 1
        return ms

    def avg_iter_ms(self) -> float:
        if not self.iter_times_ms:
            return 0.0
        return sum(self.iter_times_ms) / len(self.iter_times_ms)

    # -- persistence -------------------------------------------------------
    def avg_fwd_ms(self) -> float:
        if not self.fwd_times_ms:
            return 0.0
        return sum(self.fwd_times_ms) / len(self.fwd_times_ms)

    def save_time_profile(self, path: str, key: str) -> None:
        """Append {key: avg FORWARD ms} into a computation_profiling JSON
        (reference key format 'layernum[N]_bsz{b}_seq{s}'; that value is
        the search's fct).  The full iteration time rides along under an
        'iter_'-prefixed key (prefix, not suffix, so the search's
        layernum regex never matches it)."""
        data = {}
        if os.path.exists(path):
            with open(path) as f:
                data = json.load(f)
        data[key] = self.avg_fwd_ms() or self.avg_iter_ms()
        data[f"iter_{key}"] = self.avg_iter_ms()
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        with open(path, "w") as f:
            json.dump(data, f, indent=4)

    def save_memory_profile(self, path: str, key_prefix: str) -> None:
        data = {}
        if os.path.exists(path):
            with open(path) as f:
                data = json.load(f)
        s = self.memory_summary()
        data[f"{key_prefix}_ms"] = s["model_states_mb"]
        data[f"{key_prefix}_act"] = s["activation_mb"]
        data[f"{key_prefix}_act_peak"] = s["peak_activation_mb"]
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        with open(path, "w") as f:
            json.dump(data, f, indent=4)

    # -- logging -----------------------------------------------------------
    def log_iteration(self, loss: float, lr: float, grad_norm: float,
                      interval: int = 1) -> None:
        if self.rank == 0 and self.iteration % max(interval, 1) == 0:
            ms = self.iter_times_ms[-1] if self.iter_times_ms else None
            ms_s = f"{ms:.1f} ms" if ms is not None else "- ms"
            print(f"iter {self.iteration:5d} | loss {loss:.4f} | "
                  f"lr {lr:.3e} | grad-norm {grad_norm:.3f} | "
                  f"{ms_s}", flush=True)


def get_runtime_profiler(cfg, device=None) -> RuntimeProfiler:
    rank = dist.get_rank() if dist.is_initialized() else 0
    return RuntimeProfiler(enabled=bool(cfg.profile.profile) or True,
                           device=device, rank=rank)
