"""Metrics logging sinks: tensorboard / wandb / JSONL.

Reference: galvatron/core/runtime/parallel_state.py:85-130 (tensorboard +
wandb writer singletons configured from LoggingConfig) — here one
MetricsLogger with graceful degradation: tensorboard and wandb attach when
their packages exist (this image ships neither), and a JSONL sink
(<dir>/metrics.jsonl) always records, so runs are observable offline.
"""
from __future__ import annotations

import json
import os
import time
from typing import Dict


class MetricsLogger:
    def __init__(self, cfg=None, rank: int = 0):
        self.rank = rank
        self.enabled = rank == 0
        self._tb = None
        self._wandb = None
        self._jsonl = None
        if not self.enabled or cfg is None:
            return
        log = cfg.logging
        if log.tensorboard_dir:
            try:
                from torch.utils.tensorboard import SummaryWriter
                self._tb = SummaryWriter(log_dir=log.tensorboard_dir)
            except ImportError:
                pass
            os.makedirs(log.tensorboard_dir, exist_ok=True)
            self._jsonl = open(
                os.path.join(log.tensorboard_dir, "metrics.jsonl"), "a")
        if log.wandb_project:
            try:
                import wandb
                self._wandb = wandb
                wandb.init(project=log.wandb_project,
                           name=log.wandb_exp_name or None)
            except ImportError:
                pass

    def log(self, metrics: Dict[str, float], step: int) -> None:
        if not self.enabled:
            return
        if self._tb is not None:
            for k, v in metrics.items():
                self._tb.add_scalar(k, v, step)
        if self._wandb is not None:
            self._wandb.log(metrics, step=step)
        if self._jsonl is not None:
            self._jsonl.write(json.dumps(
                {"step": step, "ts": time.time(), **metrics}) + "\n")
            self._jsonl.flush()

    def close(self) -> None:
        if self._tb is not None:
            self._tb.close()
        if self._wandb is not None:
            self._wandb.finish()
        if self._jsonl is not None:
            self._jsonl.close()
