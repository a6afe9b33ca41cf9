"""LR / weight-decay scheduler.

Reference: galvatron/core/runtime/optimizer/param_scheduler.py:102-385
(OptimizerParamScheduler: constant/linear/cosine/wsd decay + warmup).
"""
from __future__ import annotations

import math


class OptimizerParamScheduler:
    def __init__(self, optimizer, max_lr: float, min_lr: float = 0.0,
                 warmup_steps: int = 0, decay_steps: int = 1,
                 warmup_init_lr: float = 0.0,
                 decay_style: str = "cosine", wsd_decay_steps: int = 0,
                 start_wd: float = None, end_wd: float = None,
                 wd_incr_steps: int = 0):
        self.optimizer = optimizer
        self.max_lr = max_lr
        self.min_lr = min_lr
        self.warmup_steps = warmup_steps
        self.warmup_init_lr = warmup_init_lr
        self.decay_steps = max(decay_steps, 1)
        self.decay_style = decay_style
        self.wsd_decay_steps = wsd_decay_steps
        self.start_wd = start_wd
        self.end_wd = end_wd
        self.wd_incr_steps = wd_incr_steps
        self.num_steps = 0
        self.step(0)

    def get_lr(self) -> float:
        s = self.num_steps
        if self.warmup_steps > 0 and s <= self.warmup_steps:
            return self.warmup_init_lr + \
                (self.max_lr - self.warmup_init_lr) * s / self.warmup_steps
        if self.decay_style == "constant":
            return self.max_lr
        if s >= self.decay_steps:
            return self.min_lr
        frac = (s - self.warmup_steps) / max(self.decay_steps - self.warmup_steps, 1)
        frac = min(max(frac, 0.0), 1.0)
        dlr = self.max_lr - self.min_lr
        if self.decay_style == "linear":
            return self.max_lr - dlr * frac
        if self.decay_style == "cosine":
            return self.min_lr + dlr * 0.5 * (1 + math.cos(math.pi * frac))
        if self.decay_style == "wsd":
            # warmup-stable-decay: stable at max_lr until the final
            # wsd_decay_steps, then linear decay
            stable_end = self.decay_steps - self.wsd_decay_steps
            if s <= stable_end:
                return self.max_lr
            f = (s - stable_end) / max(self.wsd_decay_steps, 1)
            return self.max_lr - dlr * min(f, 1.0)
        raise ValueError(f"unknown decay style {self.decay_style}")

    def get_wd(self) -> float:
        if self.start_wd is None or self.end_wd is None or self.wd_incr_steps <= 0:
            return getattr(self.optimizer, "weight_decay", 0.0)
        f = min(self.num_steps / self.wd_incr_steps, 1.0)
        return self.start_wd + (self.end_wd - self.start_wd) * f

    def step(self, increment: int = 1) -> None:
        self.num_steps += increment
        lr = self.get_lr()
        self.optimizer.lr = lr
        if self.start_wd is not None:
            self.optimizer.weight_decay = self.get_wd()

    def state_dict(self) -> dict:
        return {"num_steps": self.num_steps}

    def load_state_dict(self, sd: dict) -> None:
        self.num_steps = sd["num_steps"]
        self.step(0)
