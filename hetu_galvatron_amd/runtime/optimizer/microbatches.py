"""Global-batch / microbatch calculators (constant + linear ramp-up).

Reference: core/runtime/utils/num_microbatches_calculator.py:1-508
(ConstantNumMicroBatchesCalculator, RampupBatchsizeNumMicroBatchesCalculator):
the global batch ramps linearly from `start` to the target in `increment`
steps spread over `ramp_samples` consumed samples; the number of
microbatches follows as gbs / (micro_batch_size * dp).
"""
from __future__ import annotations

from typing import Optional, Tuple


class ConstantBatchCalculator:
    def __init__(self, global_batch_size: int, micro_batch_size: int,
                 dp: int):
        assert global_batch_size % (micro_batch_size * dp) == 0
        self.global_batch_size = global_batch_size
        self.micro_batch_size = micro_batch_size
        self.num_micro_batches = global_batch_size // (micro_batch_size * dp)

    def update(self, consumed_samples: int) -> None:
        pass

    def get(self) -> Tuple[int, int]:
        return self.global_batch_size, self.num_micro_batches


class RampupBatchCalculator:
    """rampup = (start, increment, ramp_samples): gbs grows from start by
    `increment` at evenly spaced consumed-sample milestones until it
    reaches the target."""

    def __init__(self, start: int, increment: int, ramp_samples: int,
                 global_batch_size: int, micro_batch_size: int, dp: int):
        unit = micro_batch_size * dp
        assert start % unit == 0 and increment % unit == 0 \
            and global_batch_size % unit == 0, \
            "rampup sizes must divide micro_batch*dp"
        assert (global_batch_size - start) % increment == 0, \
            "(target - start) must be a multiple of increment"
        self.start = start
        self.increment = increment
        self.ramp_samples = ramp_samples
        self.target = global_batch_size
        self.micro_batch_size = micro_batch_size
        self.dp = dp
        self.num_steps = (global_batch_size - start) // increment
        self.samples_per_step = ramp_samples / max(self.num_steps, 1)
        self.global_batch_size = start
        self.num_micro_batches = start // unit

    def update(self, consumed_samples: int) -> None:
        if consumed_samples >= self.ramp_samples:
            gbs = self.target
        else:
            steps = int(consumed_samples / self.samples_per_step)
            gbs = min(self.start + steps * self.increment, self.target)
        self.global_batch_size = gbs
        self.num_micro_batches = gbs // (self.micro_batch_size * self.dp)

    def get(self) -> Tuple[int, int]:
        return self.global_batch_size, self.num_micro_batches


def build_batch_calculator(cfg, dp: int, micro_batch_size: int):
    """cfg.train.rampup_batch_size: "start,increment,ramp_samples" or None."""
    spec = getattr(cfg.train, "rampup_batch_size", None)
    gbs = cfg.train.global_train_batch_size
    if not spec:
        return ConstantBatchCalculator(gbs, micro_batch_size, dp)
    if isinstance(spec, (list, tuple)):
        parts = [int(v) for v in spec]
    else:
        parts = [int(v) for v in str(spec).replace(" ", "").split(",")]
    assert len(parts) == 3, "rampup_batch_size = start,increment,ramp_samples"
    return RampupBatchCalculator(parts[0], parts[1], parts[2], gbs,
                                 micro_batch_size, dp)
