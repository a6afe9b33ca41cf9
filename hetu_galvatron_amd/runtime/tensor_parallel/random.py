"""RNG state tracking for parallel regions.

Reference: galvatron/core/runtime/tensor_parallel/random.py:120-325
(Megatron CudaRNGStatesTracker).  TP ranks must draw DISTINCT dropout masks
inside the model-parallel region while replicated (DP) ranks draw identical
ones; this tracker forks named RNG streams per region.
"""
from __future__ import annotations

import contextlib
from typing import Dict

import torch

_MODEL_PARALLEL_RNG = "model-parallel-rng"
_EXPERT_PARALLEL_RNG = "expert-parallel-rng"


class RNGStatesTracker:
    def __init__(self) -> None:
        self.states_: Dict[str, torch.Tensor] = {}

    def reset(self) -> None:
        self.states_ = {}

    def _get_state(self):
        if torch.cuda.is_available():
            return torch.cuda.get_rng_state()
        return torch.get_rng_state()

    def _set_state(self, state):
        if torch.cuda.is_available():
            torch.cuda.set_rng_state(state)
        else:
            torch.set_rng_state(state)

    def add(self, name: str, seed: int) -> None:
        if name in self.states_:
            raise RuntimeError(f"rng state {name} already exists")
        orig = self._get_state()
        if torch.cuda.is_available():
            torch.cuda.manual_seed(seed)
        else:
            torch.manual_seed(seed)
        self.states_[name] = self._get_state()
        self._set_state(orig)

    @contextlib.contextmanager
    def fork(self, name: str = _MODEL_PARALLEL_RNG):
        if name not in self.states_:
            yield  # tracker unseeded (e.g. dropout==0 paths): no-op
            return
        orig = self._get_state()
        self._set_state(self.states_[name])
        try:
            yield
        finally:
            self.states_[name] = self._get_state()
            self._set_state(orig)


_TRACKER = RNGStatesTracker()


def get_rng_tracker() -> RNGStatesTracker:
    return _TRACKER


def model_parallel_seed(seed: int, tp_rank: int, ep_rank: int = 0) -> None:
    """Seed the named streams: model-parallel offset by tp rank so TP shards
    get distinct dropout (reference: random.py:279 set_seed_with_group)."""
    tracker = get_rng_tracker()
    tracker.reset()
    tracker.add(_MODEL_PARALLEL_RNG, seed + 2718 + tp_rank)
    tracker.add(_EXPERT_PARALLEL_RNG, seed + 5042 + tp_rank + 1000 * ep_rank)
