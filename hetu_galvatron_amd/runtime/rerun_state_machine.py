"""Rerun state machine: in-place fault attribution for unexpected results.

Reference: galvatron/core/runtime/utils/rerun_state_machine.py:127-1038
(wraps the train step; on an unexpected result — NaN/inf/spike — re-runs
the SAME iteration in place to classify it as CORRECT (flaky validation),
TRANSIENT_ERROR (same GPU, different result => transient HW fault) or
PERSISTENT_ERROR (reproducible => software or persistent HW), then
requests checkpoint-and-exit with dedicated exit codes so an outer
launcher can reschedule onto different GPUs).

Usage:
    rsm = get_rerun_state_machine()
    it = RerunDataIterator(data_iter)
    while rsm.should_run_forward_backward(it):
        loss = step(it)
        rsm.validate_result(loss)
    # <= here the iteration is final; rsm.state tells what happened
"""
from __future__ import annotations

import enum
import math
from typing import Any, Iterator, List, Optional

import torch

EXIT_CODE_FAILED_ON_RESULT_VALIDATION = 16
EXIT_CODE_SUCCESS_AFTER_RERUN = 17


class RerunState(enum.Enum):
    NOT_RUNNING_YET = 0
    RUNNING_NORMALLY = 1
    RERUNNING_IN_PLACE = 2          # first re-run: same GPU determinism check
    RERUNNING_AGAIN = 3             # second re-run: confirm
    DONE = 4


class RerunDiagnostic(enum.Enum):
    CORRECT = 0
    TRANSIENT_ERROR = 1
    PERSISTENT_ERROR = 2


class RerunDataIterator:
    """Rewindable wrapper: records fetched batches so an iteration can be
    replayed byte-identically (reference :989)."""

    def __init__(self, it: Iterator):
        self._it = it
        self._record: List[Any] = []
        self._replaying = False
        self._pos = 0

    def __iter__(self):
        return self

    def __next__(self):
        if self._replaying:
            if self._pos < len(self._record):
                v = self._record[self._pos]
                self._pos += 1
                return v
            self._replaying = False
        v = next(self._it)
        self._record.append(v)
        return v

    def rewind(self) -> None:
        self._replaying = True
        self._pos = 0

    def advance(self) -> None:
        """Iteration accepted: drop the recorded batches."""
        self._record = []
        self._replaying = False
        self._pos = 0


def _as_float(x) -> float:
    if isinstance(x, torch.Tensor):
        return float(x.detach().float().cpu())
    return float(x)


class RerunStateMachine:
    def __init__(self, enabled: bool = True, max_reruns: int = 2,
                 spike_factor: float = 10.0, history: int = 32):
        self.enabled = enabled
        self.max_reruns = max_reruns
        self.spike_factor = spike_factor
        self.state = RerunState.NOT_RUNNING_YET
        self.diagnostic: Optional[RerunDiagnostic] = None
        self.first_result: Optional[float] = None
        self.rerun_results: List[float] = []
        self._history: List[float] = []
        self._history_cap = history
        self.iteration = 0
        self.reruns_total = 0
        self.skipped: List[int] = []
        self.request_checkpoint_and_exit = False
        self.exit_code: Optional[int] = None

    # -- loop protocol -----------------------------------------------------
    def should_run_forward_backward(self, data_iter) -> bool:
        if not self.enabled:
            if self.state == RerunState.NOT_RUNNING_YET:
                self.state = RerunState.RUNNING_NORMALLY
                return True
            self.state = RerunState.NOT_RUNNING_YET
            return False
        if self.state == RerunState.NOT_RUNNING_YET:
            self.state = RerunState.RUNNING_NORMALLY
            self.diagnostic = None
            self.first_result = None
            self.rerun_results = []
            return True
        if self.state in (RerunState.RUNNING_NORMALLY, RerunState.DONE):
            # iteration finished (validation clean, or re-runs concluded)
            self.state = RerunState.NOT_RUNNING_YET
            if isinstance(data_iter, RerunDataIterator):
                data_iter.advance()
            return False
        if self.state in (RerunState.RERUNNING_IN_PLACE,
                          RerunState.RERUNNING_AGAIN):
            if isinstance(data_iter, RerunDataIterator):
                data_iter.rewind()
            self.reruns_total += 1
            return True
        return False

    def validate_result(self, result, tolerance: float = 0.0) -> None:
        """Call once per forward_backward with the loss (or any scalar
        invariant). Decides whether to re-run (reference :434)."""
        if not self.enabled:
            return
        v = _as_float(result)
        if self.state == RerunState.RUNNING_NORMALLY:
            if self._is_unexpected(v):
                self.first_result = v
                self.state = RerunState.RERUNNING_IN_PLACE
            else:
                self._note(v)
                self.iteration += 1
            return
        if self.state == RerunState.RERUNNING_IN_PLACE:
            self.rerun_results.append(v)
            if not self._is_unexpected(v):
                # different (sane) result on the same GPU => transient
                self._finish(RerunDiagnostic.TRANSIENT_ERROR, v)
            elif self._same(v, self.first_result):
                self.state = RerunState.RERUNNING_AGAIN
            else:
                self._finish(RerunDiagnostic.TRANSIENT_ERROR, v)
            return
        if self.state == RerunState.RERUNNING_AGAIN:
            self.rerun_results.append(v)
            if self._same(v, self.first_result):
                self._finish(RerunDiagnostic.PERSISTENT_ERROR, v)
            else:
                self._finish(RerunDiagnostic.TRANSIENT_ERROR, v)

    # -- internals ---------------------------------------------------------
    def _finish(self, diag: RerunDiagnostic, v: float) -> None:
        self.diagnostic = diag
        self.state = RerunState.DONE
        if diag == RerunDiagnostic.PERSISTENT_ERROR:
            # reproducible bad value: software bug or persistent HW fault —
            # checkpoint and exit so the launcher can reschedule
            self.request_checkpoint_and_exit = True
            self.exit_code = EXIT_CODE_FAILED_ON_RESULT_VALIDATION
            self.skipped.append(self.iteration)
        elif diag == RerunDiagnostic.TRANSIENT_ERROR:
            self.exit_code = EXIT_CODE_SUCCESS_AFTER_RERUN
            if not self._is_unexpected(v):
                self._note(v)
        self.iteration += 1

    def _is_unexpected(self, v: float) -> bool:
        if math.isnan(v) or math.isinf(v):
            return True
        if len(self._history) >= 4:
            mean = sum(self._history) / len(self._history)
            if abs(v) > self.spike_factor * max(abs(mean), 1e-8):
                return True
        return False

    def _same(self, a: float, b: float) -> bool:
        if a is None or b is None:
            return False
        if math.isnan(a) and math.isnan(b):
            return True
        if a == b:
            return True
        return abs(a - b) <= 1e-9 * max(abs(a), abs(b))

    def _note(self, v: float) -> None:
        self._history.append(v)
        if len(self._history) > self._history_cap:
            self._history.pop(0)

    # -- checkpoint persistence (reference :871-902) -----------------------
    def state_dict(self) -> dict:
        return {"iteration": self.iteration, "history": list(self._history),
                "skipped": list(self.skipped),
                "reruns_total": self.reruns_total}

    def load_state_dict(self, sd: dict) -> None:
        self.iteration = sd.get("iteration", 0)
        self._history = list(sd.get("history", []))
        self.skipped = list(sd.get("skipped", []))
        self.reruns_total = sd.get("reruns_total", 0)


_GLOBAL: Optional[RerunStateMachine] = None


def initialize_rerun_state_machine(enabled: bool = True, **kw) -> RerunStateMachine:
    global _GLOBAL
    _GLOBAL = RerunStateMachine(enabled=enabled, **kw)
    return _GLOBAL


def get_rerun_state_machine() -> RerunStateMachine:
    global _GLOBAL
    if _GLOBAL is None:
        _GLOBAL = RerunStateMachine(enabled=False)
    return _GLOBAL
