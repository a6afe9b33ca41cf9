"""Rerun state-machine tests (fault attribution)
(reference: runtime/utils/rerun_state_machine.py semantics)."""
import math

import pytest

from hetu_galvatron_amd.runtime.rerun_state_machine import (
    EXIT_CODE_FAILED_ON_RESULT_VALIDATION, RerunDataIterator,
    RerunDiagnostic, RerunState, RerunStateMachine)


def drive(rsm, it, results):
    """Run the loop protocol feeding `results` per forward_backward call."""
    calls = 0
    while rsm.should_run_forward_backward(it):
        next(it)  # consume a batch
        rsm.validate_result(results[min(calls, len(results) - 1)])
        calls += 1
    return calls


def test_normal_iteration_runs_once():
    rsm = RerunStateMachine(enabled=True)
    it = RerunDataIterator(iter(range(100)))
    calls = drive(rsm, it, [1.0])
    assert calls == 1
    assert rsm.diagnostic is None
    assert next(it) == 1  # advanced past the consumed batch


def test_transient_fault_detected():
    """NaN once, clean on replay of the same data => transient HW error."""
    rsm = RerunStateMachine(enabled=True)
    it = RerunDataIterator(iter(range(100)))
    calls = drive(rsm, it, [float("nan"), 1.0])
    assert calls == 2
    assert rsm.diagnostic == RerunDiagnostic.TRANSIENT_ERROR
    assert not rsm.request_checkpoint_and_exit
    # the re-run replayed the SAME batch
    assert it._record == [] and next(it) == 1


def test_persistent_fault_requests_exit():
    """Same NaN three times => persistent; checkpoint-and-exit code 16."""
    rsm = RerunStateMachine(enabled=True)
    it = RerunDataIterator(iter(range(100)))
    calls = drive(rsm, it, [float("nan"), float("nan"), float("nan")])
    assert calls == 3
    assert rsm.diagnostic == RerunDiagnostic.PERSISTENT_ERROR
    assert rsm.request_checkpoint_and_exit
    assert rsm.exit_code == EXIT_CODE_FAILED_ON_RESULT_VALIDATION
    assert rsm.skipped == [0]


def test_spike_detection_uses_history():
    rsm = RerunStateMachine(enabled=True, spike_factor=5.0)
    it = RerunDataIterator(iter(range(100)))
    for v in (1.0, 1.1, 0.9, 1.0):
        drive(rsm, it, [v])
    assert rsm._is_unexpected(50.0)
    assert not rsm._is_unexpected(1.5)


def test_replay_iterator_rewind():
    it = RerunDataIterator(iter([10, 11, 12]))
    assert next(it) == 10 and next(it) == 11
    it.rewind()
    assert next(it) == 10 and next(it) == 11 and next(it) == 12
    it.advance()


def test_state_dict_roundtrip():
    rsm = RerunStateMachine(enabled=True)
    it = RerunDataIterator(iter(range(10)))
    drive(rsm, it, [1.0])
    sd = rsm.state_dict()
    r2 = RerunStateMachine(enabled=True)
    r2.load_state_dict(sd)
    assert r2.iteration == 1 and r2._history == [1.0]


def test_disabled_passthrough():
    rsm = RerunStateMachine(enabled=False)
    it = iter(range(5))
    calls = drive(rsm, it, [float("nan")])
    assert calls == 1


def test_rerun_state_persists_in_checkpoint(tmp_path):
    """rsm state rides the distributed checkpoint (reference persists it
    at :871-902)."""
    import torch
    from hetu_galvatron_amd.config import load_config
    from hetu_galvatron_amd.runtime import (
        GalvatronModel, get_optimizer_and_param_scheduler)
    from hetu_galvatron_amd.runtime.checkpoint import (
        load_distributed_checkpoint, save_distributed_checkpoint)
    from hetu_galvatron_amd.runtime.rerun_state_machine import (
        initialize_rerun_state_machine)

    cfg = load_config(base={
        "model": {"model_name": "tiny-llama"},
        "train": {"global_train_batch_size": 2, "train_iters": 1,
                  "distributed_backend": "gloo"}})
    torch.manual_seed(0)
    model = GalvatronModel(cfg)
    opt, sched = get_optimizer_and_param_scheduler(model.stage_model, cfg)
    rsm = initialize_rerun_state_machine(enabled=True)
    rsm.skipped = [3, 7]  # fabricated attribution history
    save_distributed_checkpoint(model, opt, sched, cfg, 1, str(tmp_path),
                                rerun_state_machine=rsm)
    rsm2 = initialize_rerun_state_machine(enabled=True)
    load_distributed_checkpoint(model, opt, sched, cfg, str(tmp_path),
                                rerun_state_machine=rsm2)
    assert rsm2.state_dict() == rsm.state_dict()
