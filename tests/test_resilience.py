"""Rerun state machine + straggler detector tests (CPU)."""

import torch

from .utils import destroy, initialize_model_parallel


def test_rerun_reproducible_step_passes():
    from megatronapp_amd.core.rerun_state_machine import (
        RerunDataIterator, RerunStateMachine, RerunMode)
    initialize_model_parallel()
    sm = RerunStateMachine(RerunMode.VALIDATE_RESULTS, check_interval=1)
    data = RerunDataIterator(iter(range(100)))

    runs = 0
    torch.manual_seed(7)
    results = []
    while sm.should_run_forward_backward(data):
        x = next(data)
        val = torch.randn(4).sum() + x  # deterministic given restored RNG
        results.append(float(val))
        sm.record_result(val)
        runs += 1
    assert runs == 2, "validate mode must run the step twice"
    assert results[0] == results[1], "rerun must replay identical RNG+data"
    assert not sm.should_checkpoint_and_exit()
    destroy()


def test_rerun_detects_irreproducible():
    from megatronapp_amd.core.rerun_state_machine import (
        RerunDataIterator, RerunStateMachine, RerunMode)
    sm = RerunStateMachine(RerunMode.VALIDATE_RESULTS, check_interval=1)
    data = RerunDataIterator(iter(range(100)))
    flaky = iter([1.0, 2.0])  # simulated transient corruption
    while sm.should_run_forward_backward(data):
        next(data)
        sm.record_result(next(flaky))
    assert sm.should_checkpoint_and_exit()
    assert sm.irreproducible_steps == [1]


def test_rerun_disabled_runs_once():
    from megatronapp_amd.core.rerun_state_machine import RerunStateMachine
    sm = RerunStateMachine()
    runs = 0
    while sm.should_run_forward_backward(None):
        runs += 1
        sm.record_result(0.0)
    assert runs == 1


def test_straggler_detector_reports():
    initialize_model_parallel()
    from megatronapp_amd.core.straggler_detector import StragglerDetector
    det = StragglerDetector(report_interval=2, flops_per_step=1e12)
    for _ in range(2):
        with det:
            torch.randn(256, 256) @ torch.randn(256, 256)
    r = det.report()  # second explicit report has empty buffer -> None
    det2 = StragglerDetector(report_interval=100)
    with det2:
        pass
    out = det2.report()
    assert out is not None and out["min_rank"] == 0
    destroy()
