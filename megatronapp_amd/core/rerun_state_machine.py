"""Rerun state machine: in-place step replay to classify irreproducible
results (silent data corruption detection).

Reference: core/rerun_state_machine.py (1341 LoC;
should_run_forward_backward :255, should_checkpoint_and_exit :378),
driven from train_step (training.py:1387-1410), --rerun-mode.

Redesigned control flow (same semantics, smaller surface):
  state INITIAL  -> run step normally, record result + RNG/data snapshot
  on validation request (rerun mode "validate_results"): replay the SAME
  step with restored RNG/data; compare -> REPRODUCIBLE (determinism holds)
  or IRREPRODUCIBLE (possible transient hardware fault: report, optionally
  request checkpoint+exit).
"""

from __future__ import annotations

import enum
import logging
from typing import Any, Callable, List, Optional

import torch

logger = logging.getLogger(__name__)


class RerunMode(str, enum.Enum):
    DISABLED = "disabled"
    VALIDATE_RESULTS = "validate_results"
    REPORT_STATS = "report_stats"


class RerunState(enum.Enum):
    NOT_RUNNING_YET = 0
    FIRST_RUN = 1
    RERUNNING = 2
    DONE = 3


class RerunDataIterator:
    """Replayable wrapper: records every batch so a rerun replays the
    exact data (reference RerunDataIterator)."""

    def __init__(self, iterator):
        self.iterator = iterator
        self._history: List[Any] = []
        self._replay_pos: Optional[int] = None

    def __iter__(self):
        return self

    def __next__(self):
        if self._replay_pos is not None:
            item = self._history[self._replay_pos]
            self._replay_pos += 1
            return item
        item = next(self.iterator)
        self._history.append(item)
        return item

    def rewind(self):
        self._replay_pos = 0

    def advance(self):
        self._history.clear()
        self._replay_pos = None


class RerunStateMachine:
    def __init__(self, mode: RerunMode = RerunMode.DISABLED,
                 check_interval: int = 0):
        self.mode = RerunMode(mode)
        self.check_interval = check_interval
        self.state = RerunState.NOT_RUNNING_YET
        self.step = 0
        self._first_result = None
        self._rng_snapshot = None
        self._stats: List[float] = []
        self.irreproducible_steps: List[int] = []
        self._exit_requested = False

    # -- main protocol (mirrors reference usage in train_step) -----------
    def should_run_forward_backward(self, data_iterator) -> bool:
        """True while another (re)run of THIS step is needed; the call
        after the step completes returns False and resets for the next
        train_step invocation."""
        if self.mode == RerunMode.DISABLED:
            if self.state == RerunState.NOT_RUNNING_YET:
                self.state = RerunState.FIRST_RUN
                return True
            self.state = RerunState.NOT_RUNNING_YET
            return False
        if self.state == RerunState.NOT_RUNNING_YET:
            self.step += 1
            validate = (self.check_interval > 0 and
                        self.mode == RerunMode.VALIDATE_RESULTS and
                        self.step % self.check_interval == 0)
            self._validating = validate
            if validate:
                self._rng_snapshot = self._snapshot_rng()
                if isinstance(data_iterator, RerunDataIterator):
                    data_iterator.advance()
            self.state = RerunState.FIRST_RUN
            return True
        if self.state == RerunState.FIRST_RUN and                 getattr(self, "_validating", False):
            # restore for the rerun
            self._restore_rng(self._rng_snapshot)
            if isinstance(data_iterator, RerunDataIterator):
                data_iterator.rewind()
            self.state = RerunState.RERUNNING
            return True
        # step complete
        self.state = RerunState.NOT_RUNNING_YET
        return False

    def record_result(self, value: torch.Tensor | float) -> None:
        v = float(value.item() if torch.is_tensor(value) else value)
        if self.mode == RerunMode.REPORT_STATS:
            self._stats.append(v)
        if self.state == RerunState.FIRST_RUN:
            self._first_result = v
        elif self.state == RerunState.RERUNNING:
            if v != self._first_result:
                self.irreproducible_steps.append(self.step)
                logger.error(
                    "RerunStateMachine: step %d IRREPRODUCIBLE "
                    "(%.9g vs %.9g) — possible transient hardware fault",
                    self.step, self._first_result, v)
                self._exit_requested = True

    def should_checkpoint_and_exit(self) -> bool:
        return self._exit_requested

    def is_first_run(self) -> bool:
        return self.state == RerunState.FIRST_RUN

    # -- rng snapshot ----------------------------------------------------
    @staticmethod
    def _snapshot_rng():
        from .tensor_parallel.random import get_cuda_rng_tracker
        snap = {"cpu": torch.get_rng_state(),
                "tracker": get_cuda_rng_tracker().get_states()}
        if torch.cuda.is_available():
            snap["cuda"] = torch.cuda.get_rng_state()
        return snap

    @staticmethod
    def _restore_rng(snap):
        from .tensor_parallel.random import get_cuda_rng_tracker
        torch.set_rng_state(snap["cpu"])
        get_cuda_rng_tracker().set_states(snap["tracker"])
        if torch.cuda.is_available() and "cuda" in snap:
            torch.cuda.set_rng_state(snap["cuda"])


_RERUN_STATE_MACHINE: Optional[RerunStateMachine] = None


def initialize_rerun_state_machine(mode="disabled", check_interval=0):
    global _RERUN_STATE_MACHINE
    _RERUN_STATE_MACHINE = RerunStateMachine(RerunMode(mode), check_interval)
    return _RERUN_STATE_MACHINE


def get_rerun_state_machine() -> RerunStateMachine:
    global _RERUN_STATE_MACHINE
    if _RERUN_STATE_MACHINE is None:
        _RERUN_STATE_MACHINE = RerunStateMachine()
    return _RERUN_STATE_MACHINE
