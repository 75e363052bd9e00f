"""Hierarchical timers with optional barriers (reference core/timers.py, 450 LoC)."""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


class _TimerBase:
    def __init__(self, name):
        self.name = name

    def start(self, barrier=False):
        raise NotImplementedError

    def stop(self, barrier=False):
        raise NotImplementedError

    def reset(self):
        raise NotImplementedError

    def elapsed(self, reset=True, barrier=False):
        raise NotImplementedError


class DummyTimer(_TimerBase):
    def __init__(self):
        super().__init__("dummy")

    def start(self, barrier=False):
        pass

    def stop(self, barrier=False):
        pass

    def reset(self):
        pass

    def elapsed(self, reset=True, barrier=False):
        raise Exception("dummy timer has no elapsed time")


class Timer(_TimerBase):
    def __init__(self, name):
        super().__init__(name)
        self._elapsed = 0.0
        self._active_time = 0.0
        self._started = False
        self._start_time = time.time()

    def _sync(self, barrier):
        if barrier and dist.is_initialized():
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    def start(self, barrier=False):
        assert not self._started, f"timer {self.name} already started"
        self._sync(barrier)
        self._start_time = time.time()
        self._started = True

    def stop(self, barrier=False):
        assert self._started
        self._sync(barrier)
        dt = time.time() - self._start_time
        self._elapsed += dt
        self._active_time += dt
        self._started = False

    def reset(self):
        self._elapsed = 0.0
        self._started = False

    def elapsed(self, reset=True, barrier=False):
        was_started = self._started
        if was_started:
            self.stop(barrier=barrier)
        e = self._elapsed
        if reset:
            self.reset()
        if was_started:
            self.start(barrier=barrier)
        return e

    def active_time(self):
        return self._active_time


class Timers:
    def __init__(self, log_level: int = 0, log_option: str = "minmax"):
        self._log_level = log_level
        self._log_option = log_option
        self._timers: Dict[str, Timer] = {}
        self._log_levels: Dict[str, int] = {}
        self._dummy = DummyTimer()
        self._max_log_level = 2

    def __call__(self, name, log_level: Optional[int] = None):
        if name in self._timers:
            return self._timers[name]
        if log_level is None:
            log_level = self._max_log_level
        if log_level > self._log_level:
            return self._dummy
        self._timers[name] = Timer(name)
        self._log_levels[name] = log_level
        return self._timers[name]

    def log(self, names: List[str], rank=None, normalizer: float = 1.0,
            reset=True, barrier=False):
        output = self.get_all_timers_string(names, normalizer, reset, barrier)
        if output is None:
            return
        if rank is None or not dist.is_initialized() or dist.get_rank() == rank:
            print(output, flush=True)

    def get_all_timers_string(self, names=None, normalizer=1.0, reset=True,
                              barrier=False):
        if names is None:
            names = list(self._timers.keys())
        fields = []
        for name in names:
            if name not in self._timers:
                continue
            e = self._timers[name].elapsed(reset=reset, barrier=barrier)
            fields.append(f"{name}: {e * 1000.0 / normalizer:.2f}")
        if not fields:
            return None
        return "time (ms) | " + " | ".join(fields)

    def write(self, names, writer, iteration, normalizer=1.0, reset=True,
              barrier=False):
        for name in names:
            if name in self._timers:
                e = self._timers[name].elapsed(reset=reset, barrier=barrier)
                if writer is not None:
                    writer.add_scalar(f"timers/{name}", e / normalizer, iteration)
