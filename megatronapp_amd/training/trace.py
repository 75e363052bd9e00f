"""MegaScan tracer: async GPU-event scope timing with one sync per
iteration, windowed activation, rank-0 gather and per-rank JSON traces.

Reference behaviours replicated (trace.py — SURVEY.md §2.2/§2.6):
* scopes record pairs of ``torch.cuda.Event(enable_timing=True)`` (ROCm
  hipEvent_t) with NO host sync at scope boundaries (~µs overhead);
* ``iteration_end`` performs one ``torch.cuda.synchronize`` and converts
  event pairs to relative-timestamp records via ``elapsed_time``;
* records carry dp/pp/tp/global rank + device; collective scopes carry
  ``group`` (peer ranks) and ``data`` (bytes) from which the aggregator
  derives bandwidth;
* wall-clock ``pad_before``/``duration_wall`` per iteration allow the
  offline aggregator to stitch a global timeline without a global clock;
* windowed activation: trace ``continuous_iters`` every ``interval``
  iterations; granularity "base" keeps only schedule-level events, "full"
  adds per-layer/per-collective scopes;
* per-rank file: ``benchmark-data-{dp}-pipeline-{pp}-tensor-{tp}.json``
  (JSON array; schema in SURVEY.md §2.6) written by a background thread
  on rank 0 after a ``gather_object``.

On CPU (tests) the same code path uses perf_counter timestamps.
"""

from __future__ import annotations

import json
import os
import queue
import threading
import time
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..core import parallel_state
from ..core.trace_hooks import register_tracer

BASE_TRACING_EVENTS = {
    "iteration", "forward", "backward", "loss", "optimizer", "allreduce",
    "grad-sync", "grad-sync-dp", "grad-sync-embedding", "grad-sync-layernorm",
    "recv-forward", "recv-backward", "send-forward", "send-backward",
    "send-forward-recv-backward", "send-backward-recv-forward",
    "exchange-next", "exchange-prev",
}


class _TracerScope:
    __slots__ = ("tracer", "name", "attrs")

    def __init__(self, tracer: "Tracer", name: str, attrs: dict):
        self.tracer = tracer
        self.name = name
        self.attrs = attrs

    def __enter__(self):
        self.tracer._tick(self.name, "B", {})
        return self

    def __exit__(self, *exc):
        self.tracer._tick(self.name, "E", self.attrs)
        return False


class Tracer:
    _instance: Optional["Tracer"] = None

    def __init__(self, trace_dir: str = "trace_out", interval: int = 5,
                 continuous_iters: int = 2, granularity: str = "full",
                 max_iters: Optional[int] = None):
        self.trace_dir = trace_dir
        self.interval = max(interval, 1)
        self.continuous_iters = continuous_iters
        self.granularity = granularity
        self.max_iters = max_iters

        self.iteration = -1
        self._active = False
        self._use_cuda = torch.cuda.is_available()
        self._events: List[tuple] = []  # (name, ph, event_or_ts, attrs)
        self._iter_begin_event = None
        self._iter_begin_wall = None
        self._prev_iter_end_wall = None

        self._work_queue: "queue.Queue" = queue.Queue()
        self._writer_thread: Optional[threading.Thread] = None
        self._shutdown = False

        self._ranks_cached = None

    # ------------------------------------------------------------------
    @classmethod
    def initialize(cls, trace_dir="trace_out", interval=5, continuous_iters=2,
                   granularity="full", max_iters=None) -> "Tracer":
        tracer = cls(trace_dir, interval, continuous_iters, granularity,
                     max_iters)
        cls._instance = tracer
        register_tracer(tracer)
        if (not dist.is_initialized()) or dist.get_rank() == 0:
            os.makedirs(trace_dir, exist_ok=True)
            tracer._writer_thread = threading.Thread(
                target=tracer._save_traces_to_disk_thread, daemon=True)
            tracer._writer_thread.start()
        return tracer

    @classmethod
    def get(cls) -> Optional["Tracer"]:
        return cls._instance

    # ------------------------------------------------------------------
    def _rank_info(self):
        if self._ranks_cached is None:
            if parallel_state.model_parallel_is_initialized():
                self._ranks_cached = dict(
                    dp_rk=parallel_state.get_data_parallel_rank(),
                    pp_rk=parallel_state.get_pipeline_model_parallel_rank(),
                    tp_rk=parallel_state.get_tensor_model_parallel_rank(),
                    g_rk=dist.get_rank() if dist.is_initialized() else 0,
                )
            else:
                self._ranks_cached = dict(dp_rk=0, pp_rk=0, tp_rk=0, g_rk=0)
            self._ranks_cached["dev"] = (
                torch.cuda.current_device() if self._use_cuda else -1)
        return self._ranks_cached

    def is_tracing_active(self) -> bool:
        return self._active

    def _window_active(self, iteration: int) -> bool:
        if self.max_iters is not None and iteration >= self.max_iters:
            return False
        return (iteration % self.interval) < self.continuous_iters

    # ------------------------------------------------------------------
    def scope(self, name: str, **attrs) -> _TracerScope:
        return _TracerScope(self, name, attrs)

    def instant(self, name: str, **attrs) -> None:
        self._tick(name, "i", attrs)

    def _keep(self, name: str) -> bool:
        if self.granularity == "full":
            return True
        return name in BASE_TRACING_EVENTS

    def _tick(self, name: str, ph: str, attrs: dict) -> None:
        if not self._active or not self._keep(name):
            return
        if self._use_cuda:
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
        else:
            ev = time.perf_counter_ns()
        self._events.append((name, ph, ev, attrs))

    # ------------------------------------------------------------------
    def iteration_begin(self, iteration: int) -> None:
        self.iteration = iteration
        self._active = self._window_active(iteration)
        if not self._active:
            return
        self._events = []
        self._iter_begin_wall = time.time_ns()
        if self._use_cuda:
            self._iter_begin_event = torch.cuda.Event(enable_timing=True)
            self._iter_begin_event.record()
        else:
            self._iter_begin_event = time.perf_counter_ns()

    def iteration_end(self) -> None:
        if not self._active:
            return
        if self._use_cuda:
            torch.cuda.synchronize()
        end_wall = time.time_ns()
        records = self._process_pending_scopes(end_wall)
        self._active = False
        self._log(records)
        self._prev_iter_end_wall = end_wall

    def _elapsed_ns(self, ev) -> int:
        if self._use_cuda:
            return int(self._iter_begin_event.elapsed_time(ev) * 1e6)
        return ev - self._iter_begin_event

    def _process_pending_scopes(self, end_wall: int) -> List[dict]:
        info = self._rank_info()
        pad_before = (0 if self._prev_iter_end_wall is None
                      else self._iter_begin_wall - self._prev_iter_end_wall)
        duration_wall = end_wall - self._iter_begin_wall
        records: List[dict] = [{
            "name": "iteration", "ph": "B", "rel_ts": 0,
            "iteration": self.iteration, "pad_before": pad_before, **info,
        }]
        last_ts = 0
        for name, ph, ev, attrs in self._events:
            rel_ts = self._elapsed_ns(ev)
            last_ts = max(last_ts, rel_ts)
            rec = {"name": name, "ph": ph, "rel_ts": rel_ts, **info}
            if attrs:
                for k, v in attrs.items():
                    if v is not None:
                        rec[k] = v
                if "data" in rec and ph == "E":
                    # bandwidth in Gbps over the scope duration
                    dur = self._find_scope_duration(name, rel_ts)
                    if dur and dur > 0:
                        rec["bandwidth"] = rec["data"] * 8.0 / dur
            records.append(rec)
        records.append({
            "name": "iteration", "ph": "E", "rel_ts": last_ts,
            "iteration": self.iteration, "duration_wall": duration_wall,
            "duration_cuda": last_ts, **info,
        })
        return records

    def _find_scope_duration(self, name: str, end_ts: int) -> Optional[int]:
        # nearest preceding B of same name (records are in program order)
        begin = None
        for n, ph, ev, _ in self._events:
            if n == name:
                ts = self._elapsed_ns(ev)
                if ph == "B" and ts <= end_ts:
                    begin = ts
        return None if begin is None else end_ts - begin

    # ------------------------------------------------------------------
    def _log(self, records: List[dict]) -> None:
        if not dist.is_initialized() or dist.get_world_size() == 1:
            gathered = [records]
        else:
            gathered = [None] * dist.get_world_size() if dist.get_rank() == 0 else None
            dist.gather_object(records, gathered, dst=0)
        if gathered is not None:
            for rank_records in gathered:
                if rank_records:
                    self._work_queue.put(rank_records)

    def _save_traces_to_disk_thread(self) -> None:
        files: Dict[str, list] = {}
        while True:
            try:
                item = self._work_queue.get(timeout=0.25)
            except queue.Empty:
                if self._shutdown:
                    break
                continue
            if item is None:
                break
            first = item[0]
            key = (f"benchmark-data-{first.get('dp_rk', 0)}-pipeline-"
                   f"{first.get('pp_rk', 0)}-tensor-{first.get('tp_rk', 0)}.json")
            files.setdefault(key, []).extend(item)
            self._flush(files)

    def _flush(self, files: Dict[str, list]) -> None:
        for fname, records in files.items():
            path = os.path.join(self.trace_dir, fname)
            with open(path, "w") as f:
                json.dump(records, f)

    def shutdown(self) -> None:
        self._shutdown = True
        if self._writer_thread is not None:
            self._work_queue.put(None)
            self._writer_thread.join(timeout=10)
        Tracer._instance = None
        register_tracer(None)


def get_tracer() -> Optional[Tracer]:
    return Tracer.get()


def get_tensor_bytes(t: torch.Tensor) -> int:
    return t.numel() * t.element_size()
