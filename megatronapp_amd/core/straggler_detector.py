"""Built-in per-rank straggler detector (reference core/utils.py
StragglerDetector:1030).

GPU-event timed sections (torch.cuda.Event = hipEvent_t) around the train
step; periodically all-gathers per-rank throughput estimates and reports
the min/max ranks so a slow GPU surfaces without MegaScan's full traces.
"""

from __future__ import annotations

import time
from typing import List, Optional

import torch
import torch.distributed as dist


class StragglerDetector:
    def __init__(self, report_interval: int = 0, flops_per_step: float = 0.0):
        self.enabled = report_interval > 0
        self.report_interval = report_interval
        self.flops_per_step = flops_per_step
        self._use_cuda = torch.cuda.is_available()
        self._events = []
        self._elapsed_ms: List[float] = []
        self._step = 0

    def __enter__(self):
        if not self.enabled:
            return self
        if self._use_cuda:
            start = torch.cuda.Event(enable_timing=True)
            start.record()
            self._events = [start]
        else:
            self._events = [time.perf_counter()]
        return self

    def __exit__(self, *exc):
        if not self.enabled:
            return False
        if self._use_cuda:
            end = torch.cuda.Event(enable_timing=True)
            end.record()
            end.synchronize()
            self._elapsed_ms.append(self._events[0].elapsed_time(end))
        else:
            self._elapsed_ms.append(
                (time.perf_counter() - self._events[0]) * 1000.0)
        self._step += 1
        if self._step % self.report_interval == 0:
            self.report()
        return False

    def report(self):
        if not self._elapsed_ms:
            return None
        mean_ms = sum(self._elapsed_ms) / len(self._elapsed_ms)
        self._elapsed_ms = []
        etpt = (self.flops_per_step / (mean_ms / 1000.0) / 1e12
                if self.flops_per_step else 1000.0 / mean_ms)
        if not dist.is_initialized() or dist.get_world_size() == 1:
            return {"min_rank": 0, "max_rank": 0, "min": etpt, "max": etpt}
        world = dist.get_world_size()
        device = "cuda" if self._use_cuda else "cpu"
        mine = torch.tensor([etpt], dtype=torch.float64, device=device)
        all_ = [torch.empty_like(mine) for _ in range(world)]
        dist.all_gather(all_, mine)
        vals = [t.item() for t in all_]
        lo = min(range(world), key=lambda r: vals[r])
        hi = max(range(world), key=lambda r: vals[r])
        result = {"min_rank": lo, "max_rank": hi, "min": vals[lo],
                  "max": vals[hi],
                  "ratio": vals[lo] / vals[hi] if vals[hi] else 1.0}
        if dist.get_rank() == 0:
            unit = "TFLOP/s" if self.flops_per_step else "steps/s"
            print(f"[straggler] slowest rank {lo} ({vals[lo]:.2f} {unit}) | "
                  f"fastest rank {hi} ({vals[hi]:.2f} {unit}) | "
                  f"ratio {result['ratio']:.3f}", flush=True)
            if result["ratio"] < 0.85:
                print(f"[straggler] WARNING: rank {lo} is >15% slower than "
                      "the fastest rank — possible degraded GPU", flush=True)
        return result
