"""SIGTERM all-gather -> graceful exit (reference dist_signal_handler.py)."""

from __future__ import annotations

import signal

import torch
import torch.distributed as dist


class DistributedSignalHandler:
    def __init__(self, sig=signal.SIGTERM):
        self.sig = sig
        self._signal_received = False
        self._prev_handler = None

    def signals_received(self):
        """All-gather the local flag: every rank learns if ANY rank got it."""
        local = torch.tensor([1.0 if self._signal_received else 0.0])
        if dist.is_initialized():
            dist.all_reduce(local, op=dist.ReduceOp.MAX)
        return [bool(local.item())] * (
            dist.get_world_size() if dist.is_initialized() else 1)

    def __enter__(self):
        def handler(signum, frame):
            self._signal_received = True
        self._prev_handler = signal.getsignal(self.sig)
        signal.signal(self.sig, handler)
        return self

    def __exit__(self, *exc):
        if self._prev_handler is not None:
            signal.signal(self.sig, self._prev_handler)
        return False
