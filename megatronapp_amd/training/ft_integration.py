"""Fault-tolerance hooks (reference training/ft_integration.py).

The reference wraps nvidia_resiliency_ext heartbeats; that package does
not exist for this stack, so the same surface is provided with a
dependency-free implementation:

* heartbeat sections — each rank touches
  ``<dir>/heartbeat_rank{R}.json`` with the current section and timestamp;
  an external watchdog (or MegaScan's detector) flags ranks whose file
  goes stale.
* simulated faults — ``--simulate-fault {hang,exit} --simulate-fault-rank R
  --simulate-fault-delay S`` arms a timer that hangs or kills rank R mid
  training, for exercising the failure-detection tooling.
"""

from __future__ import annotations

import json
import os
import threading
import time

import torch.distributed as dist

_STATE = {"dir": None, "rank": 0}


def setup(args) -> None:
    if not getattr(args, "ft_heartbeat_dir", None):
        return
    _STATE["dir"] = args.ft_heartbeat_dir
    _STATE["rank"] = args.rank
    os.makedirs(args.ft_heartbeat_dir, exist_ok=True)
    on_section("setup")


def on_section(section: str) -> None:
    d = _STATE["dir"]
    if d is None:
        return
    path = os.path.join(d, f"heartbeat_rank{_STATE['rank']}.json")
    tmp = path + ".tmp"
    with open(tmp, "w") as f:
        json.dump({"section": section, "ts": time.time(),
                   "rank": _STATE["rank"]}, f)
    os.replace(tmp, path)


def on_training_step_start() -> None:
    on_section("train_step")


def on_training_step_end() -> None:
    on_section("idle")


def on_checkpointing_start() -> None:
    on_section("checkpoint")


def on_checkpointing_end(is_async_finalization: bool = False) -> None:
    on_section("idle")


def maybe_setup_simulated_fault(args) -> None:
    kind = getattr(args, "simulate_fault", None)
    if not kind:
        return
    if args.rank != getattr(args, "simulate_fault_rank", 0):
        return
    delay = getattr(args, "simulate_fault_delay", 30.0)

    def trigger():
        time.sleep(delay)
        print(f"[ft] simulating '{kind}' fault on rank {args.rank}",
              flush=True)
        if kind == "exit":
            os._exit(42)
        while True:  # hang
            time.sleep(3600)

    threading.Thread(target=trigger, daemon=True).start()
