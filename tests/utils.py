"""Shared test utilities: single/multi-process distributed init.

Mirrors the reference's tests/unit_tests/test_utilities.py Utils pattern
(SURVEY.md §4): real collectives (gloo on CPU, RCCL on GPU), re-init per
topology.
"""

from __future__ import annotations

import os

import torch
import torch.distributed as dist

from megatronapp_amd.core import parallel_state


def init_distributed(backend=None):
    if dist.is_initialized():
        return
    backend = backend or ("nccl" if torch.cuda.is_available() else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29511")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    dist.init_process_group(backend=backend,
                            rank=int(os.environ["RANK"]),
                            world_size=int(os.environ["WORLD_SIZE"]))
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))


def initialize_model_parallel(tp=1, pp=1, vpp=None, cp=1, ep=1):
    init_distributed()
    if parallel_state.model_parallel_is_initialized():
        parallel_state.destroy_model_parallel()
    parallel_state.initialize_model_parallel(
        tensor_model_parallel_size=tp, pipeline_model_parallel_size=pp,
        virtual_pipeline_model_parallel_size=vpp, context_parallel_size=cp,
        expert_model_parallel_size=ep)


def destroy():
    parallel_state.destroy_model_parallel()


def spawn_ranks(fn, world_size=2, backend="gloo", args=()):
    """Run fn(rank, world_size, *args) in world_size processes (gloo/CPU)."""
    import torch.multiprocessing as mp
    port = _free_port()
    ctx = mp.spawn(_spawn_entry, args=(world_size, backend, port, fn, args),
                   nprocs=world_size, join=True)
    return ctx


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _spawn_entry(rank, world_size, backend, port, fn, args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
    try:
        fn(rank, world_size, *args)
    finally:
        dist.destroy_process_group()
