"""DPP transport: policy-ordered sender over the shm mailbox channel."""

from __future__ import annotations

import os
import queue
import threading
from typing import Optional, Tuple

import torch

from ..core import parallel_state

_C = None
_TRANSPORT: Optional["DPPTransport"] = None


def build_dpp_extension(verbose=False):
    """Build megatronapp_amd/dpp/_C_dpp.so in-tree."""
    import shutil
    from torch.utils.cpp_extension import load
    here = os.path.dirname(os.path.abspath(__file__))
    build_dir = os.path.join(here, "build")
    os.makedirs(build_dir, exist_ok=True)
    load(name="_C_dpp",
         sources=[os.path.join(here, "csrc", "shm_tensor.cpp")],
         build_directory=build_dir,
         extra_cflags=["-O3", "-I/opt/rocm/include",
                       "-D__HIP_PLATFORM_AMD__=1"],
         extra_ldflags=["-L/opt/rocm/lib", "-lamdhip64"],
         verbose=verbose, is_python_module=True)
    shutil.copyfile(os.path.join(build_dir, "_C_dpp.so"),
                    os.path.join(here, "_C_dpp.so"))
    return os.path.join(here, "_C_dpp.so")


def _load():
    global _C
    if _C is not None:
        return _C
    import importlib.util
    so = os.path.join(os.path.dirname(__file__), "_C_dpp.so")
    if not os.path.exists(so):
        raise RuntimeError(
            "DPP extension not built; run python -c "
            "'from megatronapp_amd.dpp import build_dpp_extension; "
            "build_dpp_extension()'")
    spec = importlib.util.spec_from_file_location(
        "megatronapp_amd.dpp._C_dpp", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _C = mod
    return _C


class _SendQueue:
    """Pending sends, drained by a worker thread in policy order."""

    def __init__(self, policy: str):
        self.policy = policy
        self.items = []  # (chunk, mb, direction, src, dst, tensor)
        self.cond = threading.Condition()
        self.closed = False

    def push(self, item):
        with self.cond:
            self.items.append(item)
            self.cond.notify()

    def pop(self):
        with self.cond:
            while not self.items and not self.closed:
                self.cond.wait(timeout=0.5)
            if not self.items:
                return None
            if self.policy == "breadth_first":
                # lowest microbatch first (then chunk): round-robin chunks
                key = min(range(len(self.items)),
                          key=lambda i: (self.items[i][1], self.items[i][0]))
            elif self.policy == "depth_first":
                # model-chunk-major greedy order — the reference C++ sender
                # policy (shm_tensor_new_rdma.cpp:1478-1525)
                key = min(range(len(self.items)),
                          key=lambda i: (self.items[i][0], self.items[i][1]))
            else:  # greedy FIFO
                key = 0
            return self.items.pop(key)

    def close(self):
        with self.cond:
            self.closed = True
            self.cond.notify_all()


class DPPTransport:
    """Pipeline p2p over the shm channel with tagged slots."""

    def __init__(self, slot_bytes: int, nslots: int = 8,
                 policy: str = "depth_first"):
        self.c = _load()
        self.rank = torch.distributed.get_rank()
        self.prev = parallel_state.get_pipeline_model_parallel_prev_rank()
        self.next = parallel_state.get_pipeline_model_parallel_next_rank()
        self.pp_rank = parallel_state.get_pipeline_model_parallel_rank()
        self.pp_world = parallel_state.get_pipeline_model_parallel_world_size()
        self.slot_bytes = slot_bytes
        self.policy = policy

        # channels: receiver creates; sender opens.
        # fwd: prev -> me, me -> next;  bwd: next -> me, me -> prev
        if self.pp_world > 1:
            self.c.init_channel("fwd", self.prev, self.rank, slot_bytes,
                                nslots, True)
            self.c.init_channel("bwd", self.next, self.rank, slot_bytes,
                                nslots, True)
            torch.distributed.barrier(
                group=parallel_state.get_pipeline_model_parallel_group())
            self.c.init_channel("fwd", self.rank, self.next, slot_bytes,
                                nslots, False)
            self.c.init_channel("bwd", self.rank, self.prev, slot_bytes,
                                nslots, False)

        self.queue = _SendQueue(policy)
        self.workers = [threading.Thread(target=self._sender_loop, daemon=True)
                        for _ in range(2)]
        for w in self.workers:
            w.start()

    # ------------------------------------------------------------- sending
    def _sender_loop(self):
        while True:
            item = self.queue.pop()
            if item is None:
                return
            chunk, mb, direction, src, dst, tensor = item
            self.c.put_tensor(direction, src, dst, chunk, mb, tensor)

    def send_forward(self, tensor: torch.Tensor, chunk: int, mb: int):
        self.queue.push((chunk, mb, "fwd", self.rank, self.next,
                         tensor.detach().contiguous()))

    def send_backward(self, tensor: torch.Tensor, chunk: int, mb: int):
        self.queue.push((chunk, mb, "bwd", self.rank, self.prev,
                         tensor.detach().contiguous()))

    # ------------------------------------------------------------ receiving
    def recv_forward(self, shape, dtype, chunk: int, mb: int) -> torch.Tensor:
        device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
        out = torch.empty(shape, dtype=dtype, device=device)
        self.c.get_tensor("fwd", self.prev, self.rank, chunk, mb, out)
        out.requires_grad_(True)
        return out

    def recv_backward(self, shape, dtype, chunk: int, mb: int) -> torch.Tensor:
        device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
        out = torch.empty(shape, dtype=dtype, device=device)
        self.c.get_tensor("bwd", self.next, self.rank, chunk, mb, out)
        return out

    def shutdown(self):
        self.queue.close()
        for w in self.workers:
            w.join(timeout=5)
        self.c.clean_channels()


def initialize_dpp(args, config) -> Optional[DPPTransport]:
    """Create the transport from args (--use-dpp)."""
    global _TRANSPORT
    if _TRANSPORT is not None:
        return _TRANSPORT
    h = config.hidden_size
    # slot = one [s, b, h] activation in pipeline dtype (reference sizing:
    # shm_tensor_new_rdma.cpp:130-135)
    elem = 2 if config.pipeline_dtype in (torch.bfloat16, torch.float16) else 4
    s = args.seq_length // config.context_parallel_size
    if config.sequence_parallel:
        s //= config.tensor_model_parallel_size
    slot_bytes = s * args.micro_batch_size * h * elem
    _TRANSPORT = DPPTransport(slot_bytes, nslots=8, policy=args.dpp_policy)
    return _TRANSPORT


def get_transport() -> Optional[DPPTransport]:
    return _TRANSPORT


def shutdown_dpp():
    global _TRANSPORT
    if _TRANSPORT is not None:
        _TRANSPORT.shutdown()
        _TRANSPORT = None
