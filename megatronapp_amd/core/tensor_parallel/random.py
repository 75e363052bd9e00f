"""Parallel RNG state tracking + activation checkpointing.

Reference: tensor_parallel/random.py:124 (CudaRNGStatesTracker), :461
(checkpoint).  TP ranks need two RNG streams: one identical across the TP
group (dropout on replicated activations) and one different per rank
(dropout on sharded activations, param init of sharded weights).  On ROCm
``torch.cuda.*_rng_state`` drives the Philox generator exactly like CUDA.
"""

from __future__ import annotations

import contextlib

import torch

from .. import parallel_state

_MODEL_PARALLEL_RNG_TRACKER_NAME = "model-parallel-rng"
_EXPERT_PARALLEL_RNG_TRACKER_NAME = "expert-parallel-rng"


def _device_rng_supported() -> bool:
    return torch.cuda.is_available()


class CudaRNGStatesTracker:
    def __init__(self):
        self.reset()

    def reset(self):
        self.states_ = {}
        self.seeds_ = set()

    def get_states(self):
        return dict(self.states_)

    def set_states(self, states):
        self.states_ = states

    def add(self, name, seed):
        if seed in self.seeds_:
            raise RuntimeError(f"seed {seed} already present")
        self.seeds_.add(seed)
        if name in self.states_:
            raise RuntimeError(f"rng state {name} already present")
        if not _device_rng_supported():
            # CPU fallback: track torch CPU generator states
            orig = torch.get_rng_state()
            torch.manual_seed(seed)
            self.states_[name] = torch.get_rng_state()
            torch.set_rng_state(orig)
            return
        orig = torch.cuda.get_rng_state()
        torch.cuda.manual_seed(seed)
        self.states_[name] = torch.cuda.get_rng_state()
        torch.cuda.set_rng_state(orig)

    @contextlib.contextmanager
    def fork(self, name=_MODEL_PARALLEL_RNG_TRACKER_NAME):
        if name not in self.states_:
            raise RuntimeError(f"rng state {name} not added")
        if not _device_rng_supported():
            orig = torch.get_rng_state()
            torch.set_rng_state(self.states_[name])
            try:
                yield
            finally:
                self.states_[name] = torch.get_rng_state()
                torch.set_rng_state(orig)
            return
        orig = torch.cuda.get_rng_state()
        torch.cuda.set_rng_state(self.states_[name])
        try:
            yield
        finally:
            self.states_[name] = torch.cuda.get_rng_state()
            torch.cuda.set_rng_state(orig)


_CUDA_RNG_STATE_TRACKER = CudaRNGStatesTracker()


def get_cuda_rng_tracker():
    return _CUDA_RNG_STATE_TRACKER


def get_expert_parallel_rng_tracker_name():
    return _EXPERT_PARALLEL_RNG_TRACKER_NAME


def model_parallel_cuda_manual_seed(seed: int) -> None:
    """Seed layout (reference random.py): data-parallel-identical default
    generator; TP-rank-offset model-parallel stream; EP-offset expert stream."""
    tp_rank = parallel_state.get_tensor_model_parallel_rank()
    pp_rank = parallel_state.get_pipeline_model_parallel_rank()
    offset = seed + 2718
    tp_seed = offset + tp_rank + pp_rank * 1024
    ep_seed = seed + 1007 + tp_rank + parallel_state.get_expert_model_parallel_rank() * 4096

    _CUDA_RNG_STATE_TRACKER.reset()
    if _device_rng_supported():
        torch.cuda.manual_seed(seed)
    torch.manual_seed(seed)
    _CUDA_RNG_STATE_TRACKER.add(_MODEL_PARALLEL_RNG_TRACKER_NAME, tp_seed)
    _CUDA_RNG_STATE_TRACKER.add(_EXPERT_PARALLEL_RNG_TRACKER_NAME, ep_seed)


class CheckpointFunction(torch.autograd.Function):
    """Activation checkpointing with TP-RNG restore (reference random.py:461)."""

    @staticmethod
    def forward(ctx, run_function, distribute_saved_activations, *args):
        ctx.run_function = run_function
        ctx.fwd_cpu_rng_state = torch.get_rng_state()
        ctx.had_device = _device_rng_supported()
        if ctx.had_device:
            ctx.fwd_device_rng_state = torch.cuda.get_rng_state()
        ctx.fwd_tracker_states = _CUDA_RNG_STATE_TRACKER.get_states()
        with torch.no_grad():
            outputs = run_function(*args)
        ctx.save_for_backward(*[a for a in args if isinstance(a, torch.Tensor)])
        ctx.arg_is_tensor = [isinstance(a, torch.Tensor) for a in args]
        ctx.non_tensor_args = [a for a in args if not isinstance(a, torch.Tensor)]
        return outputs

    @staticmethod
    def backward(ctx, *grad_outputs):
        tensors = list(ctx.saved_tensors)
        non_tensors = list(ctx.non_tensor_args)
        args = []
        for is_t in ctx.arg_is_tensor:
            args.append(tensors.pop(0) if is_t else non_tensors.pop(0))
        detached = [a.detach().requires_grad_(a.requires_grad)
                    if isinstance(a, torch.Tensor) else a for a in args]

        # restore RNG to forward-time state, rerun, restore current state
        cpu_state = torch.get_rng_state()
        torch.set_rng_state(ctx.fwd_cpu_rng_state)
        if ctx.had_device:
            device_state = torch.cuda.get_rng_state()
            torch.cuda.set_rng_state(ctx.fwd_device_rng_state)
        tracker_states = _CUDA_RNG_STATE_TRACKER.get_states()
        _CUDA_RNG_STATE_TRACKER.set_states(ctx.fwd_tracker_states)

        with torch.enable_grad():
            outputs = ctx.run_function(*detached)

        torch.set_rng_state(cpu_state)
        if ctx.had_device:
            torch.cuda.set_rng_state(device_state)
        _CUDA_RNG_STATE_TRACKER.set_states(tracker_states)

        if isinstance(outputs, torch.Tensor):
            outputs = (outputs,)
        out_tensors = [o for o in outputs if isinstance(o, torch.Tensor) and o.requires_grad]
        grads = [g for o, g in zip(outputs, grad_outputs)
                 if isinstance(o, torch.Tensor) and o.requires_grad]
        torch.autograd.backward(out_tensors, grads)
        input_grads = tuple(a.grad if isinstance(a, torch.Tensor) else None
                            for a in detached)
        return (None, None) + input_grads


def checkpoint(function, distribute_saved_activations, *args):
    return CheckpointFunction.apply(function, distribute_saved_activations, *args)
