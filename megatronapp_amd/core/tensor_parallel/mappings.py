"""Autograd-aware TP/SP collective mappings over RCCL.

Equivalent of reference mappings.py:354-620 (copy/reduce/scatter/gather
regions, sequence-parallel all-gather / reduce-scatter, all-to-all).  Each
op is a thin autograd.Function over one RCCL collective on the TP group;
on MI355X the TP all-reduce is latency-bound over xGMI so there is no
algorithm choice to make here — bucket/overlap decisions live in the
linear layers (layers.py) and DDP.

MegaScan scope instrumentation wraps each collective (reference
mappings.py:35,113,163,213,287) via the global tracer when tracing is on.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from .. import parallel_state
from ..trace_hooks import trace_collective


def _tp_group():
    return parallel_state.get_tensor_model_parallel_group()


def _tp_world():
    return parallel_state.get_tensor_model_parallel_world_size()


def _tp_rank():
    return parallel_state.get_tensor_model_parallel_rank()


# ---------------------------------------------------------------------------
# primitive collectives
# ---------------------------------------------------------------------------

def _reduce(input_: torch.Tensor) -> torch.Tensor:
    if _tp_world() == 1:
        return input_
    with trace_collective("_reduce", input_, parallel_state.get_tensor_model_parallel_global_ranks()):
        dist.all_reduce(input_.contiguous(), group=_tp_group())
    return input_


def _split_along_last_dim(input_: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return input_
    last = input_.size(-1)
    assert last % world == 0
    chunks = torch.split(input_, last // world, dim=-1)
    return chunks[_tp_rank()].contiguous()


def _split_along_first_dim(input_: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return input_
    first = input_.size(0)
    assert first % world == 0
    sz = first // world
    return input_[_tp_rank() * sz:(_tp_rank() + 1) * sz].contiguous()


def _gather_along_last_dim(input_: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return input_
    input_ = input_.contiguous()
    shape = list(input_.shape)
    shape[0] *= world
    with trace_collective("_gather_along_last_dim", input_,
                          parallel_state.get_tensor_model_parallel_global_ranks()):
        out = torch.empty(shape, dtype=input_.dtype, device=input_.device)
        dist.all_gather_into_tensor(out, input_, group=_tp_group())
    return torch.cat(out.chunk(world, dim=0), dim=-1)


def _gather_along_first_dim(input_: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return input_
    input_ = input_.contiguous()
    shape = list(input_.shape)
    shape[0] *= world
    out = torch.empty(shape, dtype=input_.dtype, device=input_.device)
    with trace_collective("_gather_along_first_dim", input_,
                          parallel_state.get_tensor_model_parallel_global_ranks()):
        dist.all_gather_into_tensor(out, input_, group=_tp_group())
    return out


def _reduce_scatter_along_first_dim(input_: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return input_
    input_ = input_.contiguous()
    shape = list(input_.shape)
    assert shape[0] % world == 0
    shape[0] //= world
    out = torch.empty(shape, dtype=input_.dtype, device=input_.device)
    with trace_collective("_reduce_scatter_along_first_dim", input_,
                          parallel_state.get_tensor_model_parallel_global_ranks()):
        dist.reduce_scatter_tensor(out, input_, group=_tp_group())
    return out


def _reduce_scatter_along_last_dim(input_: torch.Tensor) -> torch.Tensor:
    world = _tp_world()
    if world == 1:
        return input_
    # transpose-free: chunk last dim, stack on first, reduce-scatter
    chunks = torch.cat([c.contiguous() for c in input_.chunk(world, dim=-1)], dim=0)
    return _reduce_scatter_along_first_dim(chunks)


# ---------------------------------------------------------------------------
# autograd wrappers
# ---------------------------------------------------------------------------

class _CopyToModelParallelRegion(torch.autograd.Function):
    """Identity forward; all-reduce backward (column-parallel input)."""

    @staticmethod
    def forward(ctx, input_):
        return input_

    @staticmethod
    def backward(ctx, grad_output):
        return _reduce(grad_output)


class _ReduceFromModelParallelRegion(torch.autograd.Function):
    """All-reduce forward; identity backward (row-parallel output)."""

    @staticmethod
    def forward(ctx, input_):
        return _reduce(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return grad_output


class _ScatterToModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _split_along_last_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _gather_along_last_dim(grad_output)


class _GatherFromModelParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _gather_along_last_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _split_along_last_dim(grad_output)


class _ScatterToSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _split_along_first_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _gather_along_first_dim(grad_output)


class _GatherFromSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_, tensor_parallel_output_grad=True):
        ctx.tensor_parallel_output_grad = tensor_parallel_output_grad
        return _gather_along_first_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        if ctx.tensor_parallel_output_grad:
            return _reduce_scatter_along_first_dim(grad_output), None
        return _split_along_first_dim(grad_output), None


class _ReduceScatterToSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_):
        return _reduce_scatter_along_first_dim(input_)

    @staticmethod
    def backward(ctx, grad_output):
        return _gather_along_first_dim(grad_output)


class _AllToAll(torch.autograd.Function):
    """all-to-all on an arbitrary group (EP token exchange, Ulysses CP)."""

    @staticmethod
    def forward(ctx, group, input_, output_split_sizes, input_split_sizes):
        ctx.group = group
        ctx.output_split_sizes = output_split_sizes
        ctx.input_split_sizes = input_split_sizes
        world = dist.get_world_size(group=group)
        if world == 1:
            return input_
        input_ = input_.contiguous()
        if output_split_sizes is None:
            output = torch.empty_like(input_)
        else:
            shape = list(input_.shape)
            shape[0] = sum(output_split_sizes)
            output = torch.empty(shape, dtype=input_.dtype, device=input_.device)
        dist.all_to_all_single(output, input_,
                               output_split_sizes=output_split_sizes,
                               input_split_sizes=input_split_sizes, group=group)
        return output

    @staticmethod
    def backward(ctx, grad_output):
        return (None,
                _AllToAll.apply(ctx.group, grad_output,
                                ctx.input_split_sizes, ctx.output_split_sizes),
                None, None)


# public API (reference names)

def copy_to_tensor_model_parallel_region(input_):
    return _CopyToModelParallelRegion.apply(input_)


def reduce_from_tensor_model_parallel_region(input_):
    return _ReduceFromModelParallelRegion.apply(input_)


def scatter_to_tensor_model_parallel_region(input_):
    return _ScatterToModelParallelRegion.apply(input_)


def gather_from_tensor_model_parallel_region(input_):
    return _GatherFromModelParallelRegion.apply(input_)


def scatter_to_sequence_parallel_region(input_):
    return _ScatterToSequenceParallelRegion.apply(input_)


def gather_from_sequence_parallel_region(input_, tensor_parallel_output_grad=True):
    return _GatherFromSequenceParallelRegion.apply(input_, tensor_parallel_output_grad)


def reduce_scatter_to_sequence_parallel_region(input_):
    return _ReduceScatterToSequenceParallelRegion.apply(input_)


def all_to_all(group, input_, output_split_sizes=None, input_split_sizes=None):
    return _AllToAll.apply(group, input_, output_split_sizes, input_split_sizes)


def all_to_all_sp2hp(input_):
    """[s/tp, b, h] -> [s, b, h/tp] over the TP group (Ulysses transpose)."""
    world = _tp_world()
    if world == 1:
        return input_
    s, b, h = input_.shape
    inp = input_.reshape(-1, h)
    split = torch.cat(inp.chunk(world, dim=-1), dim=0).contiguous()
    out = all_to_all(_tp_group(), split)
    return out.reshape(s * world, b, h // world)


def all_to_all_hp2sp(input_):
    world = _tp_world()
    if world == 1:
        return input_
    s, b, h = input_.shape
    inp = input_.reshape(-1, h)
    out = all_to_all(_tp_group(), inp)
    out = torch.cat(out.chunk(world, dim=0), dim=-1)
    return out.reshape(s // world, b, h * world)
