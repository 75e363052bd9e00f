"""Pipeline-parallel point-to-point communication over RCCL/xGMI.

Reference: pipeline_parallel/p2p_communication.py (_communicate :303,
batched :125 vs per-op :191 ordering, shape exchange :27).

MI355X notes: intra-node PP neighbours are one xGMI hop apart
(7 p2p links × ≈153 GB/s); batch_isend_irecv maps to RCCL grouped
send/recv which rides a single link per peer — a [s,b,h] bf16 activation
(8 MB at s2048·b2·h2048) moves in ~60 µs, so the schedule overlaps these
behind compute via separate streams when overlap_p2p_comm is set.

MegaScan wraps each op with byte counts so the aggregator can compute
link bandwidth (reference p2p_communication.py:469-487).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from .. import parallel_state
from ..trace_hooks import trace_collective, trace_scope

# ---------------------------------------------------------------------------
# MegaDPP integration: when a transport is active, pipeline sends go through
# the tagged shm mailbox (sender-side policy ordering) instead of RCCL p2p.
# The schedules set the (chunk, microbatch) tags before each call.
# ---------------------------------------------------------------------------
import threading as _threading

_DPP_TAGS = _threading.local()


def set_dpp_tags(fwd_send=None, fwd_recv=None, bwd_send=None, bwd_recv=None):
    _DPP_TAGS.fwd_send = fwd_send
    _DPP_TAGS.fwd_recv = fwd_recv
    _DPP_TAGS.bwd_send = bwd_send
    _DPP_TAGS.bwd_recv = bwd_recv


def _tag(name):
    return getattr(_DPP_TAGS, name, None)


def _dpp():
    from ...dpp.transport import get_transport
    return get_transport()


def _dpp_dtype(config):
    return config.pipeline_dtype or torch.float32


def _shape_numel(shape):
    n = 1
    for s in shape:
        n *= s
    return n


def _communicate_shapes(tensor_send_next, tensor_send_prev, recv_prev, recv_next):
    """Exchange tensor shapes with PP neighbours (variable seq lengths)."""
    device = "cuda" if torch.cuda.is_available() else "cpu"
    ops = []
    recv_prev_shape = torch.empty(3, dtype=torch.int64, device=device)
    recv_next_shape = torch.empty(3, dtype=torch.int64, device=device)
    if tensor_send_prev is not None:
        ops.append(dist.P2POp(dist.isend,
                              torch.tensor(tensor_send_prev.shape, dtype=torch.int64,
                                           device=device),
                              parallel_state.get_pipeline_model_parallel_prev_rank()))
    if recv_prev:
        ops.append(dist.P2POp(dist.irecv, recv_prev_shape,
                              parallel_state.get_pipeline_model_parallel_prev_rank()))
    if tensor_send_next is not None:
        ops.append(dist.P2POp(dist.isend,
                              torch.tensor(tensor_send_next.shape, dtype=torch.int64,
                                           device=device),
                              parallel_state.get_pipeline_model_parallel_next_rank()))
    if recv_next:
        ops.append(dist.P2POp(dist.irecv, recv_next_shape,
                              parallel_state.get_pipeline_model_parallel_next_rank()))
    if ops:
        for req in dist.batch_isend_irecv(ops):
            req.wait()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return (tuple(recv_prev_shape.tolist()) if recv_prev else None,
            tuple(recv_next_shape.tolist()) if recv_next else None)


def _p2p_ops_even_odd(tensor_send_prev, tensor_recv_prev, tensor_send_next,
                      tensor_recv_next, fwd_group, bwd_group):
    """Per-op isend/irecv with even/odd rank ordering (deadlock-free without
    batching; reference _p2p_ops :191-216)."""
    reqs = []
    rank = parallel_state.get_pipeline_model_parallel_rank()
    even = (rank % 2 == 0)
    next_rank = parallel_state.get_pipeline_model_parallel_next_rank()
    prev_rank = parallel_state.get_pipeline_model_parallel_prev_rank()

    def send_next():
        if tensor_send_next is not None:
            reqs.append(dist.isend(tensor_send_next, next_rank, group=fwd_group))

    def recv_prev():
        if tensor_recv_prev is not None:
            reqs.append(dist.irecv(tensor_recv_prev, prev_rank, group=fwd_group))

    def send_prev():
        if tensor_send_prev is not None:
            reqs.append(dist.isend(tensor_send_prev, prev_rank, group=bwd_group))

    def recv_next():
        if tensor_recv_next is not None:
            reqs.append(dist.irecv(tensor_recv_next, next_rank, group=bwd_group))

    if even:
        send_next(); recv_prev(); send_prev(); recv_next()
    else:
        recv_prev(); send_next(); recv_next(); send_prev()
    return reqs


def _communicate(*, tensor_send_next: Optional[torch.Tensor],
                 tensor_send_prev: Optional[torch.Tensor],
                 recv_prev: bool, recv_next: bool, tensor_shape, config,
                 wait_on_reqs: bool = True):
    tensor_recv_prev = None
    tensor_recv_next = None

    if not config.variable_seq_lengths:
        recv_prev_shape = tensor_shape
        recv_next_shape = tensor_shape
    else:
        recv_prev_shape, recv_next_shape = _communicate_shapes(
            tensor_send_next, tensor_send_prev, recv_prev, recv_next)

    dtype = config.pipeline_dtype or torch.float32
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    if recv_prev:
        tensor_recv_prev = torch.empty(recv_prev_shape, device=device,
                                       dtype=dtype, requires_grad=True)
    if recv_next:
        tensor_recv_next = torch.empty(recv_next_shape, device=device,
                                       dtype=dtype, requires_grad=True)

    # Direction-split channels (see parallel_state): activation traffic
    # (send→next / recv←prev) on the FWD group, gradient traffic
    # (send→prev / recv←next) on the BWD group — mandatory at PP=2 where
    # next == prev would otherwise alias both flows onto one channel.
    fwd_group = parallel_state.get_pipeline_forward_group()
    bwd_group = parallel_state.get_pipeline_backward_group()
    if config.batch_p2p_comm:
        fwd_ops, bwd_ops = [], []
        if tensor_send_prev is not None:
            bwd_ops.append(dist.P2POp(dist.isend, tensor_send_prev.contiguous(),
                                      parallel_state.get_pipeline_model_parallel_prev_rank(),
                                      bwd_group))
        if tensor_recv_prev is not None:
            fwd_ops.append(dist.P2POp(dist.irecv, tensor_recv_prev,
                                      parallel_state.get_pipeline_model_parallel_prev_rank(),
                                      fwd_group))
        if tensor_send_next is not None:
            fwd_ops.append(dist.P2POp(dist.isend, tensor_send_next.contiguous(),
                                      parallel_state.get_pipeline_model_parallel_next_rank(),
                                      fwd_group))
        if tensor_recv_next is not None:
            bwd_ops.append(dist.P2POp(dist.irecv, tensor_recv_next,
                                      parallel_state.get_pipeline_model_parallel_next_rank(),
                                      bwd_group))
        reqs = []
        if fwd_ops:
            reqs.extend(dist.batch_isend_irecv(fwd_ops))
        if bwd_ops:
            reqs.extend(dist.batch_isend_irecv(bwd_ops))
    else:
        reqs = _p2p_ops_even_odd(
            tensor_send_prev.contiguous() if tensor_send_prev is not None else None,
            tensor_recv_prev,
            tensor_send_next.contiguous() if tensor_send_next is not None else None,
            tensor_recv_next, fwd_group, bwd_group)

    if wait_on_reqs and reqs:
        for req in reqs:
            req.wait()
        reqs = None
    return tensor_recv_prev, tensor_recv_next, reqs


def _tensor_bytes(shape, config):
    dtype = config.pipeline_dtype or torch.float32
    return _shape_numel(shape) * torch.tensor([], dtype=dtype).element_size()


def _p2p_peers():
    return [dist.get_rank(),
            parallel_state.get_pipeline_model_parallel_prev_rank(),
            parallel_state.get_pipeline_model_parallel_next_rank()]


def recv_forward(tensor_shape, config, is_first_stage: bool):
    if is_first_stage:
        return None
    dpp = _dpp()
    if dpp is not None and _tag("fwd_recv") is not None:
        with trace_scope("recv-forward"):
            return dpp.recv_forward(tensor_shape, _dpp_dtype(config),
                                    *_tag("fwd_recv"))
    with trace_scope("recv-forward"):
        input_tensor, _, _ = _communicate(
            tensor_send_next=None, tensor_send_prev=None, recv_prev=True,
            recv_next=False, tensor_shape=tensor_shape, config=config)
    return input_tensor


def recv_backward(tensor_shape, config, is_last_stage: bool):
    if is_last_stage:
        return None
    dpp = _dpp()
    if dpp is not None and _tag("bwd_recv") is not None:
        with trace_scope("recv-backward"):
            return dpp.recv_backward(tensor_shape, _dpp_dtype(config),
                                     *_tag("bwd_recv"))
    with trace_scope("recv-backward"):
        _, output_tensor_grad, _ = _communicate(
            tensor_send_next=None, tensor_send_prev=None, recv_prev=False,
            recv_next=True, tensor_shape=tensor_shape, config=config)
    return output_tensor_grad


def send_forward(output_tensor, config, is_last_stage: bool):
    if is_last_stage:
        return
    dpp = _dpp()
    if dpp is not None and _tag("fwd_send") is not None:
        with trace_scope("send-forward"):
            dpp.send_forward(output_tensor, *_tag("fwd_send"))
        return
    with trace_scope("send-forward", data=output_tensor.numel() * output_tensor.element_size(),
                     group=_p2p_peers()):
        _communicate(tensor_send_next=output_tensor, tensor_send_prev=None,
                     recv_prev=False, recv_next=False, tensor_shape=None,
                     config=config)


def send_backward(input_tensor_grad, config, is_first_stage: bool):
    if is_first_stage:
        return
    dpp = _dpp()
    if dpp is not None and _tag("bwd_send") is not None:
        with trace_scope("send-backward"):
            dpp.send_backward(input_tensor_grad, *_tag("bwd_send"))
        return
    with trace_scope("send-backward", data=input_tensor_grad.numel() * input_tensor_grad.element_size(),
                     group=_p2p_peers()):
        _communicate(tensor_send_next=None, tensor_send_prev=input_tensor_grad,
                     recv_prev=False, recv_next=False, tensor_shape=None,
                     config=config)


def send_forward_recv_backward(output_tensor, tensor_shape, config,
                               is_last_stage: bool):
    if is_last_stage:
        return None
    dpp = _dpp()
    if dpp is not None and _tag("fwd_send") is not None:
        dpp.send_forward(output_tensor, *_tag("fwd_send"))
        return dpp.recv_backward(tensor_shape, _dpp_dtype(config),
                                 *_tag("bwd_recv"))
    with trace_scope("send-forward-recv-backward",
                     data=output_tensor.numel() * output_tensor.element_size(),
                     group=_p2p_peers()):
        _, output_tensor_grad, _ = _communicate(
            tensor_send_next=output_tensor, tensor_send_prev=None,
            recv_prev=False, recv_next=True, tensor_shape=tensor_shape,
            config=config)
    return output_tensor_grad


def send_backward_recv_forward(input_tensor_grad, tensor_shape, config,
                               is_first_stage: bool):
    if is_first_stage:
        return None
    dpp = _dpp()
    if dpp is not None and _tag("bwd_send") is not None:
        dpp.send_backward(input_tensor_grad, *_tag("bwd_send"))
        if _tag("fwd_recv") is None:
            return None
        return dpp.recv_forward(tensor_shape, _dpp_dtype(config),
                                *_tag("fwd_recv"))
    with trace_scope("send-backward-recv-forward",
                     data=input_tensor_grad.numel() * input_tensor_grad.element_size(),
                     group=_p2p_peers()):
        input_tensor, _, _ = _communicate(
            tensor_send_next=None, tensor_send_prev=input_tensor_grad,
            recv_prev=True, recv_next=False, tensor_shape=tensor_shape,
            config=config)
    return input_tensor


def send_forward_recv_forward(output_tensor, recv_prev, tensor_shape, config,
                              overlap_p2p_comm=False):
    dpp = _dpp()
    if dpp is not None and (_tag("fwd_send") is not None or
                            (recv_prev and _tag("fwd_recv") is not None)):
        if output_tensor is not None and _tag("fwd_send") is not None:
            dpp.send_forward(output_tensor, *_tag("fwd_send"))
        result = None
        if recv_prev and _tag("fwd_recv") is not None:
            result = dpp.recv_forward(tensor_shape, _dpp_dtype(config),
                                      *_tag("fwd_recv"))
        if overlap_p2p_comm:
            return result, None
        return result
    with trace_scope("exchange-next"):
        input_tensor, _, wait_handles = _communicate(
            tensor_send_next=output_tensor, tensor_send_prev=None,
            recv_prev=recv_prev, recv_next=False, tensor_shape=tensor_shape,
            config=config, wait_on_reqs=(not overlap_p2p_comm))
    if overlap_p2p_comm:
        return input_tensor, wait_handles
    return input_tensor


def send_backward_recv_backward(input_tensor_grad, recv_next, tensor_shape,
                                config, overlap_p2p_comm=False):
    dpp = _dpp()
    if dpp is not None and (_tag("bwd_send") is not None or
                            (recv_next and _tag("bwd_recv") is not None)):
        if input_tensor_grad is not None and _tag("bwd_send") is not None:
            dpp.send_backward(input_tensor_grad, *_tag("bwd_send"))
        result = None
        if recv_next and _tag("bwd_recv") is not None:
            result = dpp.recv_backward(tensor_shape, _dpp_dtype(config),
                                       *_tag("bwd_recv"))
        if overlap_p2p_comm:
            return result, None
        return result
    with trace_scope("exchange-prev"):
        _, output_tensor_grad, wait_handles = _communicate(
            tensor_send_next=None, tensor_send_prev=input_tensor_grad,
            recv_prev=False, recv_next=recv_next, tensor_shape=tensor_shape,
            config=config, wait_on_reqs=(not overlap_p2p_comm))
    if overlap_p2p_comm:
        return output_tensor_grad, wait_handles
    return output_tensor_grad


def send_forward_backward_recv_forward_backward(
        output_tensor, input_tensor_grad, recv_prev, recv_next, tensor_shape,
        config):
    dpp = _dpp()
    if dpp is not None and _tag("fwd_send") is not None or \
            dpp is not None and _tag("bwd_send") is not None:
        if output_tensor is not None and _tag("fwd_send") is not None:
            dpp.send_forward(output_tensor, *_tag("fwd_send"))
        if input_tensor_grad is not None and _tag("bwd_send") is not None:
            dpp.send_backward(input_tensor_grad, *_tag("bwd_send"))
        it = ot = None
        if recv_prev and _tag("fwd_recv") is not None:
            it = dpp.recv_forward(tensor_shape, _dpp_dtype(config),
                                  *_tag("fwd_recv"))
        if recv_next and _tag("bwd_recv") is not None:
            ot = dpp.recv_backward(tensor_shape, _dpp_dtype(config),
                                   *_tag("bwd_recv"))
        return it, ot
    input_tensor, output_tensor_grad, _ = _communicate(
        tensor_send_next=output_tensor, tensor_send_prev=input_tensor_grad,
        recv_prev=recv_prev, recv_next=recv_next, tensor_shape=tensor_shape,
        config=config)
    return input_tensor, output_tensor_grad
