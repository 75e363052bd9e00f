"""Interleaved (virtual-pipeline) 1F1B schedule.

Reference: schedules.py:856-1780 (schedule table :800, depth-vs-breadth
knob ``microbatch_group_size_per_vp_stage``).

MegaDPP hook: ``megatronapp_amd.dpp`` can register a send-ordering policy
consulted at each send point (greedy model-chunk-major ordering is the
reference C++ sender-thread policy, shm_tensor_new_rdma.cpp:1478-1525);
with no policy registered this is the stock interleaved schedule.
"""

from __future__ import annotations

import contextlib
from typing import List

import torch

from .. import parallel_state
from ..enums import ModelType
from ..utils import get_model_config
from . import p2p_communication
from .schedules import backward_step, deallocate_output_tensor, forward_step


def forward_backward_pipelining_with_interleaving(
        *, forward_step_func, data_iterator, model: List, num_microbatches: int,
        seq_length: int, micro_batch_size: int, decoder_seq_length: int = None,
        forward_only: bool = False, collect_non_loss_data: bool = False,
        first_val_step: bool = None, adjust_tensor_shapes_fn=None):
    assert isinstance(model, list) and len(model) > 1, \
        "interleaved schedule requires multiple model chunks"
    assert isinstance(data_iterator, list)
    config = get_model_config(model[0])
    model_type = ModelType.encoder_or_decoder

    pp_size = parallel_state.get_pipeline_model_parallel_world_size()
    pp_rank = parallel_state.get_pipeline_model_parallel_rank()
    num_model_chunks = len(model)
    total_num_microbatches = num_microbatches * num_model_chunks

    group_size = config.microbatch_group_size_per_vp_stage or pp_size
    assert num_microbatches % group_size == 0, (
        f"num_microbatches {num_microbatches} must be divisible by "
        f"microbatch group size {group_size}")

    h = config.hidden_size
    s = seq_length // config.context_parallel_size
    if config.sequence_parallel:
        s //= config.tensor_model_parallel_size
    tensor_shape = (s, micro_batch_size, h)
    if adjust_tensor_shapes_fn is not None:
        tensor_shape = adjust_tensor_shapes_fn(tensor_shape)

    input_tensors = [[] for _ in range(num_model_chunks)]
    output_tensors = [[] for _ in range(num_model_chunks)]
    output_tensor_grads = [[] for _ in range(num_model_chunks)]
    forward_data_store: list = []
    total_num_tokens = torch.zeros(1, dtype=torch.int64,
                                   device="cuda" if torch.cuda.is_available() else "cpu")

    # grad sync held off for the whole schedule; buckets fire in finalize
    no_sync_ctxs = []
    for chunk in model:
        ctx = chunk.no_sync() if hasattr(chunk, "no_sync") else contextlib.nullcontext()
        ctx.__enter__()
        no_sync_ctxs.append(ctx)

    def get_model_chunk_id(microbatch_id: int, forward: bool) -> int:
        mb_in_group = microbatch_id % (group_size * num_model_chunks)
        chunk = mb_in_group // group_size
        if not forward:
            chunk = num_model_chunks - chunk - 1
        return chunk

    def is_first_microbatch_for_model_chunk(microbatch_id: int) -> bool:
        return microbatch_id < group_size

    # ---- MegaDPP tags: (chunk, within-chunk microbatch), with the
    # producer's chunk id on wrap-around hops (first rank's chunk c input
    # is the LAST rank's chunk c-1 output, and symmetrically for grads).
    def _wc(k: int) -> int:
        return (k // (group_size * num_model_chunks)) * group_size + \
            k % group_size

    def fwd_send_tag(k):
        return (get_model_chunk_id(k, True), _wc(k))

    def fwd_recv_tag(k):
        c = get_model_chunk_id(k, True)
        if pp_rank == 0:
            c -= 1  # wrap: produced by the last rank's previous chunk
        return (c, _wc(k))

    def bwd_send_tag(k):
        return (get_model_chunk_id(k, False), _wc(k))

    def bwd_recv_tag(k):
        c = get_model_chunk_id(k, False)
        if pp_rank == pp_size - 1:
            c += 1  # wrap: produced by rank 0's next chunk
        return (c, _wc(k))

    def forward_step_helper(microbatch_id: int):
        chunk = get_model_chunk_id(microbatch_id, forward=True)
        parallel_state.set_virtual_pipeline_model_parallel_rank(chunk)
        if parallel_state.is_pipeline_first_stage():
            if len(input_tensors[chunk]) == len(output_tensors[chunk]):
                input_tensors[chunk].append(None)
        input_tensor = input_tensors[chunk][-1]
        output_tensor, num_tokens = forward_step(
            forward_step_func, data_iterator[chunk], model[chunk],
            num_microbatches, input_tensor, forward_data_store, config,
            collect_non_loss_data,
            is_first_microbatch=is_first_microbatch_for_model_chunk(microbatch_id),
            current_microbatch=microbatch_id % num_microbatches, vp_stage=chunk)
        total_num_tokens.add_(num_tokens.item() if torch.is_tensor(num_tokens)
                              else num_tokens)
        output_tensors[chunk].append(output_tensor)
        if forward_only:
            input_tensors[chunk].pop()
            output_tensors[chunk].pop()
        return output_tensor

    def backward_step_helper(microbatch_id: int):
        chunk = get_model_chunk_id(microbatch_id, forward=False)
        parallel_state.set_virtual_pipeline_model_parallel_rank(chunk)
        if parallel_state.is_pipeline_last_stage():
            if len(output_tensor_grads[chunk]) == 0:
                output_tensor_grads[chunk].append(None)
        input_tensor = input_tensors[chunk].pop(0)
        output_tensor = output_tensors[chunk].pop(0)
        output_tensor_grad = output_tensor_grads[chunk].pop(0)
        return backward_step(input_tensor, output_tensor, output_tensor_grad,
                             model_type, config)

    # ---- warmup ----
    if num_microbatches == pp_size:
        num_warmup = total_num_microbatches
        all_warmup = True
    else:
        num_warmup = min((pp_size - pp_rank - 1) * 2 +
                         (num_model_chunks - 1) * group_size,
                         total_num_microbatches)
        all_warmup = num_warmup == total_num_microbatches
    num_remaining = total_num_microbatches - num_warmup

    parallel_state.set_virtual_pipeline_model_parallel_rank(0)
    p2p_communication.set_dpp_tags(fwd_recv=fwd_recv_tag(0))
    input_tensors[0].append(p2p_communication.recv_forward(
        tensor_shape, config, parallel_state.is_pipeline_first_stage()))

    for k in range(num_warmup):
        output_tensor = forward_step_helper(k)

        next_forward_chunk = get_model_chunk_id(k + 1, forward=True)
        recv_prev = True
        if parallel_state.is_pipeline_first_stage(ignore_virtual=True):
            if next_forward_chunk == 0:
                recv_prev = False
        if k == total_num_microbatches - 1:
            recv_prev = False

        # current chunk's virtual rank still set from forward_step_helper
        if parallel_state.is_pipeline_last_stage():
            output_tensor = None

        cur_chunk = get_model_chunk_id(k, forward=True)
        if (k == num_warmup - 1 and not forward_only and not all_warmup):
            input_tensor_grad = None
            recv_next = True
            if parallel_state.is_pipeline_last_stage(ignore_virtual=True):
                recv_next = False
            p2p_communication.set_dpp_tags(
                fwd_send=fwd_send_tag(k),
                fwd_recv=fwd_recv_tag(k + 1),
                bwd_recv=bwd_recv_tag(0))
            (input_tensor, output_tensor_grad) = (
                p2p_communication.send_forward_backward_recv_forward_backward(
                    output_tensor, input_tensor_grad, recv_prev=recv_prev,
                    recv_next=recv_next, tensor_shape=tensor_shape,
                    config=config))
            if recv_next:
                output_tensor_grads[num_model_chunks - 1].append(output_tensor_grad)
        else:
            p2p_communication.set_dpp_tags(
                fwd_send=fwd_send_tag(k),
                fwd_recv=fwd_recv_tag(k + 1))
            input_tensor = p2p_communication.send_forward_recv_forward(
                output_tensor, recv_prev, tensor_shape, config)
        if recv_prev:
            input_tensors[next_forward_chunk].append(input_tensor)
        deallocate_output_tensor(output_tensor, config.deallocate_pipeline_outputs)

    # ---- steady 1F1B ----
    for k in range(num_remaining):
        forward_k = k + num_warmup
        output_tensor = forward_step_helper(forward_k)
        forward_chunk = get_model_chunk_id(forward_k, forward=True)
        parallel_state.set_virtual_pipeline_model_parallel_rank(forward_chunk)
        if parallel_state.is_pipeline_last_stage():
            output_tensor = None

        backward_k = k
        input_tensor_grad = backward_step_helper(backward_k)
        backward_chunk = get_model_chunk_id(backward_k, forward=False)
        parallel_state.set_virtual_pipeline_model_parallel_rank(backward_chunk)
        if parallel_state.is_pipeline_first_stage():
            input_tensor_grad = None

        next_forward_chunk = get_model_chunk_id(forward_k + 1, forward=True)
        recv_prev = True
        if parallel_state.is_pipeline_first_stage(ignore_virtual=True) and \
                next_forward_chunk == 0:
            recv_prev = False
        next_backward_chunk = get_model_chunk_id(backward_k + 1, forward=False)
        recv_next = True
        if parallel_state.is_pipeline_last_stage(ignore_virtual=True) and \
                next_backward_chunk == num_model_chunks - 1:
            recv_next = False
        if forward_k == total_num_microbatches - 1:
            recv_prev = False

        p2p_communication.set_dpp_tags(
            fwd_send=fwd_send_tag(forward_k),
            bwd_send=bwd_send_tag(backward_k),
            fwd_recv=fwd_recv_tag(forward_k + 1),
            bwd_recv=bwd_recv_tag(backward_k + 1))
        (input_tensor, output_tensor_grad) = (
            p2p_communication.send_forward_backward_recv_forward_backward(
                output_tensor, input_tensor_grad, recv_prev=recv_prev,
                recv_next=recv_next, tensor_shape=tensor_shape, config=config))
        if recv_prev:
            input_tensors[next_forward_chunk].append(input_tensor)
        if recv_next:
            output_tensor_grads[next_backward_chunk].append(output_tensor_grad)
        deallocate_output_tensor(output_tensor, config.deallocate_pipeline_outputs)

    # ---- cooldown backwards ----
    if not forward_only:
        if all_warmup:
            p2p_communication.set_dpp_tags(bwd_recv=bwd_recv_tag(0))
            output_tensor_grads[num_model_chunks - 1].append(
                p2p_communication.recv_backward(
                    tensor_shape, config,
                    parallel_state.is_pipeline_last_stage(ignore_virtual=True)))
        for k in range(num_remaining, total_num_microbatches):
            input_tensor_grad = backward_step_helper(k)
            next_backward_chunk = get_model_chunk_id(k + 1, forward=False)
            recv_next = True
            if parallel_state.is_pipeline_last_stage(ignore_virtual=True) and \
                    next_backward_chunk == num_model_chunks - 1:
                recv_next = False
            if k == total_num_microbatches - 1:
                recv_next = False
            backward_chunk = get_model_chunk_id(k, forward=False)
            parallel_state.set_virtual_pipeline_model_parallel_rank(backward_chunk)
            if parallel_state.is_pipeline_first_stage():
                input_tensor_grad = None
            p2p_communication.set_dpp_tags(
                bwd_send=bwd_send_tag(k),
                bwd_recv=bwd_recv_tag(k + 1))
            output_tensor_grad = p2p_communication.send_backward_recv_backward(
                input_tensor_grad, recv_next, tensor_shape, config)
            if recv_next:
                output_tensor_grads[next_backward_chunk].append(output_tensor_grad)

    for ctx in no_sync_ctxs:
        ctx.__exit__(None, None, None)

    if not forward_only and config.finalize_model_grads_func is not None:
        config.finalize_model_grads_func(
            model, total_num_tokens if config.calculate_per_token_loss else None)
    return forward_data_store
