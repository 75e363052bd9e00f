"""Pipeline schedules: no-pipelining, 1F1B, and interleaved (VPP).

Reference: pipeline_parallel/schedules.py (get_forward_backward_func :28,
no_pipelining :618, interleaved :856, 1F1B :1918).

MegaScan instrumentation: this module emits the schedule-level events the
offline aggregator/detector key on — "forward", "backward", "loss",
"recv-*"/"send-*"/"exchange-*" (from p2p_communication) — closing the gap
noted in SURVEY.md §2.2 (the reference snapshot documents but does not
emit them).

MegaDPP: the interleaved schedule consults a pluggable microbatch
ordering policy (megatronapp_amd.dpp.policy) instead of a fixed
depth-first order; the same schedule runs stock when DPP is off.
"""

from __future__ import annotations

import contextlib
from typing import Callable, Iterator, List, Optional, Union

import torch

from .. import parallel_state
from ..enums import ModelType
from ..trace_hooks import trace_scope
from ..utils import get_model_config
from . import p2p_communication


def get_forward_backward_func():
    pp = parallel_state.get_pipeline_model_parallel_world_size()
    if pp > 1:
        if parallel_state.get_virtual_pipeline_model_parallel_world_size() is not None:
            return forward_backward_pipelining_with_interleaving
        return forward_backward_pipelining_without_interleaving
    return forward_backward_no_pipelining


def deallocate_output_tensor(out: Optional[torch.Tensor], deallocate=True):
    """Free activation memory while keeping the autograd graph node alive
    (reference deallocate_output_tensor)."""
    if out is None or not deallocate or not isinstance(out, torch.Tensor):
        return
    if out._base is not None:
        return
    out.data = torch.empty((1,), device=out.device, dtype=out.dtype)


def custom_backward(output: torch.Tensor, grad_output: Optional[torch.Tensor]):
    """Backward that tolerates a deallocated (shape-(1,)) output tensor.

    Bypasses torch.autograd.backward's shape check — the graph node still
    expects the original-shape grad even though ``output.data`` was
    deallocated by deallocate_output_tensor (reference custom_backward)."""
    if grad_output is None:
        assert output.numel() == 1, "implicit grad requires scalar output"
        torch.autograd.backward(output)
        return
    torch.autograd.Variable._execution_engine.run_backward(
        tensors=(output,), grad_tensors=(grad_output,), keep_graph=False,
        create_graph=False, inputs=tuple(), allow_unreachable=True,
        accumulate_grad=True)


def set_current_microbatch(model, i):
    pass


def forward_step(forward_step_func: Callable, data_iterator, model,
                 num_microbatches: int, input_tensor, forward_data_store: list,
                 config, collect_non_loss_data: bool = False,
                 is_first_microbatch: bool = False,
                 current_microbatch: Optional[int] = None,
                 vp_stage: Optional[int] = None):
    """One microbatch forward (reference forward_step :179)."""
    unwrapped_model = model
    while hasattr(unwrapped_model, "module"):
        unwrapped_model = unwrapped_model.module
    unwrapped_model.set_input_tensor(input_tensor)

    with trace_scope("forward"):
        output_tensor, loss_func = forward_step_func(data_iterator, model)

    num_tokens = torch.tensor(0, dtype=torch.int64)
    if parallel_state.is_pipeline_last_stage():
        with trace_scope("loss"):
            if not collect_non_loss_data:
                loss_out = loss_func(output_tensor)
                if len(loss_out) == 3:
                    loss, num_tokens, loss_reduced = loss_out
                    if not config.calculate_per_token_loss:
                        loss = loss / num_microbatches
                else:
                    loss, loss_reduced = loss_out
                    loss = loss / num_microbatches
                forward_data_store.append(loss_reduced)
                output_tensor = loss
            else:
                data = loss_func(output_tensor, non_loss_data=True)
                forward_data_store.append(data)
    return output_tensor, num_tokens


def backward_step(input_tensor, output_tensor, output_tensor_grad, model_type,
                  config):
    """One microbatch backward (reference backward_step :533)."""
    unwrap_input_tensor_grad = False
    if not isinstance(input_tensor, list):
        input_tensor = [input_tensor]
        unwrap_input_tensor_grad = True
    for t in input_tensor:
        if t is not None:
            t.retain_grad()
    if not isinstance(output_tensor, list):
        output_tensor = [output_tensor]
    if not isinstance(output_tensor_grad, list):
        output_tensor_grad = [output_tensor_grad]

    with trace_scope("backward"):
        if output_tensor_grad[0] is None and config.grad_scale_func is not None:
            output_tensor[0] = config.grad_scale_func(output_tensor[0])
        custom_backward(output_tensor[0], output_tensor_grad[0])

    input_tensor_grad = [None]
    if input_tensor is not None:
        input_tensor_grad = []
        for t in input_tensor:
            input_tensor_grad.append(None if t is None else t.grad)
    if unwrap_input_tensor_grad:
        input_tensor_grad = input_tensor_grad[0]
    return input_tensor_grad


def _no_sync_context(config, model):
    if config.no_sync_func is not None:
        return config.no_sync_func()
    if hasattr(model, "no_sync"):
        return model.no_sync()
    return contextlib.nullcontext()


def finish_embedding_wgrad_compute(config, embedding_module):
    pass


def forward_backward_no_pipelining(
        *, forward_step_func, data_iterator, model, num_microbatches: int,
        seq_length: int = None, micro_batch_size: int = None,
        decoder_seq_length: int = None, forward_only: bool = False,
        collect_non_loss_data: bool = False, first_val_step: bool = None,
        adjust_tensor_shapes_fn=None):
    """PP=1: run microbatches back-to-back, DDP grad sync only on the last
    (reference :618)."""
    if isinstance(model, list):
        model = model[0]
    if isinstance(data_iterator, list):
        data_iterator = data_iterator[0]
    config = get_model_config(model)

    forward_data_store: list = []
    input_tensor, output_tensor_grad = None, None
    total_num_tokens = torch.zeros(1, dtype=torch.int64,
                                   device="cuda" if torch.cuda.is_available() else "cpu")
    model_type = ModelType.encoder_or_decoder

    with _no_sync_context(config, model):
        for i in range(num_microbatches - 1):
            output_tensor, num_tokens = forward_step(
                forward_step_func, data_iterator, model, num_microbatches,
                input_tensor, forward_data_store, config,
                collect_non_loss_data, is_first_microbatch=(i == 0),
                current_microbatch=i)
            total_num_tokens += num_tokens.item() if torch.is_tensor(num_tokens) else num_tokens
            if not forward_only:
                backward_step(input_tensor, output_tensor, output_tensor_grad,
                              model_type, config)

    # last microbatch outside no_sync: grads sync here
    output_tensor, num_tokens = forward_step(
        forward_step_func, data_iterator, model, num_microbatches,
        input_tensor, forward_data_store, config, collect_non_loss_data,
        current_microbatch=num_microbatches - 1)
    total_num_tokens += num_tokens.item() if torch.is_tensor(num_tokens) else num_tokens
    if not forward_only:
        backward_step(input_tensor, output_tensor, output_tensor_grad,
                      model_type, config)
        if config.finalize_model_grads_func is not None:
            config.finalize_model_grads_func(
                [model], total_num_tokens if config.calculate_per_token_loss else None)
    return forward_data_store


def forward_backward_pipelining_without_interleaving(
        *, forward_step_func, data_iterator, model, num_microbatches: int,
        seq_length: int, micro_batch_size: int, decoder_seq_length: int = None,
        forward_only: bool = False, collect_non_loss_data: bool = False,
        first_val_step: bool = None, adjust_tensor_shapes_fn=None):
    """1F1B (reference :1918): warmup forwards, steady 1F1B, cooldown
    backwards; grad sync enabled for the final backward only."""
    if isinstance(model, list):
        assert len(model) == 1
        model = model[0]
    if isinstance(data_iterator, list):
        assert len(data_iterator) == 1
        data_iterator = data_iterator[0]
    config = get_model_config(model)
    model_type = ModelType.encoder_or_decoder

    pp_size = parallel_state.get_pipeline_model_parallel_world_size()
    pp_rank = parallel_state.get_pipeline_model_parallel_rank()
    num_warmup = min(pp_size - pp_rank - 1, num_microbatches)
    num_steady = num_microbatches - num_warmup

    # shape on the wire: [s, b, h] (divided by tp for sequence parallel, cp)
    h = config.hidden_size
    s = seq_length // config.context_parallel_size
    if config.sequence_parallel:
        s = s // config.tensor_model_parallel_size
    tensor_shape = (s, micro_batch_size, h)
    if adjust_tensor_shapes_fn is not None:
        tensor_shape = adjust_tensor_shapes_fn(tensor_shape)

    is_first = parallel_state.is_pipeline_first_stage()
    is_last = parallel_state.is_pipeline_last_stage()

    forward_data_store: list = []
    input_tensors: List = []
    output_tensors: List = []
    total_num_tokens = torch.zeros(1, dtype=torch.int64,
                                   device="cuda" if torch.cuda.is_available() else "cpu")

    class _SyncGate:
        """Holds DDP grad sync off until the final backward."""

        def __init__(self):
            self.ctx = _no_sync_context(config, model)
            self.ctx.__enter__()
            self.open = True

        def release(self):
            if self.open:
                self.ctx.__exit__(None, None, None)
                self.open = False

    sync_gate = _SyncGate()

    def _fwd(input_tensor, i):
        output_tensor, num_tokens = forward_step(
            forward_step_func, data_iterator, model, num_microbatches,
            input_tensor, forward_data_store, config, collect_non_loss_data,
            is_first_microbatch=(i == 0), current_microbatch=i)
        total_num_tokens.add_(num_tokens.item() if torch.is_tensor(num_tokens)
                              else num_tokens)
        return output_tensor

    # --- warmup forwards ---
    for i in range(num_warmup):
        p2p_communication.set_dpp_tags(fwd_recv=(0, i), fwd_send=(0, i))
        input_tensor = p2p_communication.recv_forward(tensor_shape, config, is_first)
        output_tensor = _fwd(input_tensor, i)
        p2p_communication.send_forward(output_tensor, config, is_last)
        if not forward_only:
            input_tensors.append(input_tensor)
            output_tensors.append(output_tensor)
            deallocate_output_tensor(output_tensor, config.deallocate_pipeline_outputs)

    # --- steady 1F1B ---
    if num_steady > 0:
        p2p_communication.set_dpp_tags(fwd_recv=(0, num_warmup))
        input_tensor = p2p_communication.recv_forward(tensor_shape, config, is_first)
    for i in range(num_steady):
        last_iteration = (i == num_steady - 1)
        f_mb = num_warmup + i
        output_tensor = _fwd(input_tensor, f_mb)
        if forward_only:
            p2p_communication.set_dpp_tags(fwd_send=(0, f_mb),
                                           fwd_recv=(0, f_mb + 1))
            p2p_communication.send_forward(output_tensor, config, is_last)
            if not last_iteration:
                input_tensor = p2p_communication.recv_forward(
                    tensor_shape, config, is_first)
            continue

        p2p_communication.set_dpp_tags(fwd_send=(0, f_mb), bwd_recv=(0, i))
        output_tensor_grad = p2p_communication.send_forward_recv_backward(
            output_tensor, tensor_shape, config, is_last)
        input_tensors.append(input_tensor)
        output_tensors.append(output_tensor)
        deallocate_output_tensor(output_tensor, config.deallocate_pipeline_outputs)

        input_tensor_b = input_tensors.pop(0)
        output_tensor_b = output_tensors.pop(0)

        if num_warmup == 0 and last_iteration:
            sync_gate.release()

        input_tensor_grad = backward_step(
            input_tensor_b, output_tensor_b, output_tensor_grad, model_type,
            config)
        if last_iteration:
            input_tensor = None
            p2p_communication.set_dpp_tags(bwd_send=(0, i))
            p2p_communication.send_backward(input_tensor_grad, config, is_first)
        else:
            p2p_communication.set_dpp_tags(bwd_send=(0, i),
                                           fwd_recv=(0, f_mb + 1))
            input_tensor = p2p_communication.send_backward_recv_forward(
                input_tensor_grad, tensor_shape, config, is_first)

    # --- cooldown backwards ---
    if not forward_only:
        for i in range(num_warmup):
            if i == num_warmup - 1:
                sync_gate.release()
            input_tensor_b = input_tensors.pop(0)
            output_tensor_b = output_tensors.pop(0)
            b_mb = num_steady + i
            p2p_communication.set_dpp_tags(bwd_recv=(0, b_mb),
                                           bwd_send=(0, b_mb))
            output_tensor_grad = p2p_communication.recv_backward(
                tensor_shape, config, is_last)
            input_tensor_grad = backward_step(
                input_tensor_b, output_tensor_b, output_tensor_grad,
                model_type, config)
            p2p_communication.send_backward(input_tensor_grad, config, is_first)

    sync_gate.release()

    if not forward_only and config.finalize_model_grads_func is not None:
        with trace_scope("grad-sync"):
            config.finalize_model_grads_func(
                [model], total_num_tokens if config.calculate_per_token_loss else None)
    return forward_data_store


# interleaved schedule lives in its own module for clarity
def forward_backward_pipelining_with_interleaving(**kwargs):
    from .interleaved import forward_backward_pipelining_with_interleaving as f
    return f(**kwargs)
