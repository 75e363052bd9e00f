"""MegaFBD disaggregated pipeline schedule.

Reference: schedules.py:2208-2505 (forward_or_backward_pipelining_
without_interleaving, forward_step_no_grad :355,
send/recv_corresponding_forward :1866/:655).

Forward instances stream gradient-free microbatch forwards down the
forward pipeline and ship every stage input to their dual backward
instance.  Backward instances recompute the stage forward WITH autograd
from the shipped input, then run the backward pipeline.  Because the
forward pipe never blocks on gradients, it runs ahead and keeps the
backward pipe saturated — the FBD decoupling.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..core import parallel_state
from ..core.enums import ModelType
from ..core.pipeline_parallel import p2p_communication
from ..core.pipeline_parallel.schedules import (
    backward_step,
    deallocate_output_tensor,
    forward_step,
)
from ..core.utils import get_model_config


def _dual_isend(tensor: torch.Tensor, dual_rank: int, pending: list):
    """Async ship of a stage input to the dual backward rank; the
    forward pipe keeps running ahead instead of blocking per microbatch
    (round-1 used blocking dist.send)."""
    t = tensor.detach().contiguous()
    pending.append((dist.isend(t, dst=dual_rank), t))
    # bound the in-flight queue so activations don't pile up unboundedly
    while len(pending) > 4:
        req, _ = pending.pop(0)
        req.wait()


def _dual_irecv(shape, dtype, dual_rank: int):
    device = torch.cuda.current_device() if torch.cuda.is_available() else "cpu"
    t = torch.empty(shape, dtype=dtype, device=device, requires_grad=True)
    req = dist.irecv(t, src=dual_rank)
    return req, t


def forward_backward_disaggregated(
        *, forward_step_func, data_iterator, model, num_microbatches: int,
        seq_length: int, micro_batch_size: int, decoder_seq_length=None,
        forward_only: bool = False, collect_non_loss_data: bool = False,
        first_val_step=None, adjust_tensor_shapes_fn=None):
    if isinstance(model, list):
        model = model[0]
    if isinstance(data_iterator, list):
        data_iterator = data_iterator[0]
    config = get_model_config(model)
    model_type = ModelType.encoder_or_decoder

    h = config.hidden_size
    s = seq_length // config.context_parallel_size
    if config.sequence_parallel:
        s //= config.tensor_model_parallel_size
    tensor_shape = (s, micro_batch_size, h)
    dtype = config.pipeline_dtype or torch.float32

    is_first = parallel_state.is_pipeline_first_stage()
    is_last = parallel_state.is_pipeline_last_stage()
    dual = parallel_state.get_forward_backward_parallel_dual_rank()
    forward_data_store: list = []

    if parallel_state.is_forward_stage():
        # ---------------- forward instance: no autograd, run ahead -------
        pending: list = []
        with torch.no_grad():
            for i in range(num_microbatches):
                input_tensor = p2p_communication.recv_forward(
                    tensor_shape, config, is_first)
                output_tensor, _ = forward_step(
                    forward_step_func, data_iterator, model,
                    num_microbatches, input_tensor, forward_data_store,
                    config, collect_non_loss_data, current_microbatch=i)
                p2p_communication.send_forward(output_tensor, config, is_last)
                # ship the stage input to the dual backward instance
                # (first stage reads tokens from its own iterator instead)
                if not is_first:
                    _dual_isend(input_tensor, dual, pending)
        for req, _ in pending:
            req.wait()
        return forward_data_store

    # -------------------- backward instance: recompute + backward --------
    import contextlib
    no_sync = model.no_sync() if hasattr(model, "no_sync") else \
        contextlib.nullcontext()
    no_sync.__enter__()
    sync_open = True
    # prefetch the first shipped input; post the next irecv before
    # computing so transfer overlaps the recompute+backward
    next_req = None
    if not is_first:
        next_req = _dual_irecv(tensor_shape, dtype, dual)
    for i in range(num_microbatches):
        if is_first:
            input_tensor = None
        else:
            req, input_tensor = next_req
            req.wait()
            if i + 1 < num_microbatches:
                next_req = _dual_irecv(tensor_shape, dtype, dual)
        output_tensor, _ = forward_step(
            forward_step_func, data_iterator, model, num_microbatches,
            input_tensor, forward_data_store, config, collect_non_loss_data,
            current_microbatch=i)
        if not forward_only:
            if i == num_microbatches - 1 and sync_open:
                no_sync.__exit__(None, None, None)
                sync_open = False
            output_tensor_grad = p2p_communication.recv_backward(
                tensor_shape, config, is_last)
            input_tensor_grad = backward_step(
                input_tensor, output_tensor, output_tensor_grad, model_type,
                config)
            p2p_communication.send_backward(input_tensor_grad, config,
                                            is_first)
        deallocate_output_tensor(output_tensor,
                                 config.deallocate_pipeline_outputs)

    if sync_open:
        no_sync.__exit__(None, None, None)
    if not forward_only and config.finalize_model_grads_func is not None:
        config.finalize_model_grads_func([model], None)
    return forward_data_store
