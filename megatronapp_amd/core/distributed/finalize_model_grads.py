"""Cross-group gradient reductions after backward.

Reference: distributed/finalize_model_grads.py:257 — (1) finish DP bucket
reduction, (2) tied embedding grads all-reduced across the PP embedding
group (:120-188), (3) sequence-parallel norm grads all-reduced across TP
(:190-227), (4) optional per-token grad scaling.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from .. import parallel_state
from ..trace_hooks import trace_scope
from ..utils import get_model_config, unwrap_model


def _allreduce_embedding_grads(model: List[torch.nn.Module], config):
    if (parallel_state.get_pipeline_model_parallel_world_size() == 1 or
            not parallel_state.is_rank_in_embedding_group(ignore_virtual=True)):
        return
    group = parallel_state.get_embedding_group()
    if group is None:
        return
    unwrapped = unwrap_model(model)
    for chunk in (unwrapped if isinstance(unwrapped, list) else [unwrapped]):
        if not getattr(chunk, "share_embeddings_and_output_weights", False):
            continue
        weight = chunk.shared_embedding_or_output_weight()
        if weight is None:
            continue
        grad = getattr(weight, "main_grad", None)
        if grad is None:
            grad = weight.grad
        if grad is not None:
            dist.all_reduce(grad, group=group)


def _allreduce_layernorm_grads(model: List[torch.nn.Module], config):
    if not (config.sequence_parallel or config.qk_layernorm):
        return
    if parallel_state.get_tensor_model_parallel_world_size() == 1:
        return
    grads = []
    for m in model:
        for param in m.parameters():
            if getattr(param, "sequence_parallel", False):
                grad = getattr(param, "main_grad", None)
                if grad is None:
                    grad = param.grad
                if grad is not None:
                    grads.append(grad.data)
    if not grads:
        return
    flat = torch._utils._flatten_dense_tensors(grads)
    dist.all_reduce(flat, group=parallel_state.get_tensor_model_parallel_group())
    for buf, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
        buf.copy_(synced)


def finalize_model_grads(model: List[torch.nn.Module],
                         num_tokens: Optional[torch.Tensor] = None):
    config = get_model_config(model[0])

    with trace_scope("grad-sync-dp"):
        for chunk in model:
            if hasattr(chunk, "finish_grad_sync"):
                chunk.finish_grad_sync()

    with trace_scope("grad-sync-embedding"):
        _allreduce_embedding_grads(model, config)
    with trace_scope("grad-sync-layernorm"):
        _allreduce_layernorm_grads(model, config)

    if num_tokens is not None:
        # normalize per-token loss: all-reduce token count over DP, scale grads
        dp_group = parallel_state.get_data_parallel_group(with_context_parallel=True)
        dist.all_reduce(num_tokens, group=dp_group)
        if num_tokens.item() > 0:
            scaling = 1.0 / num_tokens.item()
            for chunk in model:
                if hasattr(chunk, "buffers"):
                    for buf in chunk.buffers:
                        buf.grad_data.mul_(scaling)
