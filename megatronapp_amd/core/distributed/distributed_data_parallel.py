"""DistributedDataParallel: grad buffering + overlapped RCCL reduction.

Reference: distributed/distributed_data_parallel.py:21.  Wraps one model
chunk; grads accumulate into fp32 contiguous buffers (param.main_grad);
post-accumulate hooks count bucket readiness and fire the bucket
all-reduce / reduce-scatter asynchronously during the last microbatch's
backward.
"""

from __future__ import annotations

import contextlib
import logging
from typing import Dict, List

import torch
import torch.distributed as dist

from .. import parallel_state
from ..transformer_config import TransformerConfig
from .distributed_data_parallel_config import DistributedDataParallelConfig
from .param_and_grad_buffer import _ParamAndGradBuffer

logger = logging.getLogger(__name__)


class DistributedDataParallel(torch.nn.Module):
    def __init__(self, config: TransformerConfig,
                 ddp_config: DistributedDataParallelConfig,
                 module: torch.nn.Module,
                 disable_bucketing: bool = False):
        super().__init__()
        self.config = config
        self.ddp_config = ddp_config
        self.module = module

        dp_group = parallel_state.get_data_parallel_group(with_context_parallel=True)
        ep_dp_group = parallel_state.get_expert_data_parallel_group()
        self.data_parallel_group = dp_group

        dense_params: List[torch.nn.Parameter] = []
        expert_params: List[torch.nn.Parameter] = []
        self.param_to_name: Dict[torch.nn.Parameter, str] = {}
        for name, param in module.named_parameters():
            if not param.requires_grad:
                continue
            self.param_to_name[param] = name
            param.grad_added_to_main_grad = False
            if getattr(param, "allreduce", True):
                dense_params.append(param)
            else:
                expert_params.append(param)

        grad_dtype = torch.float32 if ddp_config.grad_reduce_in_fp32 else \
            config.params_dtype
        bucket_size = ddp_config.bucket_size
        if disable_bucketing:
            bucket_size = None

        dp_world = dist.get_world_size(dp_group)
        scale = 1.0 / dp_world if dp_world > 1 else 1.0

        self.buffers: List[_ParamAndGradBuffer] = []
        if dense_params:
            self.buffers.append(_ParamAndGradBuffer(
                ddp_config, config.params_dtype, grad_dtype, dense_params,
                dp_group, bucket_size, self.param_to_name,
                gradient_scaling_factor=scale))
        if expert_params and ep_dp_group is not None:
            ep_world = dist.get_world_size(ep_dp_group)
            self.buffers.append(_ParamAndGradBuffer(
                ddp_config, config.params_dtype, grad_dtype, expert_params,
                ep_dp_group, bucket_size, self.param_to_name,
                gradient_scaling_factor=1.0 / ep_world if ep_world > 1 else 1.0))

        # per-param bucket map for the hooks
        self.param_to_bucket = {}
        for buf in self.buffers:
            self.param_to_bucket.update(buf.param_to_bucket)

        self.is_grad_sync_enabled = True
        self._hook_handles = []
        for param in list(dense_params) + list(expert_params):
            self._hook_handles.append(param.register_post_accumulate_grad_hook(
                self._make_post_accumulate_hook()))

    # ------------------------------------------------------------------
    def _make_post_accumulate_hook(self):
        def hook(param):
            if param.grad is not None and not param.grad_added_to_main_grad:
                param.main_grad.add_(param.grad.data)
            param.grad = None
            param.grad_added_to_main_grad = False
            if self.ddp_config.overlap_grad_reduce and self.is_grad_sync_enabled:
                self.param_to_bucket[param].register_grad_ready(param)
        return hook

    @contextlib.contextmanager
    def no_sync(self):
        self.is_grad_sync_enabled = False
        try:
            yield
        finally:
            self.is_grad_sync_enabled = True

    def start_grad_sync(self):
        for buf in self.buffers:
            buf.start_grad_sync()

    def finish_grad_sync(self):
        for buf in self.buffers:
            buf.finish_grad_sync()

    def start_param_sync(self, *unused, force_sync: bool = False):
        handles = [buf.start_param_sync(async_op=not force_sync)
                   for buf in self.buffers]
        return handles

    def zero_grad_buffer(self):
        for param in self.param_to_name:
            param.grad_added_to_main_grad = False
        for buf in self.buffers:
            buf.reset()

    def broadcast_params(self):
        for param in self.module.parameters():
            is_expert = not getattr(param, "allreduce", True)
            group = (parallel_state.get_expert_data_parallel_group() if is_expert
                     else self.data_parallel_group)
            if group is None:
                continue
            src = dist.get_process_group_ranks(group)[0]
            dist.broadcast(param.data, src_rank := src, group=group)

    def forward(self, *inputs, **kwargs):
        return self.module(*inputs, **kwargs)

    def set_input_tensor(self, input_tensor):
        return self.module.set_input_tensor(input_tensor)

    def state_dict(self, destination=None, prefix="", keep_vars=False):
        return self.module.state_dict(destination, prefix, keep_vars)

    def state_dict_for_save_checkpoint(self, prefix="", keep_vars=False):
        return self.module.state_dict_for_save_checkpoint(prefix, keep_vars)

    def sharded_state_dict(self, prefix: str = "", *args, **kwargs):
        return self.module.sharded_state_dict(prefix, *args, **kwargs)

    def load_state_dict(self, state_dict, strict=True):
        return self.module.load_state_dict(state_dict, strict=strict)
