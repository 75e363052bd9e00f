"""Tensor-parallel linear layers + vocab-parallel embedding.

Reference: tensor_parallel/layers.py (ColumnParallelLinear:675,
RowParallelLinear:1019, VocabParallelEmbedding:172,
LinearWithGradAccumulationAndAsyncCommunication:404-560).

MI355X design notes:
* GEMMs go through torch.matmul -> hipBLASLt (bf16 MFMA).  The framework's
  hand-written HIP kernels cover the *fused* ops (norms, activations,
  attention) — plain projection GEMMs are hipBLASLt's job.
* The backward overlaps the TP grad all-reduce (latency-bound over xGMI)
  with the weight-gradient GEMM on the compute stream: the all-reduce is
  issued first on the communication side via async_op=True, the wgrad GEMM
  fills the gap (reference behaviour of :404).
* Gradient accumulation fuses into the DDP-owned fp32 ``main_grad`` buffer.
"""

from __future__ import annotations

import os
import warnings
from typing import Callable, Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F
from torch.nn.parameter import Parameter

from .. import parallel_state
from ... import ops as _ops
from .mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
    scatter_to_tensor_model_parallel_region,
    _gather_along_first_dim,
    _reduce_scatter_along_first_dim,
)
from .random import get_cuda_rng_tracker
from .utils import VocabUtility, divide

_MODEL_PARALLEL_ATTRIBUTE_DEFAULTS = {
    "tensor_model_parallel": False,
    "partition_dim": -1,
    "partition_stride": 1,
}


def param_is_not_tensor_parallel_duplicate(param):
    return getattr(param, "tensor_model_parallel", False) or (
        parallel_state.get_tensor_model_parallel_rank() == 0)


def set_tensor_model_parallel_attributes(tensor, is_parallel, dim, stride=1):
    tensor.tensor_model_parallel = is_parallel
    tensor.partition_dim = dim
    tensor.partition_stride = stride


def set_defaults_if_not_set_tensor_model_parallel_attributes(tensor):
    for attr, default in _MODEL_PARALLEL_ATTRIBUTE_DEFAULTS.items():
        if not hasattr(tensor, attr):
            setattr(tensor, attr, default)


def copy_tensor_model_parallel_attributes(dst, src):
    for attr in _MODEL_PARALLEL_ATTRIBUTE_DEFAULTS:
        if hasattr(src, attr):
            setattr(dst, attr, getattr(src, attr))


def _initialize_affine_weight(weight, init_method, partition_dim, stride=1,
                              expert_parallel=False):
    set_tensor_model_parallel_attributes(weight, True, partition_dim, stride)
    if torch.cuda.is_available() and weight.is_cuda:
        with get_cuda_rng_tracker().fork():
            init_method(weight)
    else:
        init_method(weight)


class _EmbeddingWithMainGradAccum(torch.autograd.Function):
    """Embedding lookup whose backward scatter-adds dE straight into the
    fp32 main_grad buffer (one fused kernel instead of the eager
    embedding_dense_backward + cast-add pass)."""

    @staticmethod
    def forward(ctx, masked_input, weight, input_mask):
        out = F.embedding(masked_input, weight)
        if input_mask is not None:
            out = out.clone()
            out[input_mask, :] = 0.0
        tokens = masked_input.to(torch.int32)
        if input_mask is not None:
            tokens = tokens.masked_fill(input_mask, -1)
        ctx.save_for_backward(tokens)
        ctx.weight_ref = weight
        return out

    @staticmethod
    def backward(ctx, dy):
        (tokens,) = ctx.saved_tensors
        weight = ctx.weight_ref
        _ops.get_ops().embedding_bwd_accum(
            dy.contiguous().reshape(-1, dy.shape[-1]),
            tokens.reshape(-1).contiguous(), weight.main_grad)
        weight.grad_added_to_main_grad = True
        dummy = torch.empty(weight.shape, dtype=weight.dtype,
                            device=weight.device)
        return None, dummy, None


class VocabParallelEmbedding(torch.nn.Module):
    """Embedding sharded along the vocab dimension (reference layers.py:172).

    Forward masks out-of-shard tokens, local lookup, then one TP
    all-reduce (or reduce-scatter when sequence-parallel) merges shards.
    """

    def __init__(self, num_embeddings, embedding_dim, *, init_method,
                 config, reduce_scatter_embeddings: bool = False):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.reduce_scatter_embeddings = reduce_scatter_embeddings
        self.tensor_model_parallel_size = parallel_state.get_tensor_model_parallel_world_size()
        (self.vocab_start_index, self.vocab_end_index) = (
            VocabUtility.vocab_range_from_global_vocab_size(
                num_embeddings, parallel_state.get_tensor_model_parallel_rank(),
                self.tensor_model_parallel_size))
        self.num_embeddings_per_partition = (
            self.vocab_end_index - self.vocab_start_index)
        self.weight = Parameter(torch.empty(
            self.num_embeddings_per_partition, embedding_dim,
            dtype=config.params_dtype))
        _initialize_affine_weight(self.weight, init_method, partition_dim=0)

    def forward(self, input_):
        if self.tensor_model_parallel_size > 1:
            input_mask = (input_ < self.vocab_start_index) | (
                input_ >= self.vocab_end_index)
            masked_input = input_.clone() - self.vocab_start_index
            masked_input[input_mask] = 0
        else:
            input_mask = None
            masked_input = input_
        if (input_.is_cuda and _ops.have_ops() and
                self.weight.dtype == torch.bfloat16 and
                hasattr(self.weight, "main_grad") and
                hasattr(self.weight, "grad_added_to_main_grad")):
            output_parallel = _EmbeddingWithMainGradAccum.apply(
                masked_input, self.weight, input_mask)
        else:
            output_parallel = F.embedding(masked_input, self.weight)
            if input_mask is not None:
                output_parallel = output_parallel.clone()
                output_parallel[input_mask, :] = 0.0
        if self.reduce_scatter_embeddings:
            # [b, s, h] -> [s, b, h] -> [s/tp, b, h]
            output_parallel = output_parallel.transpose(0, 1).contiguous()
            output = reduce_scatter_to_sequence_parallel_region(output_parallel)
        else:
            output = reduce_from_tensor_model_parallel_region(output_parallel)
        return output


class LinearWithGradAccumulationAndAsyncCommunication(torch.autograd.Function):
    """Fused linear core (reference layers.py:404-560).

    forward:  (SP? all-gather input) ; out = x @ W^T (+b)
    backward: dgrad GEMM -> issue async TP all-reduce / SP reduce-scatter of
              dgrad -> wgrad GEMM overlaps the collective -> wait.
              wgrad accumulates straight into param.main_grad when the DDP
              grad buffer owns one (gradient_accumulation_fusion).
    """

    @staticmethod
    def forward(ctx, input, weight, bias, gradient_accumulation_fusion,
                async_grad_allreduce, sequence_parallel):
        ctx.use_bias = bias is not None
        ctx.bias_param = bias
        ctx.gradient_accumulation_fusion = gradient_accumulation_fusion
        ctx.async_grad_allreduce = async_grad_allreduce
        ctx.sequence_parallel = sequence_parallel

        if sequence_parallel:
            total_input = _gather_along_first_dim(input)
        else:
            total_input = input
        ctx.save_for_backward(input, weight)
        from ..fp8 import fp8_forward, fp8_train_enabled
        ctx.fp8 = fp8_train_enabled(weight, total_input)
        if ctx.fp8:
            output = fp8_forward(total_input, weight, bias)
        elif bias is not None:
            # F.linear -> addmm: hipBLASLt fuses the bias in the GEMM
            # epilogue (a separate [s*b, out] add costs ~166 us/layer at
            # mbs16 on the QKV projection)
            output = F.linear(total_input, weight, bias)
        else:
            output = torch.matmul(total_input, weight.t())
        return output

    @staticmethod
    def backward(ctx, grad_output):
        input, weight = ctx.saved_tensors
        use_bias = ctx.use_bias

        if ctx.sequence_parallel:
            total_input = _gather_along_first_dim(input)
        else:
            total_input = input

        if getattr(ctx, "fp8", False):
            from ..fp8 import fp8_dgrad
            grad_input = fp8_dgrad(grad_output, weight)
        else:
            grad_input = grad_output.matmul(weight)

        handle = None
        if ctx.sequence_parallel:
            assert not ctx.async_grad_allreduce
            sub_grad_input = torch.empty(
                input.shape, dtype=input.dtype, device=input.device)
            handle = dist.reduce_scatter_tensor(
                sub_grad_input, grad_input.contiguous(),
                group=parallel_state.get_tensor_model_parallel_group(),
                async_op=True)
        elif ctx.async_grad_allreduce:
            handle = dist.all_reduce(
                grad_input, group=parallel_state.get_tensor_model_parallel_group(),
                async_op=True)

        # wgrad GEMM overlaps the collective above
        grad_output_2d = grad_output.reshape(-1, grad_output.shape[-1])
        total_input_2d = total_input.reshape(-1, total_input.shape[-1])
        fused_dbias = None
        if ctx.gradient_accumulation_fusion and hasattr(weight, "main_grad"):
            if (grad_output.is_cuda and _ops.have_ops()
                    and grad_output.dtype == torch.bfloat16):
                # fp32-accumulating hipblasLt wgrad straight into main_grad;
                # when this linear also owns a fused bias grad, ride the
                # BGRADB epilogue so dbias costs no extra HBM pass
                go2 = grad_output_2d.contiguous()
                ti2 = total_input_2d.contiguous()
                lt = _ops.get_ops()
                wgrad_done = False
                if getattr(ctx, "fp8", False):
                    # fp8 wgrad (transpose-quantized e4m3 GEMM, fp32 out
                    # accumulated); bias grad stays on the colsum path
                    from ..fp8 import fp8_wgrad
                    wgrad_done = fp8_wgrad(go2, ti2, weight.main_grad)
                bias_param = ctx.bias_param if use_bias else None
                want_bgrad = (
                    not wgrad_done
                    and bias_param is not None
                    and hasattr(bias_param, "main_grad")
                    and hasattr(bias_param, "grad_added_to_main_grad")
                    and "bgrad" not in os.environ.get(
                        "MEGATRONAPP_DISABLE_FUSED", ""))
                if want_bgrad:
                    dbias_tmp = torch.empty_like(bias_param.main_grad)
                    if lt.wgrad_accum_bgrad(go2, ti2, weight.main_grad,
                                            dbias_tmp):
                        fused_dbias = dbias_tmp
                    else:          # no epilogue algo for this shape
                        lt.wgrad_accum(go2, ti2, weight.main_grad)
                elif not wgrad_done:
                    lt.wgrad_accum(go2, ti2, weight.main_grad)
            else:
                weight.main_grad.add_(
                    torch.matmul(grad_output_2d.t(), total_input_2d))
            if hasattr(weight, "grad_added_to_main_grad"):
                # DDP-managed: return a dummy grad so the param's
                # post-accumulate hook still fires for bucket bookkeeping
                grad_weight = torch.empty(
                    weight.shape, dtype=weight.dtype, device=weight.device,
                    requires_grad=False)
                weight.grad_added_to_main_grad = True
            else:
                grad_weight = None
        else:
            grad_weight = grad_output_2d.t().matmul(total_input_2d)
        grad_bias = None
        if use_bias and fused_dbias is not None:
            bias_param = ctx.bias_param
            bias_param.main_grad.add_(fused_dbias)
            bias_param.grad_added_to_main_grad = True
            grad_bias = torch.empty(bias_param.shape,
                                    dtype=bias_param.dtype,
                                    device=bias_param.device)
        elif use_bias:
            bias_param = ctx.bias_param
            if (grad_output.is_cuda and _ops.have_ops() and
                    grad_output.dtype == torch.bfloat16 and
                    bias_param is not None and
                    hasattr(bias_param, "main_grad") and
                    hasattr(bias_param, "grad_added_to_main_grad")):
                _ops.get_ops().colsum_accum(grad_output_2d.contiguous(),
                                            bias_param.main_grad)
                bias_param.grad_added_to_main_grad = True
                grad_bias = torch.empty(bias_param.shape,
                                        dtype=bias_param.dtype,
                                        device=bias_param.device)
            else:
                grad_bias = grad_output_2d.sum(dim=0)

        if handle is not None:
            handle.wait()
        if ctx.sequence_parallel:
            grad_input = sub_grad_input

        return grad_input, grad_weight, grad_bias, None, None, None


def linear_with_grad_accumulation_and_async_allreduce(
        input, weight, bias, gradient_accumulation_fusion,
        async_grad_allreduce, sequence_parallel):
    # fp8 serving fast path: weights carry e4m3 data + per-row scales
    # (inference/fp8.py quantize_model_fp8); 2x MFMA peak on gfx950
    if (not torch.is_grad_enabled() and input.is_cuda
            and hasattr(weight, "fp8_data")):
        from ...inference.fp8 import fp8_linear
        out = fp8_linear(input, weight)
        if out is not None:    # None: below the fp8 token threshold
            if bias is not None:
                out = out + bias
            return out
    return LinearWithGradAccumulationAndAsyncCommunication.apply(
        input, weight, bias, gradient_accumulation_fusion,
        async_grad_allreduce, sequence_parallel)


class ColumnParallelLinear(torch.nn.Module):
    """Y = XA^T with A sharded along its output dim (reference :675)."""

    def __init__(self, input_size, output_size, *, config, init_method,
                 bias=True, gather_output=False, stride=1,
                 keep_master_weight_for_test=False, skip_bias_add=False,
                 skip_weight_param_allocation=False, is_expert=False,
                 tp_comm_buffer_name=None):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.gather_output = gather_output
        self.skip_bias_add = skip_bias_add
        self.config = config
        self.is_expert = is_expert
        world_size = (1 if is_expert else
                      parallel_state.get_tensor_model_parallel_world_size())
        self.output_size_per_partition = divide(output_size, world_size)

        if not skip_weight_param_allocation:
            self.weight = Parameter(torch.empty(
                self.output_size_per_partition, input_size,
                dtype=config.params_dtype))
            _initialize_affine_weight(self.weight, init_method, partition_dim=0,
                                      stride=stride)
            setattr(self.weight, "allreduce", not is_expert)
        else:
            self.weight = None

        if bias:
            self.bias = Parameter(torch.zeros(
                self.output_size_per_partition, dtype=config.params_dtype))
            set_tensor_model_parallel_attributes(self.bias, True, 0, stride)
            setattr(self.bias, "allreduce", not is_expert)
        else:
            self.register_parameter("bias", None)

        self.sequence_parallel = config.sequence_parallel and not is_expert
        self.async_tensor_model_parallel_allreduce = (
            config.async_tensor_model_parallel_allreduce
            and world_size > 1 and not self.sequence_parallel)
        self.gradient_accumulation_fusion = config.gradient_accumulation_fusion

    def forward(self, input_, weight=None):
        weight = weight if weight is not None else self.weight
        bias = None if self.skip_bias_add else self.bias

        if (self.async_tensor_model_parallel_allreduce or
                self.sequence_parallel or self.is_expert):
            input_parallel = input_
        else:
            input_parallel = copy_to_tensor_model_parallel_region(input_)

        output_parallel = linear_with_grad_accumulation_and_async_allreduce(
            input_parallel, weight, bias,
            self.gradient_accumulation_fusion,
            self.async_tensor_model_parallel_allreduce,
            self.sequence_parallel)
        if self.gather_output:
            output = gather_from_tensor_model_parallel_region(output_parallel)
        else:
            output = output_parallel
        output_bias = self.bias if self.skip_bias_add else None
        return output, output_bias


class RowParallelLinear(torch.nn.Module):
    """Y = XA^T with A sharded along its input dim (reference :1019).

    Forward ends with the TP all-reduce (or SP reduce-scatter); bias is
    added after the reduction (skip_bias_add defers it to the caller for
    fusion into the following kernel)."""

    def __init__(self, input_size, output_size, *, config, init_method,
                 bias=True, input_is_parallel=True, stride=1,
                 keep_master_weight_for_test=False, skip_bias_add=False,
                 is_expert=False, tp_comm_buffer_name=None):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.input_is_parallel = input_is_parallel
        self.skip_bias_add = skip_bias_add
        self.config = config
        self.is_expert = is_expert
        world_size = (1 if is_expert else
                      parallel_state.get_tensor_model_parallel_world_size())
        self.input_size_per_partition = divide(input_size, world_size)

        self.weight = Parameter(torch.empty(
            output_size, self.input_size_per_partition,
            dtype=config.params_dtype))
        _initialize_affine_weight(self.weight, init_method, partition_dim=1,
                                  stride=stride)
        setattr(self.weight, "allreduce", not is_expert)
        if bias:
            self.bias = Parameter(torch.zeros(output_size, dtype=config.params_dtype))
            setattr(self.bias, "allreduce", not is_expert)
            setattr(self.bias, "sequence_parallel", config.sequence_parallel)
        else:
            self.register_parameter("bias", None)
        self.sequence_parallel = config.sequence_parallel and not is_expert
        self.gradient_accumulation_fusion = config.gradient_accumulation_fusion

    def forward(self, input_):
        if self.input_is_parallel or self.is_expert:
            input_parallel = input_
        else:
            input_parallel = scatter_to_tensor_model_parallel_region(input_)

        output_parallel = linear_with_grad_accumulation_and_async_allreduce(
            input_parallel, self.weight, None,
            self.gradient_accumulation_fusion, False, False)
        if self.is_expert:
            output = output_parallel
        elif self.sequence_parallel:
            output = reduce_scatter_to_sequence_parallel_region(output_parallel)
        else:
            output = reduce_from_tensor_model_parallel_region(output_parallel)
        if not self.skip_bias_add:
            output = output + self.bias if self.bias is not None else output
            output_bias = None
        else:
            output_bias = self.bias
        return output, output_bias
