"""Expert MLPs (reference moe/experts.py GroupedMLP:90, SequentialMLP:856).

SequentialMLP: one MLP per local expert, applied to its token segment.
GroupedMLP: all local experts' weights stacked in two 3D tensors; on
MI355X the per-expert GEMMs run as ONE hipBLASLt grouped-GEMM launch via
torch._grouped_mm (measured 1004 TF vs 694 for the segmented loop at
mixtral-8x1b shapes).  The aten op has no autograd on ROCm, so
_GroupedMMFn supplies it: dx is another grouped mm against w^T and dw a
(2D, 2D)->3D grouped mm over the jagged token dim.  Falls back to the
segmented loop off-GPU.
"""

from __future__ import annotations

import os

import torch
from torch import nn

from .... import ops as _ops


def _grouped_mm_available(t: torch.Tensor) -> bool:
    return t.is_cuda and hasattr(torch, "_grouped_mm")


class _GroupedMMFn(torch.autograd.Function):
    """y[seg_e] = x[seg_e] @ w[e] for jagged segments given by offs
    (int32 inclusive cumsum of tokens_per_expert).

    When the stacked weight is a DDP-managed param (fp32 main_grad) and
    host_counts is provided, the backward skips the grouped bf16 dw and
    instead runs one fp32-accumulating hipBLASLt wgrad per expert slice
    straight into main_grad[e] — removing both the [E, in, out] bf16 dw
    materialization and the DDP hook's fp32 += bf16 pass over every
    expert weight (together ~2 extra full passes over the expert params
    per microbatch), and matching the dense linears' fp32 wgrad
    accumulation precision."""

    @staticmethod
    def forward(ctx, x, w, offs, host_counts=None):
        ctx.save_for_backward(x, w, offs)
        ctx.host_counts = host_counts
        ctx.wparam = w if host_counts is not None else None
        return torch._grouped_mm(x, w, offs=offs)

    @staticmethod
    def backward(ctx, dy):
        x, w, offs = ctx.saved_tensors
        dy = dy.contiguous()
        dx = torch._grouped_mm(dy, w.transpose(1, 2), offs=offs)
        wp = ctx.wparam
        if (wp is not None and hasattr(wp, "main_grad")
                and hasattr(wp, "grad_added_to_main_grad")
                and wp.main_grad.dtype == torch.float32
                and _ops.have_ops()):
            lt = _ops.get_ops()
            mg = wp.main_grad
            start = 0
            for e, n in enumerate(ctx.host_counts):
                if n > 0:
                    # mg[e][i, o] += sum_r x[r, i] * dy[r, o]
                    lt.wgrad_accum(x[start:start + n], dy[start:start + n],
                                   mg[e])
                start += n
            wp.grad_added_to_main_grad = True
            dw = torch.empty_like(w)   # dummy so the DDP hook still fires
        else:
            # dw[e] = x[seg]^T @ dy[seg]: (2D, 2D) -> 3D grouped
            dw = torch._grouped_mm(x.t().contiguous(), dy, offs=offs)
        return dx, dw, None, None

from ...fusions.fused_bias_act import bias_gelu_impl, bias_swiglu_impl
from ...transformer_config import TransformerConfig
from ..mlp import MLP, MLPSubmodules


class SequentialMLP(nn.Module):
    def __init__(self, num_local_experts: int, config: TransformerConfig,
                 submodules: MLPSubmodules = None):
        super().__init__()
        from ...tensor_parallel.layers import (ColumnParallelLinear,
                                               RowParallelLinear)
        submodules = submodules or MLPSubmodules(
            linear_fc1=ColumnParallelLinear, linear_fc2=RowParallelLinear)
        self.num_local_experts = num_local_experts
        self.config = config
        self.local_experts = nn.ModuleList([
            MLP(config, submodules, is_expert=True,
                ffn_hidden_size=config.moe_ffn_hidden_size)
            for _ in range(num_local_experts)])

    def forward(self, permuted_tokens: torch.Tensor,
                tokens_per_expert: torch.Tensor):
        outputs = []
        start = 0
        counts = tokens_per_expert.tolist()
        for expert, n in zip(self.local_experts, counts):
            seg = permuted_tokens[start:start + n]
            start += n
            if n == 0:
                outputs.append(seg)
                continue
            out, bias = expert(seg)
            if bias is not None:
                out = out + bias
            outputs.append(out)
        return torch.cat(outputs, dim=0) if outputs else permuted_tokens


class GroupedMLP(nn.Module):
    """Stacked-weight experts: w1 [E, h, f*(2 if gated)], w2 [E, f, h]."""

    def __init__(self, num_local_experts: int, config: TransformerConfig,
                 submodules=None):
        super().__init__()
        self.num_local_experts = num_local_experts
        self.config = config
        f = config.moe_ffn_hidden_size
        fc1_out = f * 2 if config.gated_linear_unit else f
        self.weight1 = nn.Parameter(torch.empty(
            num_local_experts, config.hidden_size, fc1_out,
            dtype=config.params_dtype))
        self.weight2 = nn.Parameter(torch.empty(
            num_local_experts, f, config.hidden_size,
            dtype=config.params_dtype))
        for e in range(num_local_experts):
            config.init_method(self.weight1[e])
            config.output_layer_init_method(self.weight2[e])
        for w in (self.weight1, self.weight2):
            setattr(w, "allreduce", False)   # expert-parallel params
        self.activation = (bias_swiglu_impl if config.gated_linear_unit
                           else bias_gelu_impl)

    def forward(self, permuted_tokens: torch.Tensor,
                tokens_per_expert: torch.Tensor):
        if _grouped_mm_available(permuted_tokens):
            offs = torch.cumsum(tokens_per_expert.to(
                device=permuted_tokens.device), 0).to(torch.int32)
            # host token counts enable the per-expert fp32 wgrad in
            # backward; the one-time .tolist() sync per layer is ~50 us
            # against the ~2 full expert-weight passes it removes
            dis = os.environ.get("MEGATRONAPP_DISABLE_FUSED", "")
            counts = None
            if (torch.is_grad_enabled() and _ops.have_ops()
                    and permuted_tokens.dtype == torch.bfloat16
                    and hasattr(self.weight1, "main_grad")
                    and not ("all" in dis or "moe_wgrad" in dis)):
                counts = tokens_per_expert.tolist()
            inter = _GroupedMMFn.apply(permuted_tokens, self.weight1, offs,
                                       counts)
            inter = self.activation(inter, None)
            return _GroupedMMFn.apply(inter, self.weight2, offs, counts)
        outputs = []
        start = 0
        for e, n in enumerate(tokens_per_expert.tolist()):
            seg = permuted_tokens[start:start + n]
            start += n
            if n == 0:
                outputs.append(seg)
                continue
            inter = torch.matmul(seg, self.weight1[e])
            inter = self.activation(inter, None)
            outputs.append(torch.matmul(inter, self.weight2[e]))
        return torch.cat(outputs, dim=0) if outputs else permuted_tokens
