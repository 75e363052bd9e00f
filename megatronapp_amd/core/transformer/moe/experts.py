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

import torch
from torch import nn


def _grouped_mm_available(t: torch.Tensor) -> bool:
    return t.is_cuda and hasattr(torch, "_grouped_mm")


class _GroupedMMFn(torch.autograd.Function):
    """y[seg_e] = x[seg_e] @ w[e] for jagged segments given by offs
    (int32 inclusive cumsum of tokens_per_expert)."""

    @staticmethod
    def forward(ctx, x, w, offs):
        ctx.save_for_backward(x, w, offs)
        return torch._grouped_mm(x, w, offs=offs)

    @staticmethod
    def backward(ctx, dy):
        x, w, offs = ctx.saved_tensors
        dy = dy.contiguous()
        dx = torch._grouped_mm(dy, w.transpose(1, 2), offs=offs)
        # dw[e] = x[seg]^T @ dy[seg]: (2D, 2D) -> 3D grouped over tokens
        dw = torch._grouped_mm(x.t().contiguous(), dy, offs=offs)
        return dx, dw, None

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
            inter = _GroupedMMFn.apply(permuted_tokens, self.weight1, offs)
            inter = self.activation(inter, None)
            return _GroupedMMFn.apply(inter, self.weight2, offs)
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
