"""Fused scale + mask + softmax (reference fused_softmax.py:12-96 +
FusedScaleMaskSoftmax:97).

GPU path: wave64 HIP kernels (ops/csrc/softmax.hip) — causal
(upper-triangular) and generic-mask variants, fp32 accumulation in
registers/LDS, bf16/fp16 I/O.  Used by the "fused" attention backend;
the flash backend fuses softmax into the attention kernel itself.
"""

from __future__ import annotations

import torch
from torch import nn

from ...core.enums import AttnMaskType
from ... import ops as _ops


def _sk_ok(t):
    # the wave softmax kernels vectorize rows as short8: sk % 8 == 0
    return t.shape[-1] % 8 == 0


class ScaledUpperTriangMaskedSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inputs, scale):
        if _ops.fused_enabled(inputs, "softmax") and _sk_ok(inputs):
            probs = _ops.get_ops().scaled_upper_triang_masked_softmax_fwd(
                inputs, scale)
        else:
            b, sq, sk = inputs.shape
            x = inputs.float() * scale
            mask = torch.triu(torch.ones(sq, sk, dtype=torch.bool,
                                         device=inputs.device), diagonal=1)
            x = x.masked_fill(mask, float("-inf"))
            probs = torch.softmax(x, dim=-1).to(inputs.dtype)
        ctx.save_for_backward(probs)
        ctx.scale = scale
        return probs

    @staticmethod
    def backward(ctx, dy):
        (probs,) = ctx.saved_tensors
        if _ops.fused_enabled(dy, "softmax") and _sk_ok(dy):
            # causal-aware bwd: only the valid row prefix is read
            dx = _ops.get_ops().scaled_upper_triang_masked_softmax_bwd(
                dy.contiguous(), probs, ctx.scale)
        else:
            dyf = dy.float()
            p = probs.float()
            dx = (p * (dyf - (dyf * p).sum(-1, keepdim=True)) * ctx.scale).to(dy.dtype)
        return dx, None


class ScaledMaskedSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, inputs, mask, scale):
        if _ops.fused_enabled(inputs, "softmax") and _sk_ok(inputs):
            if mask is None:
                probs = _ops.get_ops().scaled_softmax_fwd(inputs, scale)
            else:
                probs = _ops.get_ops().scaled_masked_softmax_fwd(inputs, mask,
                                                                 scale)
        else:
            x = inputs.float() * scale
            if mask is not None:
                x = x.masked_fill(mask, -10000.0)
            probs = torch.softmax(x, dim=-1).to(inputs.dtype)
        ctx.save_for_backward(probs)
        ctx.scale = scale
        return probs

    @staticmethod
    def backward(ctx, dy):
        (probs,) = ctx.saved_tensors
        if _ops.fused_enabled(dy, "softmax") and _sk_ok(dy):
            dx = _ops.get_ops().scaled_softmax_bwd(dy.contiguous(), probs, ctx.scale)
        else:
            dyf = dy.float()
            p = probs.float()
            dx = (p * (dyf - (dyf * p).sum(-1, keepdim=True)) * ctx.scale).to(dy.dtype)
        return dx, None, None


class FusedScaleMaskSoftmax(nn.Module):
    """Dispatcher mirroring reference FusedScaleMaskSoftmax:97."""

    def __init__(self, input_in_fp16, input_in_bf16, attn_mask_type,
                 scaled_masked_softmax_fusion, mask_func, softmax_in_fp32,
                 scale):
        super().__init__()
        self.input_in_float16 = input_in_fp16 or input_in_bf16
        self.attn_mask_type = attn_mask_type
        self.fusion = scaled_masked_softmax_fusion
        self.mask_func = mask_func
        self.softmax_in_fp32 = softmax_in_fp32
        self.scale = scale

    def forward(self, input, mask, attn_mask_type=None):
        # input: [b, np, sq, sk]; attn_mask_type overrides the constructed
        # default (decode steps pass no_mask: the single query row attends
        # to the whole KV prefix)
        mask_type = attn_mask_type if attn_mask_type is not None \
            else self.attn_mask_type
        scale = self.scale if self.scale is not None else 1.0
        if self.fusion and self.input_in_float16:
            b, np_, sq, sk = input.shape
            if mask_type == AttnMaskType.causal and sq == sk:
                probs = ScaledUpperTriangMaskedSoftmax.apply(
                    input.view(-1, sq, sk), scale)
                return probs.view(b, np_, sq, sk)
            return ScaledMaskedSoftmax.apply(input, mask, scale)
        # unfused fallback
        orig_dtype = input.dtype
        if self.input_in_float16 and self.softmax_in_fp32:
            input = input.float()
        if self.scale is not None:
            input = input * self.scale
        if mask_type == AttnMaskType.causal and mask is None:
            sq, sk = input.shape[-2], input.shape[-1]
            # causal with a KV prefix: query row i is global position
            # i + (sk - sq)
            mask = torch.triu(torch.ones(sq, sk, dtype=torch.bool,
                                         device=input.device),
                              diagonal=1 + sk - sq)
            mask = mask.view(1, 1, sq, sk)
        if mask is not None:
            input = self.mask_func(input, mask)
        probs = torch.softmax(input, dim=-1)
        if probs.dtype != orig_dtype:
            probs = probs.to(orig_dtype)
        return probs
