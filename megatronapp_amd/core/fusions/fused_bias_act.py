"""Fused bias + activation kernels (GeLU / SwiGLU / squared-ReLU).

Reference: core/fusions/fused_bias_gelu.py, fused_bias_swiglu.py (torch-JIT
there; hand HIP elementwise here — ops/csrc/elementwise.hip, bf16x8
vectorised, HBM-bound).
"""

from __future__ import annotations

import math

import torch

from ... import ops as _ops


def _gelu_tanh(x):
    return 0.5 * x * (1.0 + torch.tanh(0.7978845608028654 * (x + 0.044715 * x * x * x)))


def _gelu_tanh_grad(x):
    t = torch.tanh(0.7978845608028654 * (x + 0.044715 * x * x * x))
    return 0.5 * (1.0 + t) + 0.5 * x * (1.0 - t * t) * 0.7978845608028654 * (
        1.0 + 3 * 0.044715 * x * x)


def _bias_grad(bias, dx):
    """Bias grad for a fused activation: colsum into main_grad on GPU
    (with a dummy grad so the DDP post-accumulate hook still fires),
    eager sum otherwise."""
    if bias is None or not bias.requires_grad:
        return None
    if (dx.is_cuda and dx.dtype == torch.bfloat16
            and _ops.have_ops() and hasattr(bias, "main_grad")
            and hasattr(bias, "grad_added_to_main_grad")):
        _ops.get_ops().colsum_accum(
            dx.reshape(-1, dx.shape[-1]).contiguous(), bias.main_grad)
        bias.grad_added_to_main_grad = True
        return torch.empty(bias.shape, dtype=bias.dtype, device=bias.device)
    return dx.reshape(-1, dx.shape[-1]).sum(0).to(bias.dtype)


class _BiasGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        ctx.save_for_backward(x, bias)
        if _ops.fused_enabled(x, "bias_act"):
            return _ops.get_ops().bias_gelu_fwd(x, bias)
        xf = (x.float() + bias.float()) if bias is not None else x.float()
        return _gelu_tanh(xf).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        if _ops.fused_enabled(dy, "bias_act"):
            dx = _ops.get_ops().bias_gelu_bwd(dy.contiguous(), x, bias)
        else:
            xf = (x.float() + bias.float()) if bias is not None else x.float()
            dx = (dy.float() * _gelu_tanh_grad(xf)).to(x.dtype)
        dbias = _bias_grad(bias, dx)
        return dx, dbias


class _BiasSwigluFn(torch.autograd.Function):
    """y = silu(x1 + b1) * (x2 + b2) on interleaved halves [..., 2F]."""

    @staticmethod
    def forward(ctx, x, bias):
        ctx.save_for_backward(x, bias)
        if _ops.fused_enabled(x, "bias_act"):
            return _ops.get_ops().bias_swiglu_fwd(x, bias)
        xf = (x.float() + bias.float()) if bias is not None else x.float()
        x1, x2 = xf.chunk(2, dim=-1)
        return (torch.nn.functional.silu(x1) * x2).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        if _ops.fused_enabled(dy, "bias_act"):
            dx = _ops.get_ops().bias_swiglu_bwd(dy.contiguous(), x, bias)
        else:
            xf = (x.float() + bias.float()) if bias is not None else x.float()
            x1, x2 = xf.chunk(2, dim=-1)
            sig = torch.sigmoid(x1)
            silu = x1 * sig
            dyf = dy.float()
            d1 = dyf * x2 * (sig * (1 + x1 * (1 - sig)))
            d2 = dyf * silu
            dx = torch.cat([d1, d2], dim=-1).to(x.dtype)
        dbias = _bias_grad(bias, dx)
        return dx, dbias


class _BiasGegluFn(torch.autograd.Function):
    """y = gelu(x1 + b1) * (x2 + b2) on interleaved halves [..., 2F]
    (reference fused_bias_geglu.py)."""

    @staticmethod
    def forward(ctx, x, bias):
        ctx.save_for_backward(x, bias)
        if _ops.fused_enabled(x, "bias_act"):
            return _ops.get_ops().bias_geglu_fwd(x, bias)
        xf = (x.float() + bias.float()) if bias is not None else x.float()
        x1, x2 = xf.chunk(2, dim=-1)
        return (torch.nn.functional.gelu(x1, approximate="tanh") * x2
                ).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        if _ops.fused_enabled(dy, "bias_act"):
            dx = _ops.get_ops().bias_geglu_bwd(dy.contiguous(), x, bias)
        else:
            xf = (x.float() + bias.float()) if bias is not None else x.float()
            x1, x2 = xf.chunk(2, dim=-1)
            c0, c1 = 0.7978845608028654, 0.044715
            t = torch.tanh(c0 * (x1 + c1 * x1 ** 3))
            g = 0.5 * x1 * (1 + t)
            dgelu = 0.5 * (1 + t) + \
                0.5 * x1 * (1 - t * t) * c0 * (1 + 3 * c1 * x1 ** 2)
            dyf = dy.float()
            dx = torch.cat([dyf * x2 * dgelu, dyf * g], dim=-1).to(x.dtype)
        dbias = _bias_grad(bias, dx)
        return dx, dbias


class _SquaredReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        ctx.save_for_backward(x, bias)
        xf = (x.float() + bias.float()) if bias is not None else x.float()
        return torch.relu(xf).pow(2).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        xf = (x.float() + bias.float()) if bias is not None else x.float()
        dx = (dy.float() * 2.0 * torch.relu(xf)).to(x.dtype)
        dbias = dx.reshape(-1, dx.shape[-1]).sum(0).to(bias.dtype) \
            if bias is not None and bias.requires_grad else None
        return dx, dbias


def bias_gelu_impl(x, bias=None):
    return _BiasGeluFn.apply(x, bias)


def bias_swiglu_impl(x, bias=None):
    return _BiasSwigluFn.apply(x, bias)


def bias_geglu_impl(x, bias=None):
    return _BiasGegluFn.apply(x, bias)


def bias_squared_relu_impl(x, bias=None):
    return _SquaredReluFn.apply(x, bias)
