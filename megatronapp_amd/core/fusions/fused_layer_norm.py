"""Fused LayerNorm / RMSNorm (HIP kernels, reference fused_layer_norm.py).

GPU path: one-pass Welford/−sum-of-squares HIP kernels (ops/csrc/norms.hip)
vectorised bf16x8, one workgroup per row group.  CPU path: fp32 torch
reference (the numerics oracle the GPU tests compare against).
"""

from __future__ import annotations

import torch
from torch import nn

from ... import ops as _ops


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        if _ops.fused_enabled(x, "norms"):
            x = x.contiguous()
            y, invrms = _ops.get_ops().rmsnorm_fwd(x, weight, eps)
        else:
            xf = x.float()
            invrms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
            y = (xf * invrms * weight.float()).to(x.dtype)
            invrms = invrms.squeeze(-1)
        ctx.save_for_backward(x, weight, invrms)
        ctx.eps = eps
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, invrms = ctx.saved_tensors
        if _ops.fused_enabled(dy, "norms"):
            if hasattr(weight, "main_grad") and \
                    hasattr(weight, "grad_added_to_main_grad"):
                # dw accumulates straight into the DDP fp32 grad buffer;
                # autograd gets a dummy so the bucket hook still fires
                dx, _ = _ops.get_ops().rmsnorm_bwd(
                    dy.contiguous(), x, weight, invrms, weight.main_grad)
                weight.grad_added_to_main_grad = True
                dw = torch.empty(weight.shape, dtype=weight.dtype,
                                 device=weight.device)
                return dx, dw, None
            dx, dw = _ops.get_ops().rmsnorm_bwd(dy.contiguous(), x, weight, invrms)
            return dx, dw.to(weight.dtype), None
        xf = x.float()
        dyf = dy.float()
        wf = weight.float()
        r = invrms.unsqueeze(-1)
        xhat = xf * r
        H = x.shape[-1]
        dxhat = dyf * wf
        dx = r * (dxhat - xhat * (dxhat * xhat).mean(-1, keepdim=True))
        dw = (dyf * xhat).reshape(-1, H).sum(0)
        return dx.to(x.dtype), dw.to(weight.dtype), None


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ctx.bias_param = bias
        if _ops.fused_enabled(x, "norms"):
            x = x.contiguous()
            y, mean, invstd = _ops.get_ops().layernorm_fwd(x, weight, bias, eps)
        else:
            xf = x.float()
            mean = xf.mean(-1)
            var = xf.var(-1, unbiased=False)
            invstd = torch.rsqrt(var + eps)
            xhat = (xf - mean.unsqueeze(-1)) * invstd.unsqueeze(-1)
            y = (xhat * weight.float() + bias.float()).to(x.dtype)
        ctx.save_for_backward(x, weight, mean, invstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, invstd = ctx.saved_tensors
        bias = ctx.bias_param
        if _ops.fused_enabled(dy, "norms"):
            if hasattr(weight, "main_grad") and bias is not None and \
                    hasattr(bias, "main_grad") and \
                    hasattr(weight, "grad_added_to_main_grad"):
                dx, _, _ = _ops.get_ops().layernorm_bwd(
                    dy.contiguous(), x, weight, mean, invstd,
                    weight.main_grad, bias.main_grad)
                weight.grad_added_to_main_grad = True
                bias.grad_added_to_main_grad = True
                dw = torch.empty(weight.shape, dtype=weight.dtype,
                                 device=weight.device)
                db = torch.empty(bias.shape, dtype=bias.dtype,
                                 device=bias.device)
                return dx, dw, db, None
            dx, dw, db = _ops.get_ops().layernorm_bwd(
                dy.contiguous(), x, weight, mean, invstd)
            return dx, dw.to(weight.dtype), db.to(weight.dtype), None
        xf = x.float()
        dyf = dy.float()
        wf = weight.float()
        H = x.shape[-1]
        xhat = (xf - mean.unsqueeze(-1)) * invstd.unsqueeze(-1)
        dxhat = dyf * wf
        dx = invstd.unsqueeze(-1) * (
            dxhat - dxhat.mean(-1, keepdim=True)
            - xhat * (dxhat * xhat).mean(-1, keepdim=True))
        dw = (dyf * xhat).reshape(-1, H).sum(0)
        db = dyf.reshape(-1, H).sum(0)
        return dx.to(x.dtype), dw.to(weight.dtype), db.to(weight.dtype), None


class _RMSNormResidualFn(torch.autograd.Function):
    """norm(x) plus a passthrough of x.  TransformerLayer uses the
    passthrough as the residual, so the residual branch's grad arrives
    here and is folded into the norm backward's dx INSIDE the kernel —
    removing the standalone [rows, H] grad-join add per norm."""

    @staticmethod
    def forward(ctx, x, weight, eps):
        if _ops.fused_enabled(x, "norms"):
            x = x.contiguous()
            y, invrms = _ops.get_ops().rmsnorm_fwd(x, weight, eps)
        else:
            xf = x.float()
            invrms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
            y = (xf * invrms * weight.float()).to(x.dtype)
            invrms = invrms.squeeze(-1)
        ctx.save_for_backward(x, weight, invrms)
        return y, x.view_as(x)

    @staticmethod
    def backward(ctx, dy, dres):
        x, weight, invrms = ctx.saved_tensors
        if _ops.fused_enabled(dy, "norms"):
            mg = (weight.main_grad
                  if hasattr(weight, "main_grad")
                  and hasattr(weight, "grad_added_to_main_grad") else None)
            dx, dw_out = _ops.get_ops().rmsnorm_bwd(
                dy.contiguous(), x, weight, invrms, mg,
                dres.contiguous() if dres is not None else None)
            if mg is not None:
                weight.grad_added_to_main_grad = True
                dw = torch.empty(weight.shape, dtype=weight.dtype,
                                 device=weight.device)
            else:
                dw = dw_out.to(weight.dtype)
            return dx, dw, None
        dx, dw, _ = _RMSNormFn.backward(ctx, dy)
        if dres is not None:
            dx = dx + dres
        return dx, dw, None


class _LayerNormResidualFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ctx.bias_param = bias
        if _ops.fused_enabled(x, "norms"):
            x = x.contiguous()
            y, mean, invstd = _ops.get_ops().layernorm_fwd(x, weight, bias,
                                                           eps)
        else:
            xf = x.float()
            mean = xf.mean(-1)
            invstd = torch.rsqrt(xf.var(-1, unbiased=False) + eps)
            xhat = (xf - mean.unsqueeze(-1)) * invstd.unsqueeze(-1)
            y = (xhat * weight.float() + bias.float()).to(x.dtype)
        ctx.save_for_backward(x, weight, mean, invstd)
        return y, x.view_as(x)

    @staticmethod
    def backward(ctx, dy, dres):
        x, weight, mean, invstd = ctx.saved_tensors
        bias = ctx.bias_param
        if _ops.fused_enabled(dy, "norms"):
            fuse_wgrad = (hasattr(weight, "main_grad") and bias is not None
                          and hasattr(bias, "main_grad")
                          and hasattr(weight, "grad_added_to_main_grad"))
            mg_w = weight.main_grad if fuse_wgrad else None
            mg_b = bias.main_grad if fuse_wgrad else None
            dx, dw_out, db_out = _ops.get_ops().layernorm_bwd(
                dy.contiguous(), x, weight, mean, invstd, mg_w, mg_b,
                dres.contiguous() if dres is not None else None)
            if fuse_wgrad:
                weight.grad_added_to_main_grad = True
                bias.grad_added_to_main_grad = True
                dw = torch.empty(weight.shape, dtype=weight.dtype,
                                 device=weight.device)
                db = torch.empty(bias.shape, dtype=bias.dtype,
                                 device=bias.device)
            else:
                dw = dw_out.to(weight.dtype)
                db = db_out.to(weight.dtype)
            return dx, dw, db, None
        dx, dw, db, _ = _LayerNormFn.backward(ctx, dy)
        if dres is not None:
            dx = dx + dres
        return dx, dw, db, None


class FusedLayerNorm(nn.Module):
    """LayerNorm over the hidden dim; weights marked sequence_parallel so
    finalize_model_grads all-reduces their grads across TP when SP is on."""

    def __init__(self, config, hidden_size: int, eps: float = 1e-5, **kwargs):
        super().__init__()
        self.config = config
        self.hidden_size = hidden_size
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden_size, dtype=config.params_dtype))
        self.bias = nn.Parameter(torch.zeros(hidden_size, dtype=config.params_dtype))
        if config.sequence_parallel:
            self.weight.sequence_parallel = True
            self.bias.sequence_parallel = True

    def forward(self, x):
        return _LayerNormFn.apply(x, self.weight, self.bias, self.eps)

    def forward_with_residual(self, x):
        """(norm(x), residual passthrough) with the residual grad fused
        into the norm backward."""
        return _LayerNormResidualFn.apply(x, self.weight, self.bias,
                                          self.eps)


class FusedRMSNorm(nn.Module):
    def __init__(self, config, hidden_size: int, eps: float = 1e-5, **kwargs):
        super().__init__()
        self.config = config
        self.hidden_size = hidden_size
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden_size, dtype=config.params_dtype))
        if config.sequence_parallel:
            self.weight.sequence_parallel = True

    def forward(self, x):
        return _RMSNormFn.apply(x, self.weight, self.eps)

    def forward_with_residual(self, x):
        return _RMSNormResidualFn.apply(x, self.weight, self.eps)


def get_norm_cls(normalization: str):
    if normalization == "RMSNorm":
        return FusedRMSNorm
    return FusedLayerNorm


class WrappedTorchNorm(FusedLayerNorm):
    """Alias kept for reference-API parity (core/transformer/torch_norm.py)."""
