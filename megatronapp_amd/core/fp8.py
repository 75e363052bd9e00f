"""fp8 (OCP e4m3) TRAINING GEMMs for gfx950 — native replacement for the
reference's TransformerEngine fp8 autocast (reference core/fp8_utils.py,
--fp8-format).

Recipe ("hybrid" in reference terms, adapted to what hipBLASLt's scaled
GEMM expresses):
  forward : y  = x8 @ w8^T     x per-token rows, W per-output-row scales
  dgrad   : dx = dy8 @ w8c     dy per-token rows, W per-INPUT-channel
                               scales (a second cached quantization of W
                               along dim 1, column-major for _scaled_mm)
  wgrad   : fp8 e4m3           both operands transpose-quantized with
                               per-tensor scales, fp32 GEMM output added
                               into the fp32 main_grad (fp8_wgrad;
                               MEGATRONAPP_DISABLE_FUSED=fp8_wgrad
                               falls back to the exact bf16 wgrad).

Weight quantizations are cached per optimizer step (`bump_step()` is
called from DistributedOptimizer.step); activations/grads quantize
dynamically per call.  Measured peak: _scaled_mm 2249 TF vs 1199 bf16
(tools/probe_fp8.py on MI355X).
"""

from __future__ import annotations

import torch

FP8_MAX = 448.0
_STEP = [0]


def bump_step() -> None:
    """Invalidate per-step weight quantization caches (optimizer step)."""
    _STEP[0] += 1


def _quant_rows(t2d: torch.Tensor):
    """Per-row e4m3 quantization of a 2D tensor — one fused HIP kernel
    on GPU (the eager 6-kernel version was 24% of fp8 step time)."""
    if t2d.is_cuda and t2d.dtype == torch.bfloat16:
        from .. import ops as _ops
        if _ops.have_ops() and hasattr(_ops.get_ops(), "quantize_rows_e4m3"):
            q, s = _ops.get_ops().quantize_rows_e4m3(t2d.contiguous())
            return q, s
    s = t2d.abs().amax(dim=1, keepdim=True).float() / FP8_MAX
    s = torch.clamp(s, min=1e-12)
    q = (t2d.float() / s).clamp(-FP8_MAX, FP8_MAX).to(torch.float8_e4m3fn)
    return q, s


def _weight_cache(weight: torch.Tensor):
    c = getattr(weight, "_fp8t", None)
    if c is None or c["step"] != _STEP[0]:
        w8, ws = _quant_rows(weight.data)          # [N,K] rows (outputs)
        wt8, wts = _quant_rows(weight.data.t().contiguous())  # [K,N] rows
        c = {"step": _STEP[0], "w8": w8, "ws": ws,
             "w8c": wt8.t(), "wcs": wts.t()}       # [N,K] col-major + [1,N]
        weight._fp8t = c
    return c


def fp8_train_enabled(weight: torch.Tensor, x: torch.Tensor) -> bool:
    return (getattr(weight, "_fp8_train", False) and x.is_cuda
            and x.dtype == torch.bfloat16
            and weight.shape[0] % 16 == 0 and weight.shape[1] % 16 == 0)


def fp8_forward(x: torch.Tensor, weight: torch.Tensor,
                bias) -> torch.Tensor:
    """y[..., N] = x[..., K] @ W[N, K]^T in e4m3."""
    c = _weight_cache(weight)
    shp = x.shape
    x2 = x.reshape(-1, shp[-1])
    x8, xs = _quant_rows(x2)
    y = torch._scaled_mm(x8, c["w8"].t(), scale_a=xs, scale_b=c["ws"].t(),
                         out_dtype=x.dtype)
    if bias is not None:
        y = y + bias
    return y.reshape(*shp[:-1], weight.shape[0])


def fp8_dgrad(dy: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """dx[..., K] = dy[..., N] @ W[N, K] in e4m3 (per-input-channel W
    scales so the rowwise scaled-GEMM pattern stays legal)."""
    c = _weight_cache(weight)
    shp = dy.shape
    dy2 = dy.reshape(-1, shp[-1])
    dy8, ds = _quant_rows(dy2)
    dx = torch._scaled_mm(dy8, c["w8c"], scale_a=ds, scale_b=c["wcs"],
                          out_dtype=dy.dtype)
    return dx.reshape(*shp[:-1], weight.shape[1])


def fp8_wgrad_enabled() -> bool:
    import os
    return "fp8_wgrad" not in os.environ.get("MEGATRONAPP_DISABLE_FUSED", "")


def fp8_wgrad(g2d: torch.Tensor, x2d: torch.Tensor,
              main_grad: torch.Tensor) -> bool:
    """main_grad[out, in] += g2d[rows, out]^T @ x2d[rows, in] through an
    fp8 e4m3 GEMM: both operands transpose-quantized (token dim
    innermost, per-tensor just-in-time scales), fp32 _scaled_mm output
    added into main_grad.  The per-tensor quantization of the GRADIENT
    operand is the accuracy risk — guarded by the fp8 convergence test
    (tests/test_families_gpu.py --fp8 twin).  Returns False when the
    path does not apply (caller falls back to the exact bf16 wgrad)."""
    from .. import ops as _ops
    rows = g2d.shape[0]
    if not (g2d.is_cuda and g2d.dtype == torch.bfloat16
            and main_grad.dtype == torch.float32 and _ops.have_ops()
            and hasattr(_ops.get_ops(), "quantize_transpose_e4m3")
            # _scaled_mm wants 16-aligned dims
            and rows % 16 == 0 and g2d.shape[1] % 16 == 0
            and x2d.shape[1] % 16 == 0
            and fp8_wgrad_enabled()):
        return False
    lt = _ops.get_ops()
    qg, sg = lt.quantize_transpose_e4m3(g2d)   # [out, rows]
    qx, sx = lt.quantize_transpose_e4m3(x2d)   # [in, rows]
    out = torch._scaled_mm(qg, qx.t(), scale_a=sg, scale_b=sx,
                           out_dtype=torch.float32)
    main_grad.add_(out)
    return True


def enable_fp8_training(model: torch.nn.Module) -> int:
    """Flag every TP linear weight for fp8 GEMMs; returns count."""
    from .tensor_parallel.layers import (ColumnParallelLinear,
                                         RowParallelLinear)
    n = 0
    for mod in model.modules():
        if isinstance(mod, (ColumnParallelLinear, RowParallelLinear)) and \
                getattr(mod, "weight", None) is not None:
            mod.weight._fp8_train = True
            n += 1
    return n
