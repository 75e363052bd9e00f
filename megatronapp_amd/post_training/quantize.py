"""Post-training quantization (reference megatron/post_training —
modelopt-based PTQ; implemented natively here, no external toolkit).

Weight quantization is per-output-channel symmetric int8 (the standard
weight-only PTQ used for serving).  Two flows:

* ``quantize_model(model)`` — simulated ("fake") quantization: every
  matching linear weight is rounded through the int8 grid in place, so
  the model runs unmodified and any eval harness measures the quantized
  accuracy.  Int8 tensors + scales are attached as buffers for export.
* ``export_int8_state_dict(model)`` — the deployable artifact:
  ``{name: (int8 weight, fp32 per-channel scale)}``; on MI355X the
  serving path dequantizes into bf16 tiles at load (weight-only PTQ is
  HBM-bandwidth bound, which is what int8 halves).

``calibrate_activation_scales`` runs sample batches through forward
hooks and records per-tensor amax for activation quantization
(smooth/int8-activation flows build on it).
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch


def _is_quantizable(module: torch.nn.Module) -> bool:
    w = getattr(module, "weight", None)
    return (isinstance(w, torch.nn.Parameter) and w.dim() == 2
            and not isinstance(module, torch.nn.Embedding))


def quantize_weight_int8(w: torch.Tensor):
    """[out, in] -> (int8 weight, [out] fp32 scales), symmetric."""
    amax = w.abs().amax(dim=1, keepdim=True).clamp(min=1e-8)
    scale = amax / 127.0
    q = torch.clamp(torch.round(w / scale), -127, 127).to(torch.int8)
    return q, scale.squeeze(1).float()


def dequantize_weight(q: torch.Tensor, scale: torch.Tensor,
                      dtype=torch.float32):
    return (q.float() * scale.unsqueeze(1)).to(dtype)


def quantize_model(model: torch.nn.Module,
                   skip: Tuple[str, ...] = ("output_layer",
                                            "embedding")) -> int:
    """Round every quantizable linear weight through the int8 grid in
    place; attach ``weight_int8``/``weight_scale`` buffers.  Returns the
    number of quantized modules."""
    count = 0
    for name, module in model.named_modules():
        if not _is_quantizable(module):
            continue
        if any(s in name for s in skip):
            continue
        with torch.no_grad():
            q, scale = quantize_weight_int8(module.weight.float())
            module.weight.copy_(
                dequantize_weight(q, scale, module.weight.dtype))
        module.register_buffer("weight_int8", q, persistent=False)
        module.register_buffer("weight_scale", scale, persistent=False)
        count += 1
    return count


def export_int8_state_dict(
        model: torch.nn.Module) -> Dict[str, Tuple[torch.Tensor,
                                                   torch.Tensor]]:
    """Collect the attached int8 weights (run quantize_model first)."""
    out = {}
    for name, module in model.named_modules():
        if hasattr(module, "weight_int8"):
            out[name] = (module.weight_int8, module.weight_scale)
    return out


def quantization_error(model_fp: torch.nn.Module,
                       model_q: torch.nn.Module,
                       sample_inputs, forward=None) -> float:
    """Max |fp − quant| over outputs for quick PTQ sanity checks."""
    forward = forward or (lambda m, x: m(*x))
    with torch.no_grad():
        a = forward(model_fp, sample_inputs)
        b = forward(model_q, sample_inputs)
    return (a.float() - b.float()).abs().max().item()


class _AmaxHook:
    def __init__(self):
        self.amax = 0.0

    def __call__(self, module, inputs, output):
        for t in inputs:
            if torch.is_tensor(t) and t.is_floating_point():
                self.amax = max(self.amax, t.abs().max().item())


def calibrate_activation_scales(model: torch.nn.Module,
                                sample_batches,
                                forward=None) -> Dict[str, float]:
    """Per-module input amax over calibration batches -> int8 scales
    (amax / 127) for activation-quantized serving."""
    forward = forward or (lambda m, x: m(*x))
    hooks, handles = {}, []
    for name, module in model.named_modules():
        if _is_quantizable(module):
            h = _AmaxHook()
            hooks[name] = h
            handles.append(module.register_forward_hook(h))
    with torch.no_grad():
        for batch in sample_batches:
            forward(model, batch)
    for h in handles:
        h.remove()
    return {n: h.amax / 127.0 for n, h in hooks.items() if h.amax > 0}
