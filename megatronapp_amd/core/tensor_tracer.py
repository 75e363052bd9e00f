"""MegaScope capture core: per-layer activation taps -> TP gather ->
compression -> report callback.

Reference: megatron/core/tensor_tracer.py (TTFlags :225, TensorTracers
:125, DefaultCompressor :76, tik_result :189, tik_end :212).  The wire
protocol (FlagType values 1-7, update messages) is kept byte-compatible
with the reference frontend.

Differences from the reference by design: the compressor config maps to a
named registry instead of ``eval`` of a config string (same configs, no
arbitrary code execution).
"""

from __future__ import annotations

import enum
from typing import Callable, Dict, List, Optional

import torch
import torch.distributed as dist

from . import parallel_state


class FlagType(enum.Enum):
    QKV = 1
    RawAttentionScore = 2
    ContextLayer = 3
    MLP1 = 4
    MLP2 = 5
    MLP2_Plot = 6
    Result = 7


class TTFlags:
    """Per-layer boolean switches for each FlagType."""

    _NAME_MAP = {
        "QKV_mat_mul": FlagType.QKV,
        "Raw_attention_score": FlagType.RawAttentionScore,
        "Context_layer": FlagType.ContextLayer,
        "MLP_1": FlagType.MLP1,
        "MLP_2": FlagType.MLP2,
        "MLP_2_plot": FlagType.MLP2_Plot,
    }

    def __init__(self, num_layers: int = 0):
        self.num_layers = num_layers
        self.flags: Dict[FlagType, List[bool]] = {
            ft: [False] * max(num_layers, 1) for ft in FlagType}

    def set_flag(self, ft: FlagType, layer_id: Optional[int], value: bool):
        if layer_id is None:
            self.flags[ft] = [value] * len(self.flags[ft])
        else:
            self.flags[ft][layer_id] = value

    def get_flag(self, ft: FlagType, layer_id: int) -> bool:
        lst = self.flags[ft]
        if layer_id >= len(lst):
            return False
        return lst[layer_id]

    def set_by_configs(self, configs: dict):
        """configs: {"QKV_mat_mul": "True"|"False"|[bool per layer], ...}."""
        for name, val in (configs or {}).items():
            ft = self._NAME_MAP.get(name)
            if ft is None:
                continue
            if isinstance(val, str):
                self.set_flag(ft, None, val == "True")
            elif isinstance(val, bool):
                self.set_flag(ft, None, val)
            elif isinstance(val, list):
                for i, v in enumerate(val):
                    self.set_flag(ft, i, bool(v) if not isinstance(v, str) else v == "True")


class DefaultCompressor:
    """Chunked mean reduction to ``pixels`` values per vector (reference
    DefaultCompressor :76-113)."""

    def __init__(self, pixels: int = 64, method: str = "mean"):
        self.pixels = pixels
        self.method = method

    def __call__(self, data: torch.Tensor) -> torch.Tensor:
        # data: [..., H] -> [..., pixels]
        H = data.shape[-1]
        if H <= self.pixels:
            return data.float()
        chunk = H // self.pixels
        trimmed = data[..., :chunk * self.pixels]
        shaped = trimmed.reshape(*data.shape[:-1], self.pixels, chunk).float()
        if self.method == "max":
            return shaped.max(dim=-1)[0]
        if self.method == "norm":
            return shaped.norm(dim=-1)
        return shaped.mean(dim=-1)


COMPRESSOR_REGISTRY = {"mean": "mean", "max": "max", "norm": "norm"}


class TensorTracers:
    """Gathers tapped activations across the TP group and reports them."""

    def __init__(self):
        self.tt_flags = TTFlags()
        self.report_func: Optional[Callable] = None
        self.compressors: Dict[str, DefaultCompressor] = {}
        self.tokenizer = None
        self._mlp2_record: List[torch.Tensor] = []

    # --- config plumbing -----------------------------------------------
    def set_report(self, fn: Callable):
        self.report_func = fn

    def set_num_layers(self, n: int):
        self.tt_flags = TTFlags(n)

    def set_compressor_configs(self, configs: dict):
        self.compressors = {}
        for key, cfg in (configs or {}).items():
            pixels = int(cfg.get("pixels", 64))
            method = COMPRESSOR_REGISTRY.get(cfg.get("method", "mean"), "mean")
            self.compressors[key] = DefaultCompressor(pixels, method)

    def _compressor_for(self, ft: FlagType) -> DefaultCompressor:
        key = {FlagType.QKV: "QKV", FlagType.MLP1: "MLP", FlagType.MLP2: "MLP",
               FlagType.ContextLayer: "QKV"}.get(ft, "QKV")
        return self.compressors.get(key, DefaultCompressor())

    def enabled(self, ft: FlagType, layer_id: int) -> bool:
        if self.report_func is None:
            return False
        return self.tt_flags.get_flag(ft, layer_id - 1 if layer_id > 0 else 0)

    # --- capture -------------------------------------------------------
    def _tp_gather_last_dim(self, t: torch.Tensor) -> torch.Tensor:
        world = parallel_state.get_tensor_model_parallel_world_size()
        if world == 1:
            return t
        group = parallel_state.get_tensor_model_parallel_group()
        gathered = [torch.empty_like(t) for _ in range(world)]
        dist.all_gather(gathered, t.contiguous(), group=group)
        return torch.cat(gathered, dim=-1)

    def report(self, ft: FlagType, layer_id: int, data):
        """Tap entry point called from attention/MLP forward."""
        if not self.enabled(ft, layer_id):
            return
        if parallel_state.get_tensor_model_parallel_rank() != 0 and \
                parallel_state.get_tensor_model_parallel_world_size() > 1:
            # non-src ranks only participate in the gather
            pass
        with torch.no_grad():
            if ft == FlagType.QKV:
                q, k, v = data
                full = torch.cat([
                    self._tp_gather_last_dim(x.flatten(start_dim=2)) for x in (q, k, v)
                ], dim=-1)
            elif ft == FlagType.RawAttentionScore:
                # [b, np, sq, sk] heads cat along dim 1 across TP
                world = parallel_state.get_tensor_model_parallel_world_size()
                if world > 1:
                    group = parallel_state.get_tensor_model_parallel_group()
                    gathered = [torch.empty_like(data) for _ in range(world)]
                    dist.all_gather(gathered, data.contiguous(), group=group)
                    full = torch.cat(gathered, dim=1)
                else:
                    full = data
                self._emit(ft, layer_id, full.float())
                return
            elif ft in (FlagType.MLP1, FlagType.MLP2, FlagType.MLP2_Plot,
                        FlagType.ContextLayer):
                full = self._tp_gather_last_dim(data)
            else:
                full = data
            if ft == FlagType.MLP2_Plot:
                self._mlp2_record.append(
                    full.float().reshape(-1, full.shape[-1]).mean(0).cpu())
                return
            compressed = self._compressor_for(ft)(full)
            self._emit(ft, layer_id, compressed)

    def _emit(self, ft: FlagType, layer_id: int, tensor: torch.Tensor):
        if self.report_func is None:
            return
        tensor = tensor.detach().float().cpu()
        self.report_func({
            "type": "update",
            "update_type": ft.value,
            "layer_id": layer_id,
            "args": list(tensor.shape),
            "result": tensor.flatten().tolist(),
        })

    # --- result-path helpers (reference tik_result :189, tik_end :212) ---
    def tik_result(self, logits: torch.Tensor, sampled_token=None, topk: int = 20):
        """Per-step top-k candidate report from last-position logits."""
        if self.report_func is None:
            return
        with torch.no_grad():
            probs = torch.softmax(logits.float(), dim=-1)
            top_p, top_i = probs.topk(topk, dim=-1)
            for b in range(top_p.shape[0]):
                cands = []
                for p, i in zip(top_p[b].tolist(), top_i[b].tolist()):
                    tok = (self.tokenizer.decoder.get(i, str(i))
                           if self.tokenizer is not None and
                           hasattr(self.tokenizer, "decoder") else str(i))
                    cands.append({"token_id": i, "token": tok, "prob": p})
                msg = {"type": "update", "update_type": FlagType.Result.value,
                       "batch": b, "result": cands}
                if sampled_token is not None:
                    msg["sampled"] = int(sampled_token[b])
                self.report_func(msg)

    def tik_end(self):
        """PCA(2) over accumulated MLP2 records for the PCA tab."""
        if not self._mlp2_record or self.report_func is None:
            self._mlp2_record = []
            return
        import numpy as np
        X = torch.stack(self._mlp2_record).numpy()
        self._mlp2_record = []
        X = (X - X.mean(0)) / (X.std(0) + 1e-6)
        # PCA via SVD (sklearn-free; identical up to sign)
        U, S, _ = np.linalg.svd(X - X.mean(0), full_matrices=False)
        pts = (U[:, :2] * S[:2]).tolist() if X.shape[0] >= 2 else [[0.0, 0.0]]
        self.report_func({"type": "update", "update_type": FlagType.MLP2_Plot.value,
                          "result": pts})


_TENSOR_TRACERS: Optional[TensorTracers] = None


def get_tensor_tracers() -> Optional[TensorTracers]:
    return _TENSOR_TRACERS


def enable_tensor_tracers() -> TensorTracers:
    global _TENSOR_TRACERS
    if _TENSOR_TRACERS is None:
        _TENSOR_TRACERS = TensorTracers()
    return _TENSOR_TRACERS


def get_tt_flags() -> Optional[TTFlags]:
    return _TENSOR_TRACERS.tt_flags if _TENSOR_TRACERS is not None else None
