"""megatronapp_amd.ops — hand-written CDNA4 (gfx950) HIP kernels.

The extension is built IN-TREE (``python -m megatronapp_amd.ops.setup`` or
``__graft_entry__.build()``) to ``megatronapp_amd/ops/_C.so`` so it ships
with the repo snapshot to GPU boxes.

Dispatch policy (no dual CUDA/HIP paths — HIP is *the* GPU path):
* tensor on GPU  -> HIP kernel; a missing extension raises loudly rather
  than silently falling back to slow eager ops.
* tensor on CPU  -> plain fp32 torch reference implementation (the same
  one the numerics tests compare the kernels against).
"""

from __future__ import annotations

import os

import torch

_C = None
_LOAD_ERROR = None


def _try_load():
    global _C, _LOAD_ERROR
    if _C is not None:
        return _C
    try:
        import importlib.util
        so_path = os.path.join(os.path.dirname(__file__), "_C.so")
        if not os.path.exists(so_path):
            raise ImportError(f"{so_path} not built")
        spec = importlib.util.spec_from_file_location(
            "megatronapp_amd.ops._C", so_path)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _C = mod
    except Exception as e:  # noqa: BLE001
        _LOAD_ERROR = e
        _C = None
    return _C


def have_ops() -> bool:
    if os.environ.get("MEGATRONAPP_FORCE_EAGER") == "1":
        return False
    return _try_load() is not None


def get_ops():
    """Return the native module; raise with context if unavailable on GPU."""
    mod = _try_load()
    if mod is None:
        raise RuntimeError(
            "megatronapp_amd HIP extension (_C.so) is not built/loadable "
            f"(load error: {_LOAD_ERROR}). On a GPU box this is a hard "
            "error: build it with `python -m megatronapp_amd.ops.setup` "
            "or __graft_entry__.build().")
    return mod


def fused_enabled(t, group: str) -> bool:
    """Gate for bf16 fused-kernel autograd paths.  MEGATRONAPP_DISABLE_FUSED
    can name comma-separated groups (softmax, norms, bias_act, rope, all)
    to force the torch fallback — used for GPU numerics bisection."""
    import torch
    if not (t.is_cuda and t.dtype == torch.bfloat16):
        return False
    dis = os.environ.get("MEGATRONAPP_DISABLE_FUSED", "")
    if dis and ("all" in dis or group in dis):
        return False
    return have_ops()
