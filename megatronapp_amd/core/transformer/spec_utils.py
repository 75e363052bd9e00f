"""ModuleSpec system: declarative layer composition.

Equivalent of the reference's spec system (spec_utils.py:106,
build_module) — a spec names a module class plus per-submodule specs, so
model families swap attention/MLP/norm implementations (torch fallback vs
HIP-fused) without subclassing.
"""

from __future__ import annotations

import types
from dataclasses import dataclass, field
from typing import Optional, Union


@dataclass
class ModuleSpec:
    module: Union[type, types.FunctionType, None] = None
    params: dict = field(default_factory=dict)
    submodules: Optional[object] = None


class IdentityOp:
    """Placeholder module slot: returns its input unchanged."""

    def __init__(self, *args, **kwargs):
        pass

    def __call__(self, x, *args, **kwargs):
        return x


class IdentityFuncOp(IdentityOp):
    """Placeholder for a function-returning slot."""

    def __call__(self, *args, **kwargs):
        return super().__call__


def import_module(path: str):
    import importlib
    mod, _, name = path.rpartition(".")
    return getattr(importlib.import_module(mod), name)


def build_module(spec_or_module, *args, **kwargs):
    """Instantiate a ModuleSpec (or pass a class/instance through)."""
    if spec_or_module is None:
        return None
    if isinstance(spec_or_module, types.FunctionType):
        return spec_or_module
    if isinstance(spec_or_module, ModuleSpec):
        if isinstance(spec_or_module.module, types.FunctionType):
            return spec_or_module.module
        module = spec_or_module.module
        if isinstance(module, str):
            module = import_module(module)
        merged = {**spec_or_module.params, **kwargs}
        if spec_or_module.submodules is not None:
            merged["submodules"] = spec_or_module.submodules
        return module(*args, **merged)
    if isinstance(spec_or_module, type):
        return spec_or_module(*args, **kwargs)
    # already an instance
    return spec_or_module
