"""Low-overhead tracer hook points used by core hot paths.

The MegaScan tracer (megatronapp_amd/training/trace.py) registers itself
here at startup; until then every hook is a no-op with near-zero cost.
Keeping the hook indirection in core (instead of importing training from
core) avoids a layering cycle — reference equivalent: the
``get_tracer().scope(...)`` sites listed in SURVEY.md §2.2.
"""

from __future__ import annotations

import contextlib

_TRACER = None


def register_tracer(tracer) -> None:
    global _TRACER
    _TRACER = tracer


def get_tracer():
    return _TRACER


_NULL = contextlib.nullcontext()


def trace_scope(name: str, **attrs):
    """Scope context manager; no-op when tracing is off."""
    if _TRACER is None or not _TRACER.is_tracing_active():
        return _NULL
    return _TRACER.scope(name, **attrs)


def trace_collective(name: str, tensor, group_ranks):
    """Scope for a collective: records byte count + peer ranks."""
    if _TRACER is None or not _TRACER.is_tracing_active():
        return _NULL
    return _TRACER.scope(name, data=tensor.numel() * tensor.element_size(),
                         group=group_ranks)


def trace_instant(name: str, **attrs) -> None:
    if _TRACER is not None and _TRACER.is_tracing_active():
        _TRACER.instant(name, **attrs)
