"""Dump resolved configs to a directory (reference core/config_logger.py).

Enabled with --config-logger-dir; every dataclass config and the argparse
namespace are serialized once at startup so a run's effective settings are
inspectable after the fact.
"""

from __future__ import annotations

import dataclasses
import json
import os


def _jsonable(v):
    if isinstance(v, (int, float, str, bool, type(None))):
        return v
    if isinstance(v, (list, tuple)):
        return [_jsonable(x) for x in v]
    if isinstance(v, dict):
        return {str(k): _jsonable(x) for k, x in v.items()}
    return repr(v)


def log_config_to_dir(directory: str, rank: int = 0, **named_configs):
    """Write each named config (dataclass or Namespace) as JSON."""
    if not directory or rank != 0:
        return
    os.makedirs(directory, exist_ok=True)
    for name, cfg in named_configs.items():
        if cfg is None:
            continue
        if dataclasses.is_dataclass(cfg):
            data = dataclasses.asdict(cfg)
        elif hasattr(cfg, "__dict__"):
            data = vars(cfg)
        else:
            data = {"value": cfg}
        with open(os.path.join(directory, f"{name}.json"), "w") as f:
            json.dump(_jsonable(data), f, indent=2, sort_keys=True)
