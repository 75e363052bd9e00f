"""YAML configuration front-end (reference training/yaml_arguments.py:1-458).

Lets a launch be described as a YAML file instead of several hundred CLI
flags.  Semantics:

* ``${ENV_VAR}`` inside any string value is interpolated from the
  environment (missing variables are an error).
* nested mappings are flattened — ``model_parallel.tensor_model_parallel_size``
  becomes ``tensor_model_parallel_size`` — so section headers are purely
  organisational, as in the reference.
* keys may use dashes or underscores.
* values override the argparse defaults; flags given explicitly on the
  command line override the YAML (CLI wins, matching the reference).

Use with ``--yaml-cfg path.yml`` (the flag is handled in
``arguments.parse_args``) or call :func:`load_yaml_config` directly.
"""

from __future__ import annotations

import os
import re
from typing import Dict

import yaml

_ENV_RE = re.compile(r"\$\{([A-Za-z_][A-Za-z0-9_]*)\}")


def _interp_env(value):
    if isinstance(value, str):
        def sub(m):
            name = m.group(1)
            if name not in os.environ:
                raise KeyError(f"environment variable {name} referenced in "
                               f"yaml config is not set")
            return os.environ[name]
        return _ENV_RE.sub(sub, value)
    return value


def _flatten(tree: dict, out: Dict[str, object]):
    for key, value in tree.items():
        key = str(key).replace("-", "_")
        if isinstance(value, dict):
            _flatten(value, out)
        else:
            out[key] = _interp_env(value)


def load_yaml_config(path: str) -> Dict[str, object]:
    """Parse a YAML config file into a flat {arg_name: value} dict."""
    with open(path) as f:
        tree = yaml.safe_load(f)
    if tree is None:
        return {}
    if not isinstance(tree, dict):
        raise ValueError(f"{path}: top level must be a mapping")
    flat: Dict[str, object] = {}
    _flatten(tree, flat)
    return flat


def apply_yaml_config(args, path: str, explicit: set = None):
    """Overlay YAML values onto an argparse Namespace in place.

    ``explicit`` is the set of dest names the user passed on the command
    line; those keep their CLI value.
    """
    explicit = explicit or set()
    flat = load_yaml_config(path)
    unknown = []
    for key, value in flat.items():
        if not hasattr(args, key):
            unknown.append(key)
            continue
        if key in explicit:
            continue
        setattr(args, key, value)
    if unknown:
        raise ValueError(f"yaml config {path} sets unknown arguments: "
                         f"{sorted(unknown)}")
    return args
