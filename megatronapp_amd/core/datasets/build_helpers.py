"""Build/load the native dataset index helpers (_helpers.so).

Reference compiles helpers.cpp on the fly at initialize (initialize.py:
177-188); we build in-tree so the .so ships to the GPU box with the
snapshot.  Pure CPU code — g++ only.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "_helpers.so")
_SRC = os.path.join(_DIR, "csrc", "helpers.cpp")


def build_helpers(force: bool = False) -> str:
    if os.path.exists(_SO) and not force and \
            os.path.getmtime(_SO) >= os.path.getmtime(_SRC):
        return _SO
    import pybind11
    cmd = ["g++", "-O3", "-shared", "-std=c++17", "-fPIC",
           f"-I{pybind11.get_include()}",
           f"-I{sysconfig.get_paths()['include']}",
           _SRC, "-o", _SO]
    subprocess.check_call(cmd)
    return _SO


def load_helpers():
    """Import the extension, building it if needed; returns the module."""
    try:
        from . import _helpers  # type: ignore
        return _helpers
    except ImportError:
        pass
    build_helpers()
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "megatronapp_amd.core.datasets._helpers", _SO)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    sys.modules["megatronapp_amd.core.datasets._helpers"] = mod
    return mod
