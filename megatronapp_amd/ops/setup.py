"""In-tree build of the CDNA4 HIP extension.

Usage:  python -m megatronapp_amd.ops.setup
Builds megatronapp_amd/ops/_C.so for gfx950 (cross-compiles fine on a
box with no GPU).  The .so is git-ignored but ships to GPU boxes with
the repo snapshot.
"""

from __future__ import annotations

import os
import shutil
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")

SOURCES = [
    "bindings.cpp",
    "norms.hip",
    "elementwise.hip",
    "rope.hip",
    "softmax.hip",
    "adam.hip", "scan.hip", "ce.hip", "fp8quant.hip", "moe.hip", "dropout.hip",
    "wgrad.cpp",
    "attention.hip",
]


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load

    sources = [os.path.join(CSRC, s) for s in SOURCES
               if os.path.exists(os.path.join(CSRC, s))]
    build_dir = os.path.join(HERE, "build")
    os.makedirs(build_dir, exist_ok=True)
    mod = load(
        name="_C",
        sources=sources,
        build_directory=build_dir,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950", "-std=c++17"],
        extra_ldflags=["-L/opt/rocm/lib", "-lhipblaslt"],
        verbose=verbose,
        is_python_module=True,
        keep_intermediates=True,
    )
    so_path = os.path.join(build_dir, "_C.so")
    dst = os.path.join(HERE, "_C.so")
    shutil.copyfile(so_path, dst)
    return dst


if __name__ == "__main__":
    print(build())
