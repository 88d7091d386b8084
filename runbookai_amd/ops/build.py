"""Build the gfx950 HIP extension IN-TREE.

The built .so lands in runbookai_amd/ops/_build/ so it travels with the
gpurun snapshot (a JIT cache under ~/.cache would not). hipcc
cross-compiles for gfx950 without a GPU present.
"""
from __future__ import annotations

import os
import sys

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
BUILD_DIR = os.path.join(OPS_DIR, "_build")
EXT_NAME = "runbookai_hip_ops"

SOURCES = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "norm_elem.hip"),
    os.path.join(CSRC, "attention.hip"),
    os.path.join(CSRC, "flash_prefill.hip"),
    os.path.join(CSRC, "flash_prefill2.hip"),
    os.path.join(CSRC, "skinny_gemm.hip"),
    os.path.join(CSRC, "decode_gemv.hip"),
    os.path.join(CSRC, "retrieval_sampling.hip"),
]


def build(verbose: bool = False):
    """Compile (or reuse) the extension; returns the imported module."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    module = load(
        name=EXT_NAME,
        sources=SOURCES,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        with_cuda=True,  # ROCm build: drives hipcc for the .hip sources
    )
    return module


def load_prebuilt():
    """Import a previously-built .so without invoking the compiler."""
    import importlib.util

    for fn in os.listdir(BUILD_DIR) if os.path.isdir(BUILD_DIR) else []:
        if fn.startswith(EXT_NAME) and fn.endswith(".so"):
            spec = importlib.util.spec_from_file_location(EXT_NAME,
                                                          os.path.join(BUILD_DIR, fn))
            mod = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(mod)
            return mod
    return None


if __name__ == "__main__":
    build(verbose=True)
    print("built", EXT_NAME, "->", BUILD_DIR)
