"""In-tree build of the gfx950 HIP extension.

Builds ``_hip_ops.so`` under ``gpuschedule_amd/ops/_build/`` via
torch.utils.cpp_extension (which drives hipcc).  The .so lives in-tree
so the gpurun snapshot carries it to the GPU box; hipcc cross-compiles
fine on a GPU-less machine.

Usage: ``python -m gpuschedule_amd.ops.build`` or ``build_ext()``.
"""
from __future__ import annotations

import os
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
BUILD_DIR = os.path.join(HERE, "_build")
SOURCES = [os.path.join(HERE, "hip", f)
           for f in ("bindings.cpp", "norms.hip", "gemm_mfma.hip",
                     "optim.hip", "attention.hip", "bn.hip", "ce.hip",
                     "elementwise.hip", "embedding.hip")]
EXT_NAME = "_hip_ops"


def build_ext(verbose: bool = False):
    """Compile (if needed) and return the extension module."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load
    sources = [s for s in SOURCES if os.path.exists(s)]
    mod = load(
        name=EXT_NAME,
        sources=sources,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
    )
    return mod


if __name__ == "__main__":
    build_ext(verbose="-q" not in sys.argv)
    print(f"built {EXT_NAME} -> {BUILD_DIR}")
