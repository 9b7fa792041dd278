"""In-tree build of the HIP health-probe extension for gfx950.

Uses torch.utils.cpp_extension (drives hipcc) with PYTORCH_ROCM_ARCH=gfx950.
The built .so lives in ops/_build/ so repo snapshots carry it to GPU boxes.
"""
from __future__ import annotations

import os
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
BUILD_DIR = os.path.join(HERE, "_build")
SOURCES = [os.path.join(HERE, "hived_ops.hip")]


def build(verbose: bool = True):
    """Compile (if needed) and import the hived_ops extension module."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    return load(
        name="hived_ops",
        sources=SOURCES,
        build_directory=BUILD_DIR,
        extra_cflags=["-O3"],
        verbose=verbose,
    )


if __name__ == "__main__":
    mod = build()
    print("built:", mod.__file__, file=sys.stderr)
