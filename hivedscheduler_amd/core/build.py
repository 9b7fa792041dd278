"""In-tree build of the C++ scheduler core (hivedcore pybind11 extension).

The .so is built into the package directory so it travels with repo snapshots
(gpurun) and is found by plain `import hivedscheduler_amd`.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

HERE = os.path.dirname(os.path.abspath(__file__))
PKG_DIR = os.path.dirname(HERE)

SOURCES = [
    "cells.cpp",
    "topo_sched.cpp",
    "build.cpp",
    "alloc.cpp",
    "algorithm.cpp",
    "debug.cpp",
    "bindings.cpp",
]


def ext_path() -> str:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return os.path.join(PKG_DIR, "hivedcore" + suffix)


def needs_rebuild() -> bool:
    out = ext_path()
    if not os.path.exists(out):
        return True
    out_mtime = os.path.getmtime(out)
    deps = [os.path.join(HERE, s) for s in SOURCES] + [os.path.join(HERE, "core.hpp")]
    return any(os.path.getmtime(d) > out_mtime for d in deps)


def build(force: bool = False, verbose: bool = True) -> str:
    out = ext_path()
    if not force and not needs_rebuild():
        return out
    import pybind11

    py_include = sysconfig.get_paths()["include"]
    cmd = [
        os.environ.get("CXX", "g++"),
        "-O2",
        "-g",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        f"-I{py_include}",
        f"-I{pybind11.get_include()}",
        f"-I{HERE}",
    ] + [os.path.join(HERE, s) for s in SOURCES] + ["-o", out]
    if verbose:
        print("[hivedcore build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
