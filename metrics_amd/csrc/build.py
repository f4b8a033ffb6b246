"""Build the in-tree HIP kernel library for gfx950.

The library is a plain HIP shared object (no torch C++ ABI linkage): Python
binds it with ctypes and passes raw device pointers + the current torch HIP
stream. That keeps the build a single fast hipcc invocation that
cross-compiles on CPU-only boxes, and the resulting .so lives in-tree
(metrics_amd/_lib/) so it travels with the repo snapshot to GPU machines.

Usage: python -m metrics_amd.csrc.build
"""
from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

CSRC = Path(__file__).resolve().parent
LIBDIR = CSRC.parent / "_lib"
LIBNAME = "libmetrics_hip.so"
ARCH = os.environ.get("METRICS_AMD_ARCH", "gfx950")


def build(verbose: bool = True) -> Path:
    """Compile csrc/*.hip (gfx950) and csrc/*.cpp (host) into metrics_amd/_lib/."""
    LIBDIR.mkdir(exist_ok=True)
    out = LIBDIR / LIBNAME
    srcs = sorted(str(p) for p in CSRC.glob("*.hip"))
    cmd = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        *srcs,
        "-o",
        str(out),
    ]
    if verbose:
        print("+", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)

    # host-side native components (COCO matcher): plain g++, no GPU linkage
    cpp_srcs = sorted(str(p) for p in CSRC.glob("*.cpp"))
    if cpp_srcs:
        cpu_out = LIBDIR / "libmetrics_cpu.so"
        cmd = ["g++", "-O3", "-std=c++17", "-fPIC", "-shared", "-fopenmp", *cpp_srcs, "-o", str(cpu_out)]
        if verbose:
            print("+", " ".join(cmd), file=sys.stderr)
        subprocess.run(cmd, check=True)
    return out


def cpu_lib_path() -> Path:
    return LIBDIR / "libmetrics_cpu.so"


def lib_path() -> Path:
    return LIBDIR / LIBNAME


def is_built() -> bool:
    return lib_path().exists()


if __name__ == "__main__":
    build()
