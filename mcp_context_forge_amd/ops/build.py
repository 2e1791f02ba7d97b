"""Build the in-tree HIP kernel library for gfx950.

Direct hipcc (no torch-extension ABI): kernels take raw pointers + a
hipStream_t, Python binds via ctypes on torch tensors' data_ptr(), launching
on torch's current HIP stream — zero-copy, no sync. The .so is built
IN-TREE so the gpurun snapshot ships it.
"""

from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

CSRC = Path(__file__).parent / "csrc"
LIB = Path(__file__).parent / "libforge_hip.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")

SOURCES = ["scan.hip", "featurize.hip", "json_guard.hip", "gemm_bf16.hip", "runtime.hip",
           "gemm_v2.hip", "envelope.cpp", "upstream.cpp", "fastpath.cpp", "rewrite.cpp", "pool.cpp"]


PYBRIDGE = Path(__file__).parent / "forge_pybridge.so"
EDGE = Path(__file__).parent / "forge_edge.so"
HEY = Path(__file__).parent / "forge_hey"


def needs_build() -> bool:
    if not LIB.exists():
        return True
    lib_mtime = LIB.stat().st_mtime
    for f in SOURCES + ["common.h"]:
        if (CSRC / f).stat().st_mtime > lib_mtime:
            return True
    return False


def build_pybridge(force: bool = False, verbose: bool = True) -> Path:
    """CPython C extension for the response-assembly hot loops (plain gcc,
    no hip): forge_pybridge.so in-tree (ships with the gpurun snapshot)."""
    src = CSRC / "pybridge.c"
    if not force and PYBRIDGE.exists() and PYBRIDGE.stat().st_mtime > src.stat().st_mtime:
        return PYBRIDGE
    import sysconfig

    cc = os.environ.get("CC", "gcc")
    cmd = [cc, "-O2", "-fPIC", "-shared", f"-I{sysconfig.get_paths()['include']}",
           "-o", str(PYBRIDGE), str(src)]
    if verbose:
        print("[forge-pybridge]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return PYBRIDGE


def build_edge(force: bool = False, verbose: bool = True) -> Path:
    """Native epoll HTTP edge (CPython extension, plain g++ — no HIP):
    forge_edge.so in-tree so the gpurun snapshot ships it."""
    src = CSRC / "edge.cpp"
    if not force and EDGE.exists() and EDGE.stat().st_mtime > src.stat().st_mtime:
        return EDGE
    import sysconfig

    cxx = os.environ.get("CXX", "g++")
    cmd = [cxx, "-O2", "-std=c++17", "-fPIC", "-shared", "-pthread",
           f"-I{sysconfig.get_paths()['include']}", "-o", str(EDGE), str(src)]
    if verbose:
        print("[forge-edge]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return EDGE


def build_hey(force: bool = False, verbose: bool = True) -> Path:
    """Native closed-loop HTTP load generator (hey analog) used by bench.py."""
    src = CSRC / "forge_hey.cpp"
    if not force and HEY.exists() and HEY.stat().st_mtime > src.stat().st_mtime:
        return HEY
    cxx = os.environ.get("CXX", "g++")
    cmd = [cxx, "-O2", "-std=c++17", "-pthread", "-o", str(HEY), str(src)]
    if verbose:
        print("[forge-hey]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return HEY


def build(force: bool = False, verbose: bool = True) -> Path:
    build_pybridge(force=force, verbose=verbose)
    build_edge(force=force, verbose=verbose)
    build_hey(force=force, verbose=verbose)
    if not force and not needs_build():
        return LIB
    cmd = [
        HIPCC,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-o",
        str(LIB),
    ] + [str(CSRC / f) for f in SOURCES]
    if verbose:
        print("[forge-hip]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return LIB


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(LIB)
