"""Build the native gfx950 GPU-validator extension in-tree.

hipcc cross-compiles for gfx950 on CPU-only machines (no GPU required to
build); the resulting ``_gpu_validator.so`` lives next to this file so it
ships with the package to GPU nodes.
"""

from __future__ import annotations

import os
import subprocess
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
SOURCE = os.path.join(PKG_DIR, "gpu_validator.hip")
OUTPUT = os.path.join(PKG_DIR, "_gpu_validator.so")
JSONOPS_SOURCE = os.path.join(PKG_DIR, "jsonops.cpp")
JSONOPS_OUTPUT = os.path.join(PKG_DIR, "_jsonops.so")

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
CXX = os.environ.get("CXX", "g++")


def is_built() -> bool:
    return os.path.exists(OUTPUT) and os.path.getmtime(OUTPUT) >= os.path.getmtime(SOURCE)


def build_jsonops(force: bool = False, verbose: bool = False) -> str:
    """Compile the native JSON-object accelerators (plain C++, no ROCm)."""
    if (
        not force
        and os.path.exists(JSONOPS_OUTPUT)
        and os.path.getmtime(JSONOPS_OUTPUT) >= os.path.getmtime(JSONOPS_SOURCE)
    ):
        return JSONOPS_OUTPUT
    cmd = [
        CXX, "-O3", "-std=c++17", "-fPIC", "-shared",
        f"-I{sysconfig.get_paths()['include']}",
        JSONOPS_SOURCE, "-o", JSONOPS_OUTPUT,
    ]
    if verbose:
        print(" ".join(cmd))  # noqa: T201 (CLI/build output)
    subprocess.run(cmd, check=True, capture_output=not verbose)
    return JSONOPS_OUTPUT


def build(force: bool = False, verbose: bool = False) -> str:
    """Compile gpu_validator.hip -> _gpu_validator.so for gfx950 (and the
    plain-C++ _jsonops accelerator)."""
    build_jsonops(force=force, verbose=verbose)
    if is_built() and not force:
        return OUTPUT
    import pybind11

    cmd = [
        HIPCC,
        "--offload-arch=gfx950",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        f"-I{sysconfig.get_paths()['include']}",
        f"-I{pybind11.get_include()}",
        SOURCE,
        "-o",
        OUTPUT,
    ]
    if verbose:
        print(" ".join(cmd))  # noqa: T201 (CLI/build output)
    subprocess.run(cmd, check=True, capture_output=not verbose)
    return OUTPUT


if __name__ == "__main__":
    build(force=True, verbose=True)
    print(f"built {OUTPUT}")  # noqa: T201 (CLI/build output)
