"""Build the native CPU envpool extension in-tree (envs/build/).

Plain C++ (no HIP) — compiles with the host toolchain anywhere, including
GPU boxes and CPU-only CI. Built .so lives in-tree so it travels with the
repo snapshot.
"""
from __future__ import annotations

import os
from pathlib import Path

HERE = Path(__file__).parent
CSRC = HERE / "csrc"
BUILD_DIR = HERE / "build"


def build(verbose: bool = False):
    from torch.utils.cpp_extension import load

    os.environ.setdefault("MAX_JOBS", "4")
    BUILD_DIR.mkdir(exist_ok=True)
    return load(
        name="stoix_amd_envpool",
        sources=[str(CSRC / "envpool_cpu.cpp"), str(CSRC / "envpool_games2.cpp")],
        extra_cflags=["-O3", "-std=c++17"],
        build_directory=str(BUILD_DIR),
        verbose=verbose,
        with_cuda=False,
    )


def load_built(verbose: bool = False):
    return build(verbose=verbose)


if __name__ == "__main__":
    build(verbose=True)
    print("stoix_amd envpool (CPU) extension built OK")
