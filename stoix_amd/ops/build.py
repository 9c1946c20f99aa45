"""Build the gfx950 HIP extension in-tree.

Uses torch.utils.cpp_extension with PYTORCH_ROCM_ARCH=gfx950; the build
directory is inside the repo (stoix_amd/ops/build/) so the produced .so
travels to the GPU box with the gpurun snapshot (a JIT cache under ~/.cache
would not). hipcc cross-compiles gfx950 without a GPU present.
"""
from __future__ import annotations

import os
from pathlib import Path

HERE = Path(__file__).parent
CSRC = HERE / "csrc"
BUILD_DIR = HERE / "build"

SOURCES = [
    str(CSRC / "bind.cpp"),
    str(CSRC / "envs.hip"),
    str(CSRC / "scan.hip"),
    str(CSRC / "optim.hip"),
    str(CSRC / "mlp.hip"),
    str(CSRC / "wgrad.hip"),
    str(CSRC / "per.hip"),
    str(CSRC / "rnn.hip"),
    str(CSRC / "snake.hip"),
]


def build(verbose: bool = False):
    """Compile (if stale) and return the loaded module."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    BUILD_DIR.mkdir(exist_ok=True)
    sources = [s for s in SOURCES if os.path.exists(s)]
    mod = load(
        name="stoix_amd_C",
        sources=sources,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17", "--offload-arch=gfx950"],
        build_directory=str(BUILD_DIR),
        verbose=verbose,
        with_cuda=True,
    )
    return mod


if __name__ == "__main__":
    build(verbose=True)
    print("stoix_amd HIP extension built OK")
