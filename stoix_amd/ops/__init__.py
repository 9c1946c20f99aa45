"""stoix_amd.ops — hand-written gfx950 kernels with torch reference paths.

Dispatch policy (the driver's "native code must be the path that runs"
contract): on a CUDA/HIP device the extension is REQUIRED — if the built
.so is missing, importing the ops module on a GPU box raises rather than
silently falling back to eager PyTorch. On CPU the torch reference
implementations run (they are also the numerics oracles for the kernels,
tests/test_gpu_ops.py).
"""
from __future__ import annotations

import os
import sys
from typing import Optional

import torch

from stoix_amd.ops import losses, multistep, running_statistics  # noqa: F401

_EXT = None
_EXT_ERR: Optional[str] = None


def _try_load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    from pathlib import Path

    so = Path(__file__).parent / "build" / "stoix_amd_C.so"
    try:
        if so.exists():
            import importlib.util

            spec = importlib.util.spec_from_file_location("stoix_amd_C", so)
            mod = importlib.util.module_from_spec(spec)
            # torch extension modules need torch symbols loaded first
            import torch  # noqa: F811

            spec.loader.exec_module(mod)  # type: ignore[union-attr]
            _EXT = mod
            return _EXT
        # fall back to (re)building via cpp_extension load (dev path)
        from stoix_amd.ops.build import build

        _EXT = build()
        return _EXT
    except Exception as e:  # pragma: no cover
        _EXT_ERR = f"{type(e).__name__}: {e}"
        return None


def ext(required: bool = False):
    """Return the compiled extension module.

    required=True (the GPU hot path) raises loudly when unavailable —
    a GPU run must never silently drop to the eager fallback.
    """
    mod = _try_load_ext()
    if mod is None and required:
        raise RuntimeError(
            "stoix_amd HIP extension not available on a GPU device "
            f"(build error: {_EXT_ERR}). Run `python -m stoix_amd.ops.build` "
            "before GPU execution."
        )
    return mod


def have_ext() -> bool:
    return _try_load_ext() is not None
