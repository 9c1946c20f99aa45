"""Versioned checkpointing of learner state.

Parity with /root/reference/stoix/utils/checkpointing.py (:17 version, save
:99-148, best-model retention by episode_return :66-73, restore-by-template
:166-179, major-version compatibility gate :153-158). Backed by
``safetensors`` for tensors + a JSON metadata sidecar (same logical format:
pytree-of-tensors + versioned metadata, as BASELINE.json requires).
"""
from __future__ import annotations

import json
import os
import shutil
import time
from typing import Any, Dict, Optional

import torch

from stoix_amd.types import tree_flatten

CHECKPOINTER_VERSION = "2.0"


def _flatten_state(state: Any) -> Dict[str, torch.Tensor]:
    flat = tree_flatten(state)
    return {k: v.detach().cpu().contiguous() for k, v in flat.items()}


class Checkpointer:
    """Save/restore checkpoints keyed by env-step, with max_to_keep and
    best-by-metric retention."""

    def __init__(
        self,
        model_name: str,
        metadata: Optional[dict] = None,
        directory: str = "checkpoints",
        max_to_keep: Optional[int] = 1,
        keep_period: Optional[int] = None,
        save_best: bool = True,
        best_metric: str = "episode_return",
    ):
        self.directory = os.path.join(directory, model_name)
        os.makedirs(self.directory, exist_ok=True)
        self.max_to_keep = max_to_keep
        self.keep_period = keep_period
        self.save_best = save_best
        self.best_metric = best_metric
        self._best_value = float("-inf")
        self._aux: Optional[dict] = None
        self.metadata = dict(metadata or {})
        self.metadata["checkpointer_version"] = CHECKPOINTER_VERSION
        meta_path = os.path.join(self.directory, "metadata.json")
        # a loader-only Checkpointer (no metadata given) must NOT clobber
        # the saved run's metadata (and with it the version gate)
        if metadata is not None or not os.path.exists(meta_path):
            with open(meta_path, "w") as f:
                json.dump(_jsonable(self.metadata), f, indent=2)

    # ------------------------------------------------------------------ save

    def stage_aux(self, aux: Optional[dict]) -> None:
        """Attach non-template state (optimizer state_dicts, counters) to
        the NEXT save; written as aux.pt next to the safetensors payload.
        The reference checkpoints the whole learner state incl. opt_state
        (checkpointing.py save of the unreplicated LearnerState); template
        restore cannot rebuild a fresh optimizer's empty state, so aux
        rides alongside and restores through Optimizer.load_state_dict."""
        self._aux = aux


    def save(self, timestep: int, state: Any, metric_value: Optional[float] = None) -> str:
        from safetensors.torch import save_file

        path = os.path.join(self.directory, f"step_{timestep}")
        os.makedirs(path, exist_ok=True)
        flat = _flatten_state(state)
        save_file(flat, os.path.join(path, "state.safetensors"))
        with open(os.path.join(path, "info.json"), "w") as f:
            json.dump(
                {
                    "timestep": timestep,
                    "time": time.time(),
                    "metric": metric_value,
                    "checkpointer_version": CHECKPOINTER_VERSION,
                },
                f,
            )
        if self._aux is not None:
            torch.save(self._aux, os.path.join(path, "aux.pt"))
            self._aux = None
        if self.save_best and metric_value is not None and metric_value >= self._best_value:
            self._best_value = metric_value
            best = os.path.join(self.directory, "best")
            if os.path.islink(best) or os.path.exists(best):
                shutil.rmtree(best, ignore_errors=True)
            shutil.copytree(path, best)
        self._gc(timestep)
        return path

    def _gc(self, latest: int) -> None:
        if self.max_to_keep is None:
            return
        steps = sorted(
            int(d.split("_", 1)[1])
            for d in os.listdir(self.directory)
            if d.startswith("step_")
        )
        excess = steps[: max(0, len(steps) - self.max_to_keep)]
        for s in excess:
            if self.keep_period and s % self.keep_period == 0:
                continue
            shutil.rmtree(os.path.join(self.directory, f"step_{s}"), ignore_errors=True)

    # --------------------------------------------------------------- restore

    def restore_params(self, template: Any, timestep: Optional[int] = None, best: bool = False) -> Any:
        """Load tensors back into the *structure* of ``template`` (reference
        restore-by-template semantics, checkpointing.py:166-179)."""
        from safetensors.torch import load_file

        meta_path = os.path.join(self.directory, "metadata.json")
        if os.path.exists(meta_path):
            with open(meta_path) as f:
                meta = json.load(f)
            saved_ver = str(meta.get("checkpointer_version", "0.0"))
            if saved_ver.split(".")[0] != CHECKPOINTER_VERSION.split(".")[0]:
                raise ValueError(
                    f"checkpoint major version {saved_ver} incompatible with {CHECKPOINTER_VERSION}"
                )
        if best:
            path = os.path.join(self.directory, "best")
        else:
            if timestep is None:
                steps = sorted(
                    int(d.split("_", 1)[1])
                    for d in os.listdir(self.directory)
                    if d.startswith("step_")
                )
                if not steps:
                    raise FileNotFoundError(f"no checkpoints in {self.directory}")
                timestep = steps[-1]
            path = os.path.join(self.directory, f"step_{timestep}")
        flat = load_file(os.path.join(path, "state.safetensors"))
        return _unflatten_into(template, flat)

    def restore_aux(self, timestep: Optional[int] = None, best: bool = False) -> Optional[dict]:
        """Load the aux.pt (optimizer state etc.) saved next to a
        checkpoint, or None if that checkpoint has no aux payload."""
        if best:
            path = os.path.join(self.directory, "best")
        else:
            if timestep is None:
                steps = sorted(
                    int(d.split("_", 1)[1])
                    for d in os.listdir(self.directory)
                    if d.startswith("step_")
                )
                if not steps:
                    raise FileNotFoundError(f"no checkpoints in {self.directory}")
                timestep = steps[-1]
            path = os.path.join(self.directory, f"step_{timestep}")
        aux_path = os.path.join(path, "aux.pt")
        if not os.path.exists(aux_path):
            return None
        return torch.load(aux_path, weights_only=False)


def _unflatten_into(template: Any, flat: Dict[str, torch.Tensor], prefix: str = "") -> Any:
    import dataclasses

    if isinstance(template, torch.Tensor):
        t = flat[prefix]
        return t.to(device=template.device, dtype=template.dtype)
    if isinstance(template, dict):
        return {
            k: _unflatten_into(v, flat, f"{prefix}.{k}" if prefix else str(k))
            for k, v in template.items()
        }
    if isinstance(template, tuple) and hasattr(template, "_fields"):
        return type(template)(
            *(
                _unflatten_into(v, flat, f"{prefix}.{k}" if prefix else k)
                for k, v in zip(template._fields, template)
            )
        )
    if isinstance(template, (list, tuple)):
        return type(template)(
            _unflatten_into(v, flat, f"{prefix}.{i}" if prefix else str(i))
            for i, v in enumerate(template)
        )
    if dataclasses.is_dataclass(template) and not isinstance(template, type):
        return type(template)(
            **{
                f.name: _unflatten_into(getattr(template, f.name), flat, f"{prefix}.{f.name}" if prefix else f.name)
                for f in dataclasses.fields(template)
            }
        )
    return template


def _jsonable(obj: Any) -> Any:
    if isinstance(obj, dict):
        return {k: _jsonable(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [_jsonable(v) for v in obj]
    if isinstance(obj, (str, int, float, bool)) or obj is None:
        return obj
    return str(obj)
