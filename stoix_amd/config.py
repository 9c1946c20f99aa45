"""Hydra-style YAML config composition (no external deps beyond PyYAML).

The reference composes configs with Hydra:
/root/reference/stoix/configs/default/anakin/default_ff_ppo.yaml:1-11 declares
``defaults: [logger, arch, system, network, env]`` and the CLI overrides
groups (``env=gymnax/cartpole``) and leaves (``system.rollout_length=16``).
Hydra is not available offline, so this module implements the subset the
framework needs:

  * config groups as directories under ``stoix_amd/configs/``
  * a ``defaults:`` list in the root yaml, entries ``{group: name}``
  * group overrides ``group=name`` and leaf overrides ``a.b.c=value`` on the
    command line (or passed programmatically)
  * attribute access (``cfg.system.gamma``) on a plain dict subclass
"""
from __future__ import annotations

import os
from pathlib import Path
from typing import Any, Dict, Iterable, List, Optional

import yaml

CONFIG_ROOT = Path(__file__).parent / "configs"


class DotDict(dict):
    """dict with attribute access, recursive over nested dicts."""

    def __getattr__(self, k: str) -> Any:
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def __setattr__(self, k: str, v: Any) -> None:
        self[k] = v

    def __delattr__(self, k: str) -> None:
        del self[k]

    @staticmethod
    def wrap(obj: Any) -> Any:
        if isinstance(obj, dict):
            return DotDict({k: DotDict.wrap(v) for k, v in obj.items()})
        if isinstance(obj, list):
            return [DotDict.wrap(v) for v in obj]
        return obj

    def to_plain(self) -> dict:
        def rec(o: Any) -> Any:
            if isinstance(o, dict):
                return {k: rec(v) for k, v in o.items()}
            if isinstance(o, list):
                return [rec(v) for v in o]
            return o

        return rec(self)


def _deep_merge(base: dict, over: dict) -> dict:
    out = dict(base)
    for k, v in over.items():
        if k in out and isinstance(out[k], dict) and isinstance(v, dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = v
    return out


def _load_yaml(path: Path) -> dict:
    with open(path) as f:
        data = yaml.safe_load(f)
    return data or {}


def _parse_value(s: str) -> Any:
    """Parse a CLI override value with YAML semantics."""
    try:
        return yaml.safe_load(s)
    except yaml.YAMLError:
        return s


def _set_path(cfg: dict, dotted: str, value: Any) -> None:
    keys = dotted.split(".")
    cur = cfg
    for k in keys[:-1]:
        if k not in cur or not isinstance(cur[k], dict):
            cur[k] = DotDict()
        cur = cur[k]
    cur[keys[-1]] = DotDict.wrap(value)


def _resolve_group_file(root: Path, group: str, name: str) -> Path:
    """Resolve ``group=name`` to a yaml path; name may contain '/'."""
    p = root / group / f"{name}.yaml"
    if not p.exists():
        raise FileNotFoundError(f"config group '{group}' has no entry '{name}' (looked at {p})")
    return p


def compose(
    default: str,
    overrides: Optional[Iterable[str]] = None,
    config_root: Optional[Path] = None,
) -> DotDict:
    """Compose a config like hydra would.

    Args:
        default: path of the root yaml relative to the config root, e.g.
            ``default/anakin/default_ff_ppo.yaml``.
        overrides: list of strings: ``group=name`` (group switch) or
            ``a.b.c=value`` (leaf override; value parsed as YAML).
    """
    root = Path(config_root) if config_root else CONFIG_ROOT
    root_cfg = _load_yaml(root / default)
    defaults: List[Any] = root_cfg.pop("defaults", [])

    overrides = list(overrides or [])
    group_overrides: Dict[str, str] = {}
    leaf_overrides: List[tuple] = []
    for ov in overrides:
        if "=" not in ov:
            raise ValueError(f"override '{ov}' must be key=value")
        key, val = ov.split("=", 1)
        # A group override has no dot in the key AND names a group dir,
        # or is an explicit 'group/sub' style key that exists as a directory.
        if "." not in key and (root / key).is_dir():
            group_overrides[key] = val
        else:
            leaf_overrides.append((key, _parse_value(val)))

    cfg: dict = {}
    for entry in defaults:
        if isinstance(entry, str):
            if entry == "_self_":
                cfg = _deep_merge(cfg, root_cfg)
                continue
            # bare file include relative to root
            cfg = _deep_merge(cfg, _load_yaml(root / f"{entry}.yaml"))
            continue
        assert isinstance(entry, dict) and len(entry) == 1, f"bad defaults entry: {entry}"
        (group, name), = entry.items()
        name = group_overrides.pop(group, name)
        sub = _load_yaml(_resolve_group_file(root, group, str(name)))
        # Each group's yaml is merged under its group key unless the yaml
        # declares a '# @package _global_'-style root marker via a top-level
        # '_global_: true' key.
        if sub.pop("_global_", False):
            cfg = _deep_merge(cfg, sub)
        else:
            cfg = _deep_merge(cfg, {group: sub})

    if group_overrides:
        # group overrides for groups not in defaults: merge anyway
        for group, name in group_overrides.items():
            sub = _load_yaml(_resolve_group_file(root, group, name))
            if sub.pop("_global_", False):
                cfg = _deep_merge(cfg, sub)
            else:
                cfg = _deep_merge(cfg, {group: sub})

    cfg = _deep_merge(cfg, root_cfg)

    dcfg = DotDict.wrap(cfg)
    for key, val in leaf_overrides:
        _set_path(dcfg, key, val)
    return dcfg


def save_config(cfg: DotDict, path: str) -> None:
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "w") as f:
        yaml.safe_dump(cfg.to_plain() if isinstance(cfg, DotDict) else cfg, f, sort_keys=False)
