"""Core types for the MI355X-native RL engine.

Mirrors the semantics of the reference's base types
(/root/reference/stoix/base_types.py:32-220 and the external `stoa` env
protocol) but is designed around batched torch tensors
resident on one GPU (struct-of-arrays), not JAX pytrees.

Conventions (load-bearing for GAE correctness — reference
stoix/systems/ppo/anakin/ff_ppo.py:107-116):
  * termination  => ``discount == 0`` and ``step_type == TERMINATED``
  * truncation   => ``discount == 1`` and ``step_type == TRUNCATED``
  * autoreset: when an episode ends, ``step()`` returns the *reset*
    observation of the new episode as ``observation`` and the true final
    observation in ``extras["next_obs"]``.
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, NamedTuple, Tuple

import torch

Tensor = torch.Tensor
Params = Dict[str, torch.Tensor]


class StepType:
    """Integer step-type codes (stored in a uint8 tensor)."""

    FIRST = 0
    MID = 1
    TERMINATED = 2
    TRUNCATED = 3


@dataclass
class TimeStep:
    """One batched environment transition (struct of arrays, batch-first).

    Semantics of the reference's ``stoa.TimeStep``.
    """

    step_type: Tensor  # [B] uint8
    reward: Tensor  # [B] float
    discount: Tensor  # [B] float; 0 at termination, 1 at truncation
    observation: Tensor  # [B, *obs_shape] (or dict for composite obs)
    extras: Dict[str, Any] = field(default_factory=dict)

    def first(self) -> Tensor:
        return self.step_type == StepType.FIRST

    def mid(self) -> Tensor:
        return self.step_type == StepType.MID

    def last(self) -> Tensor:
        """True where the episode ended (terminated OR truncated)."""
        return self.step_type >= StepType.TERMINATED

    def terminated(self) -> Tensor:
        return self.step_type == StepType.TERMINATED

    def truncated(self) -> Tensor:
        return self.step_type == StepType.TRUNCATED

    # done as every Anakin system consumes it: ff_ppo.py:107-108
    def done(self) -> Tensor:
        return self.discount == 0.0

    def to(self, device: torch.device) -> "TimeStep":
        return TimeStep(
            step_type=self.step_type.to(device),
            reward=self.reward.to(device),
            discount=self.discount.to(device),
            observation=_map_obs(self.observation, lambda t: t.to(device)),
            extras={k: _maybe_to(v, device) for k, v in self.extras.items()},
        )


def _map_obs(obs: Any, fn: Callable[[Tensor], Tensor]) -> Any:
    if isinstance(obs, torch.Tensor):
        return fn(obs)
    if isinstance(obs, dict):
        return {k: _map_obs(v, fn) for k, v in obs.items()}
    return obs


def _maybe_to(v: Any, device: torch.device) -> Any:
    if isinstance(v, torch.Tensor):
        return v.to(device)
    if isinstance(v, dict):
        return {k: _maybe_to(x, device) for k, x in v.items()}
    return v


class Transition(NamedTuple):
    """Generic off-policy transition (reference dqn_types.py:9-15)."""

    obs: Any
    action: Tensor
    reward: Tensor
    done: Tensor
    next_obs: Any
    info: Dict[str, Any]


class PPOTransition(NamedTuple):
    """On-policy transition (reference ppo_types.py:9-20)."""

    done: Tensor
    truncated: Tensor
    action: Tensor
    value: Tensor
    reward: Tensor
    bootstrap_value: Tensor
    log_prob: Tensor
    obs: Any
    info: Dict[str, Any]


class OnlineAndTarget(NamedTuple):
    """Param pair for target networks (reference base_types.py:152-155)."""

    online: Any
    target: Any


@dataclass
class EvalOutput:
    episode_return: Tensor
    episode_length: Tensor
    extra_metrics: Dict[str, Tensor] = field(default_factory=dict)


def tree_map(fn: Callable, obj: Any) -> Any:
    """Map fn over every tensor leaf of a nested container."""
    if isinstance(obj, torch.Tensor):
        return fn(obj)
    if isinstance(obj, dict):
        return {k: tree_map(fn, v) for k, v in obj.items()}
    if isinstance(obj, tuple) and hasattr(obj, "_fields"):  # NamedTuple
        return type(obj)(*(tree_map(fn, v) for v in obj))
    if isinstance(obj, (list, tuple)):
        return type(obj)(tree_map(fn, v) for v in obj)
    if dataclasses.is_dataclass(obj) and not isinstance(obj, type):
        return type(obj)(**{f.name: tree_map(fn, getattr(obj, f.name)) for f in dataclasses.fields(obj)})
    return obj


def tree_flatten(obj: Any, prefix: str = "") -> Dict[str, torch.Tensor]:
    """Flatten a nested container of tensors into a flat dict."""
    out: Dict[str, torch.Tensor] = {}

    def rec(o: Any, p: str) -> None:
        if isinstance(o, torch.Tensor):
            out[p] = o
        elif isinstance(o, dict):
            for k, v in o.items():
                rec(v, f"{p}.{k}" if p else str(k))
        elif isinstance(o, tuple) and hasattr(o, "_fields"):
            for k, v in zip(o._fields, o):
                rec(v, f"{p}.{k}" if p else str(k))
        elif isinstance(o, (list, tuple)):
            for i, v in enumerate(o):
                rec(v, f"{p}.{i}" if p else str(i))
        elif dataclasses.is_dataclass(o) and not isinstance(o, type):
            for f in dataclasses.fields(o):
                rec(getattr(o, f.name), f"{p}.{f.name}" if p else f.name)

    rec(obj, prefix)
    return out


def stack_timesteps(steps: list) -> TimeStep:
    """Stack a list of per-step TimeSteps into time-major [T, B, ...]."""
    extras_keys = steps[0].extras.keys()
    return TimeStep(
        step_type=torch.stack([s.step_type for s in steps]),
        reward=torch.stack([s.reward for s in steps]),
        discount=torch.stack([s.discount for s in steps]),
        observation=_stack_any([s.observation for s in steps]),
        extras={k: _stack_any([s.extras[k] for s in steps]) for k in extras_keys},
    )


def _stack_any(items: list) -> Any:
    if isinstance(items[0], torch.Tensor):
        return torch.stack(items)
    if isinstance(items[0], dict):
        return {k: _stack_any([it[k] for it in items]) for k in items[0]}
    return items
