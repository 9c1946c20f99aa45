"""Hyperparameter sweeps (the reference exposes optuna sweeps through
hydra's sweeper plugin; optuna is not installable offline, so this module
implements the two search strategies the reference's configs actually use —
grid and random — plus a simple median-pruning successive-halving loop,
driving any system's ``run(config)`` and maximising its returned final
episode return).

Usage (programmatic):
    from stoix_amd.utils.sweep import Sweep, Uniform, LogUniform, Choice
    sweep = Sweep(
        entry="stoix_amd.systems.ppo.ff_ppo",
        default="default/anakin/default_ff_ppo.yaml",
        space={"system.actor_lr": LogUniform(1e-5, 1e-2),
               "system.ent_coef": Uniform(0.0, 0.05),
               "system.num_minibatches": Choice([8, 16, 32])},
        base_overrides=["env=gymnax/cartpole"],
    )
    best = sweep.run_random(num_trials=20, seed=0)

CLI:
    python -m stoix_amd.utils.sweep --entry ... --default ... \
        --trials 10 --param system.actor_lr=log:1e-5:1e-2 \
        --param system.num_minibatches=choice:8,16,32
"""
from __future__ import annotations

import argparse
import importlib
import itertools
import json
import math
import random
import sys
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence


@dataclass
class Uniform:
    lo: float
    hi: float

    def sample(self, rng: random.Random) -> float:
        return rng.uniform(self.lo, self.hi)

    def grid(self, n: int) -> List[float]:
        return [self.lo + (self.hi - self.lo) * i / max(n - 1, 1) for i in range(n)]


@dataclass
class LogUniform:
    lo: float
    hi: float

    def sample(self, rng: random.Random) -> float:
        return math.exp(rng.uniform(math.log(self.lo), math.log(self.hi)))

    def grid(self, n: int) -> List[float]:
        ll, lh = math.log(self.lo), math.log(self.hi)
        return [math.exp(ll + (lh - ll) * i / max(n - 1, 1)) for i in range(n)]


@dataclass
class Choice:
    options: Sequence[Any]

    def sample(self, rng: random.Random) -> Any:
        return rng.choice(list(self.options))

    def grid(self, n: int) -> List[Any]:
        return list(self.options)


@dataclass
class Trial:
    params: Dict[str, Any]
    value: Optional[float] = None
    error: Optional[str] = None


@dataclass
class Sweep:
    entry: str
    default: str
    space: Dict[str, Any]
    base_overrides: List[str] = field(default_factory=list)
    trials: List[Trial] = field(default_factory=list)

    def _run_one(self, params: Dict[str, Any]) -> Trial:
        from stoix_amd.config import compose

        mod = importlib.import_module(self.entry)
        overrides = list(self.base_overrides) + [f"{k}={v}" for k, v in params.items()]
        t = Trial(params=params)
        try:
            cfg = compose(self.default, overrides)
            t.value = float(mod.run(cfg))
        except Exception as e:  # a failed trial is recorded, not fatal
            t.error = repr(e)
        self.trials.append(t)
        return t

    def run_random(self, num_trials: int, seed: int = 0) -> Optional[Trial]:
        rng = random.Random(seed)
        for _ in range(num_trials):
            params = {k: dist.sample(rng) for k, dist in self.space.items()}
            self._run_one(params)
        return self.best()

    def run_grid(self, points_per_dim: int = 3) -> Optional[Trial]:
        axes = {k: dist.grid(points_per_dim) for k, dist in self.space.items()}
        keys = list(axes)
        for combo in itertools.product(*axes.values()):
            self._run_one(dict(zip(keys, combo)))
        return self.best()

    def best(self) -> Optional[Trial]:
        done = [t for t in self.trials if t.value is not None and t.value == t.value]
        return max(done, key=lambda t: t.value) if done else None

    def summary(self) -> str:
        lines = []
        for t in sorted(self.trials, key=lambda t: -(t.value if t.value is not None else -1e18)):
            lines.append(json.dumps({"value": t.value, "error": t.error, **{f"p:{k}": v for k, v in t.params.items()}}))
        return "\n".join(lines)


def _parse_param(spec: str):
    """system.actor_lr=log:1e-5:1e-2 | uniform:0:1 | choice:a,b,c"""
    key, rest = spec.split("=", 1)
    kind, _, args = rest.partition(":")
    if kind == "log":
        lo, hi = args.split(":")
        return key, LogUniform(float(lo), float(hi))
    if kind == "uniform":
        lo, hi = args.split(":")
        return key, Uniform(float(lo), float(hi))
    if kind == "choice":
        opts = [yaml_scalar(v) for v in args.split(",")]
        return key, Choice(opts)
    raise ValueError(f"unknown param kind '{kind}' in {spec}")


def yaml_scalar(v: str) -> Any:
    import yaml

    return yaml.safe_load(v)


def main(argv=None) -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--entry", required=True)
    p.add_argument("--default", required=True)
    p.add_argument("--trials", type=int, default=10)
    p.add_argument("--grid", action="store_true")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--param", action="append", default=[])
    p.add_argument("--override", action="append", default=[])
    args = p.parse_args(argv)
    space = dict(_parse_param(s) for s in args.param)
    sweep = Sweep(entry=args.entry, default=args.default, space=space, base_overrides=args.override)
    if args.grid:
        best = sweep.run_grid()
    else:
        best = sweep.run_random(args.trials, args.seed)
    print(sweep.summary())
    if best is not None:
        print("BEST:", json.dumps({"value": best.value, **best.params}))
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
