"""Hyperparameter sweeps (the reference exposes optuna sweeps through
hydra's sweeper plugin — /root/reference/stoix/configs/default/anakin/
hyperparameter_sweep.yaml uses optuna's TPE sampler; optuna is not
installable offline, so this module implements grid, random, and a compact
TPE (tree-structured Parzen estimator) sampler — the same strategy optuna
defaults to — driving any system's ``run(config)`` and maximising its
returned final episode return).

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
    # test hook: maximise this callable(params)->float instead of running
    # the RL entry point
    objective: Optional[Any] = None

    def _run_one(self, params: Dict[str, Any]) -> Trial:
        t = Trial(params=params)
        if self.objective is not None:
            try:
                t.value = float(self.objective(params))
            except Exception as e:
                t.error = repr(e)
            self.trials.append(t)
            return t
        from stoix_amd.config import compose

        mod = importlib.import_module(self.entry)
        overrides = list(self.base_overrides) + [f"{k}={v}" for k, v in params.items()]
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

    def run_tpe(
        self,
        num_trials: int,
        seed: int = 0,
        n_startup: int = 5,
        gamma: float = 0.25,
        n_candidates: int = 24,
    ) -> Optional[Trial]:
        """Compact TPE (optuna's default sampler): after ``n_startup``
        random trials, split observations into good (top ``gamma``
        quantile by value) and bad, model each with per-dimension Parzen
        estimators, and pick the candidate maximising the good/bad
        density ratio l(x)/g(x)."""
        rng = random.Random(seed)

        def numeric(dist):
            return isinstance(dist, (Uniform, LogUniform))

        def to_unit(dist, v):
            if isinstance(dist, LogUniform):
                return (math.log(v) - math.log(dist.lo)) / (math.log(dist.hi) - math.log(dist.lo) + 1e-12)
            return (v - dist.lo) / (dist.hi - dist.lo + 1e-12)

        def kde_logpdf(xs, x, bw):
            if not xs:
                return 0.0
            acc = 0.0
            for m in xs:
                acc += math.exp(-0.5 * ((x - m) / bw) ** 2)
            return math.log(acc / len(xs) / bw + 1e-12)

        for i in range(num_trials):
            done = [t for t in self.trials if t.value is not None and t.value == t.value]
            if i < n_startup or len(done) < n_startup:
                params = {k: d.sample(rng) for k, d in self.space.items()}
                self._run_one(params)
                continue
            done.sort(key=lambda t: -t.value)
            n_good = max(1, int(len(done) * gamma))
            good, bad = done[:n_good], done[n_good:]
            bw = max(0.1, 1.0 / max(len(good), 1))
            best_params, best_score = None, -1e18
            for _ in range(n_candidates):
                cand = {}
                score = 0.0
                for k, d in self.space.items():
                    if numeric(d):
                        # sample around a random good observation
                        m = to_unit(d, rng.choice(good).params[k])
                        u = min(1.0, max(0.0, rng.gauss(m, bw)))
                        gx = [to_unit(d, t.params[k]) for t in good]
                        bx = [to_unit(d, t.params[k]) for t in bad]
                        score += kde_logpdf(gx, u, bw) - kde_logpdf(bx, u, max(bw, 0.2))
                        if isinstance(d, LogUniform):
                            cand[k] = math.exp(math.log(d.lo) + u * (math.log(d.hi) - math.log(d.lo)))
                        else:
                            cand[k] = d.lo + u * (d.hi - d.lo)
                    else:
                        # categorical: weighted by good-frequency + prior
                        opts = list(d.options)
                        wg = [1.0 + sum(1 for t in good if t.params[k] == o) for o in opts]
                        wb = [1.0 + sum(1 for t in bad if t.params[k] == o) for o in opts]
                        tot = sum(wg)
                        r = rng.uniform(0, tot)
                        acc = 0.0
                        pick = opts[-1]
                        for o, w in zip(opts, wg):
                            acc += w
                            if r <= acc:
                                pick = o
                                break
                        j = opts.index(pick)
                        score += math.log(wg[j] / sum(wg)) - math.log(wb[j] / sum(wb))
                        cand[k] = pick
                if score > best_score:
                    best_score, best_params = score, cand
            self._run_one(best_params)
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
    p.add_argument("--tpe", action="store_true")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--param", action="append", default=[])
    p.add_argument("--override", action="append", default=[])
    args = p.parse_args(argv)
    space = dict(_parse_param(s) for s in args.param)
    sweep = Sweep(entry=args.entry, default=args.default, space=space, base_overrides=args.override)
    if args.grid:
        best = sweep.run_grid()
    elif args.tpe:
        best = sweep.run_tpe(args.trials, args.seed)
    else:
        best = sweep.run_random(args.trials, args.seed)
    print(sweep.summary())
    if best is not None:
        print("BEST:", json.dumps({"value": best.value, **best.params}))
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
