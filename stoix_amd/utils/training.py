"""Learning-rate schedules (parity: /root/reference/stoix/utils/
training.py:6-49 — the simple linear decay by update count suggested by
the PPO-implementation-details blog; ``system.decay_learning_rates``
selects it, otherwise the rate is constant).

MI355X note: under hip-graph capture the optimiser step is recorded once,
so a python-float lr would be frozen into the graph. ``LinearLRDecay``
therefore drives torch optimisers through their param-group ``lr`` BETWEEN
replays (the eager/capturable-Adam path re-reads it), and learners that
use the fused Adam kernel keep decay out of eligibility (the flagship
configs all ship ``decay_learning_rates: false``, like the reference's).
"""
from __future__ import annotations

from typing import Iterable

import torch


class LinearLRDecay:
    """lr(update) = init_lr * (1 - update / num_updates), applied to every
    param group of the given optimisers once per update step."""

    def __init__(self, optimizers: Iterable[torch.optim.Optimizer], num_updates: int):
        self.opts = list(optimizers)
        self.num_updates = max(1, int(num_updates))
        self._init = [[g["lr"] for g in opt.param_groups] for opt in self.opts]
        self._count = 0

    def step(self) -> float:
        """Advance one update; returns the current decay fraction."""
        self._count += 1
        frac = max(0.0, 1.0 - self._count / self.num_updates)
        for opt, inits in zip(self.opts, self._init):
            for g, lr0 in zip(opt.param_groups, inits):
                new_lr = lr0 * frac
                if torch.is_tensor(g["lr"]):
                    g["lr"].fill_(new_lr)
                else:
                    g["lr"] = new_lr
        return frac


def maybe_lr_decay(config, *optimizers) -> LinearLRDecay | None:
    """Build the decay driver when ``system.decay_learning_rates`` is set
    (reference make_learning_rate, training.py:31-49)."""
    if bool(getattr(config.system, "decay_learning_rates", False)):
        return LinearLRDecay(optimizers, int(config.arch.num_updates))
    return None
