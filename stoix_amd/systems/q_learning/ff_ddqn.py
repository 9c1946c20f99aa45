"""Anakin Double DQN (parity: /root/reference/stoix/systems/q_learning/ff_ddqn.py)."""
from __future__ import annotations

import sys
from typing import Dict, Tuple

import torch

from stoix_amd.config import compose
from stoix_amd.ops.losses import double_q_learning
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.q_learning.base import OffPolicyQLearner

Tensor = torch.Tensor


class DDQNLearner(OffPolicyQLearner):
    def loss_fn(self, batch: Dict[str, Tensor]) -> Tuple[Tensor, Dict[str, Tensor]]:
        q_tm1 = self.q_values(self.q_online, batch["obs"])
        with torch.no_grad():
            q_t_value = self.q_values(self.q_target, batch["next_obs"])
            q_t_selector = self.q_values(self.q_online, batch["next_obs"])
        loss = double_q_learning(
            q_tm1,
            q_t_value,
            batch["action"],
            batch["reward"],
            self.gamma * batch["discount"],
            q_t_selector,
            float(getattr(self.sys, "huber_loss_parameter", 0.0)),
        )
        return loss, {"q_loss": loss.detach(), "q_mean": q_tm1.mean().detach()}


def learner_factory(config, env, device) -> DDQNLearner:
    return DDQNLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_ddqn.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
