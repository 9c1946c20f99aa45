"""Anakin DDPG (parity: /root/reference/stoix/systems/ddpg/ff_ddpg.py)."""
from __future__ import annotations

import sys
from typing import Dict, Tuple

import torch

from stoix_amd.config import compose
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.ddpg.base import DDPGFamilyLearner

Tensor = torch.Tensor


class DDPGLearner(DDPGFamilyLearner):
    n_critics = 1

    def critic_loss(self, batch: Dict[str, Tensor]) -> Tuple[Tensor, Dict[str, Tensor]]:
        with torch.no_grad():
            a_next = self._actor_action(self.actor_target, batch["next_obs"])
            q_next = self.q_target(batch["next_obs"], a_next)
            target = batch["reward"] + self.gamma * batch["discount"] * q_next
        q_pred = self.q_online(batch["obs"], batch["action"])
        loss = 0.5 * ((q_pred - target) ** 2).mean()
        return loss, {"q_loss": loss.detach(), "q_mean": q_pred.mean().detach()}


def learner_factory(config, env, device) -> DDPGLearner:
    return DDPGLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_ddpg.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
