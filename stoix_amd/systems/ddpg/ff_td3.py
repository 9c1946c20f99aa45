"""Anakin TD3 (parity: /root/reference/stoix/systems/ddpg/ff_td3.py):
twin critics, target-policy smoothing noise clip(N(0, policy_noise),
+-noise_clip) (:185-195), twin-min target, delayed actor updates (:396)."""
from __future__ import annotations

import sys
from typing import Dict, Tuple

import torch

from stoix_amd.config import compose
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.ddpg.base import DDPGFamilyLearner

Tensor = torch.Tensor


class TD3Learner(DDPGFamilyLearner):
    n_critics = 2

    def critic_loss(self, batch: Dict[str, Tensor]) -> Tuple[Tensor, Dict[str, Tensor]]:
        policy_noise = float(getattr(self.sys, "policy_noise", 0.2))
        noise_clip = float(getattr(self.sys, "noise_clip", 0.5))
        with torch.no_grad():
            a_next = self._actor_action(self.actor_target, batch["next_obs"])
            noise = (
                torch.randn(a_next.shape, device=a_next.device, generator=self.gen) * policy_noise
            ).clamp(-noise_clip, noise_clip)
            a_next = (a_next + noise).clamp(self.act_min, self.act_max)
            q_next = self.q_target(batch["next_obs"], a_next).min(dim=0).values
            target = batch["reward"] + self.gamma * batch["discount"] * q_next
        q_pred = self.q_online(batch["obs"], batch["action"])  # [2, B]
        loss = 0.5 * ((q_pred - target.unsqueeze(0)) ** 2).mean()
        return loss, {"q_loss": loss.detach(), "q_mean": q_pred.mean().detach()}

    def actor_loss(self, batch: Dict[str, Tensor]) -> Tensor:
        a = self._actor_action(self.actor, batch["obs"])
        q = self.q_online(batch["obs"], a)[0]  # first critic, standard TD3
        return -q.mean()


def learner_factory(config, env, device) -> TD3Learner:
    return TD3Learner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_td3.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
