"""Anakin PPO-penalty (parity: /root/reference/stoix/systems/ppo/anakin/
ff_ppo_penalty.py and _continuous.py): KL-penalty surrogate instead of
clipping (loss.py:35-47), kl_penalty_coef=3.0."""
from __future__ import annotations

import sys

import torch

from stoix_amd.config import compose
from stoix_amd.ops.losses import ppo_penalty_loss
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.ppo.ff_ppo import PPOLearner

Tensor = torch.Tensor


class PPOPenaltyLearner(PPOLearner):
    def policy_loss(self, new_logp: Tensor, old_logp: Tensor, adv: Tensor) -> Tensor:
        loss, _kl = ppo_penalty_loss(new_logp, old_logp, adv, float(self.sys.kl_penalty_coef))
        return loss


def learner_factory(config, env, device) -> PPOPenaltyLearner:
    return PPOPenaltyLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_ppo_penalty.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
