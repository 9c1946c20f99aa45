"""Anakin DPO — drift policy optimisation (parity: /root/reference/stoix/
systems/ppo/anakin/ff_dpo_continuous.py): drift-formulated surrogate
(loss.py:50-65) with alpha/beta hyperparameters."""
from __future__ import annotations

import sys

import torch

from stoix_amd.config import compose
from stoix_amd.ops.losses import dpo_loss
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.ppo.ff_ppo import PPOLearner

Tensor = torch.Tensor


class DPOLearner(PPOLearner):
    def policy_loss(self, new_logp: Tensor, old_logp: Tensor, adv: Tensor) -> Tensor:
        return dpo_loss(new_logp, old_logp, adv, float(self.sys.dpo_alpha), float(self.sys.dpo_beta))


def learner_factory(config, env, device) -> DPOLearner:
    return DPOLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_dpo_continuous.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
