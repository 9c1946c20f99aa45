"""Anakin QR-DQN (parity: /root/reference/stoix/systems/q_learning/ff_qr_dqn.py):
QuantileDiscreteQNetwork head + quantile-regression pinball loss."""
from __future__ import annotations

import sys
from typing import Dict, Tuple

import torch

from stoix_amd.config import compose
from stoix_amd.ops.losses import quantile_q_learning
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.q_learning.base import OffPolicyQLearner

Tensor = torch.Tensor


class QRDQNLearner(OffPolicyQLearner):
    def q_values(self, net, obs: Tensor) -> Tensor:
        return net(obs).q_values

    def loss_fn(self, batch: Dict[str, Tensor]) -> Tuple[Tensor, Dict[str, Tensor]]:
        out_tm1 = self.q_online(batch["obs"])
        with torch.no_grad():
            out_t = self.q_target(batch["next_obs"])
            sel = self.q_online(batch["next_obs"])
        loss = quantile_q_learning(
            out_tm1.q_dist,
            out_tm1.taus,
            batch["action"],
            batch["reward"],
            self.gamma * batch["discount"],
            sel.q_dist,
            out_t.q_dist,
            float(getattr(self.sys, "huber_loss_parameter", 1.0)),
        )
        return loss, {"q_loss": loss.detach(), "q_mean": out_tm1.q_values.mean().detach()}


def learner_factory(config, env, device) -> QRDQNLearner:
    return QRDQNLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_qr_dqn.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
