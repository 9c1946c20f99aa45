"""Anakin D4PG (parity: /root/reference/stoix/systems/ddpg/ff_d4pg.py):
distributional critic (DistributionalContinuousQNetworkHead returning
(value, logits, atoms), heads.py:259-274), categorical TD target with the
target-actor action (:183-222), actor loss -E[Q] (:224-243).

Simplification vs the reference: the reference samples n-step windows from a
trajectory buffer (:477) and builds n-step rewards; here 1-step targets from
the item buffer (the categorical-projection machinery is identical; n-step
windows arrive with the trajectory-buffer integration of rainbow)."""
from __future__ import annotations

import sys
from typing import Dict, Tuple

import torch
import torch.nn.functional as F

from stoix_amd.config import compose
from stoix_amd.ops.losses import categorical_td_learning
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.ddpg.base import DDPGFamilyLearner

Tensor = torch.Tensor


class D4PGLearner(DDPGFamilyLearner):
    n_critics = 1

    def critic_loss(self, batch: Dict[str, Tensor]) -> Tuple[Tensor, Dict[str, Tensor]]:
        with torch.no_grad():
            a_next = self._actor_action(self.actor_target, batch["next_obs"])
            out_t = self.q_target(batch["next_obs"], a_next)
        out_tm1 = self.q_online(batch["obs"], batch["action"])
        loss = categorical_td_learning(
            out_tm1.logits,
            out_tm1.atoms,
            batch["reward"],
            self.gamma * batch["discount"],
            out_t.logits,
            out_t.atoms,
        )
        return loss, {"q_loss": loss.detach(), "q_mean": out_tm1.value.mean().detach()}

    def actor_loss(self, batch: Dict[str, Tensor]) -> Tensor:
        a = self._actor_action(self.actor, batch["obs"])
        out = self.q_online(batch["obs"], a)
        return -out.value.mean()

    def _scalar_q(self, q_out) -> Tensor:
        return q_out.value if hasattr(q_out, "value") else q_out


def learner_factory(config, env, device) -> D4PGLearner:
    return D4PGLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_d4pg.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
