"""Anakin PQN (parity: /root/reference/stoix/systems/q_learning/ff_pqn.py):
buffer-free on-policy Q(lambda) — rollout with epsilon-greedy online net,
Peng's Q(lambda) targets computed once over the trajectory (:92-175), then
shuffled minibatch epochs of TD regression; LayerNorm MLP by convention."""
from __future__ import annotations

import sys
from typing import Dict

import torch
import torch.nn as nn

from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.distributions import EpsilonGreedy
from stoix_amd.networks.factory import build_q_network
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class PQNLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)

        self.q_net = build_q_network(
            config.network.actor_network,
            env.observation_space,
            env.action_space,
            epsilon=float(self.sys.training_epsilon),
        ).to(device)
        broadcast_module(self.q_net)
        self.opt = torch.optim.Adam(self.q_net.parameters(), lr=float(self.sys.q_lr), eps=1e-5)
        self.reducer = FlatGradReducer(self.q_net.parameters(), device)
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 211)
        self.train_eps = float(self.sys.training_epsilon)
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    def _q(self, obs: Tensor) -> Tensor:
        out = self.q_net(obs)
        return out.preferences if isinstance(out, EpsilonGreedy) else out

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        q = self._q(obs)
        if greedy:
            return q.argmax(dim=-1)
        return EpsilonGreedy(q, float(getattr(self.sys, "evaluation_epsilon", 0.0))).sample(self.gen)

    def update_step(self) -> Dict[str, Tensor]:
        T, B = self.T, self.B
        obs_list, act_list, rew_list, disc_list, q_next_list = [], [], [], [], []
        ts = self.ts
        with torch.no_grad():
            for _ in range(T):
                obs = ts.observation
                q = self._q(obs)
                action = EpsilonGreedy(q, self.train_eps).sample(self.gen)
                next_ts = self.env.step(action)
                obs_list.append(obs.clone())
                act_list.append(action)
                rew_list.append(next_ts.reward)
                disc_list.append(next_ts.discount)
                q_next_list.append(self._q(next_ts.extras["next_obs"]))
                ts = next_ts
            self.ts = ts
            em = ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

            obs_t = torch.stack(obs_list)
            act_t = torch.stack(act_list)
            r_t = torch.stack(rew_list)
            d_t = torch.stack(disc_list) * self.gamma
            q_next = torch.stack(q_next_list)  # [T, B, A]
            targets = multistep.batch_q_lambda(
                q_next, act_t, r_t, d_t, q_next, float(self.sys.q_lambda)
            )

        TB = T * B
        flat_obs = obs_t.reshape(TB, *obs_t.shape[2:])
        flat_act = act_t.reshape(TB)
        flat_tgt = targets.reshape(TB)
        n_mb = int(self.sys.num_minibatches)
        mb = TB // n_mb
        metrics: Dict[str, Tensor] = {}
        for _ in range(int(self.sys.epochs)):
            perm = torch.randperm(TB, device=self.device, generator=self.gen)
            for i in range(n_mb):
                idx = perm[i * mb : (i + 1) * mb]
                q = self._q(flat_obs[idx])
                q_a = q.gather(-1, flat_act[idx].unsqueeze(-1)).squeeze(-1)
                loss = 0.5 * ((q_a - flat_tgt[idx]) ** 2).mean()
                self.opt.zero_grad(set_to_none=True)
                loss.backward()
                self.reducer.reduce()
                self.reducer.wait()
                if getattr(self.sys, "max_grad_norm", None):
                    nn.utils.clip_grad_norm_(self.q_net.parameters(), float(self.sys.max_grad_norm))
                self.opt.step()
                metrics = {"q_loss": loss.detach(), "q_mean": q.mean().detach()}
        return metrics

    def state_for_checkpoint(self):
        return {"q_net": dict(self.q_net.state_dict())}

    def snapshot_params(self):
        return {"q_net": {k: v.clone() for k, v in self.q_net.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.q_net.load_state_dict(snap["q_net"])


def learner_factory(config, env, device) -> PQNLearner:
    return PQNLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_pqn.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
