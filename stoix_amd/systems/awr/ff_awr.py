"""Anakin AWR — Advantage-Weighted Regression (parity:
/root/reference/stoix/systems/awr/ff_awr.py and _continuous.py; the head
comes from the network config).

Trajectory buffer of sequences (:431); per rollout, ``num_critic_steps``
TD(lambda) value regressions against targets from the pre-update critic,
then ``num_actor_steps`` weighted regressions with
``w = min(exp(A/beta), weight_clip)`` computed from the updated critic
(:152-300). beta=0.05, clip 20 (ff_awr.yaml).
"""
from __future__ import annotations

import copy
import sys
from typing import Dict

import torch
import torch.nn as nn

from stoix_amd.buffers import TrajectoryBuffer
from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class AWRLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)

        obs_space, act_space = env.observation_space, env.action_space
        self.actor = build_actor(config.network.actor_network, obs_space, act_space).to(device)
        self.critic = build_critic(config.network.critic_network, obs_space).to(device)
        broadcast_module(self.actor)
        broadcast_module(self.critic)
        self.actor_opt = torch.optim.Adam(self.actor.parameters(), lr=float(self.sys.actor_lr))
        self.critic_opt = torch.optim.Adam(self.critic.parameters(), lr=float(self.sys.critic_lr))
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.critic.parameters()), device
        )
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 301)

        self.seq_len = int(getattr(self.sys, "sample_sequence_length", 16))
        self.buffer = TrajectoryBuffer(
            add_batch_size=self.B,
            max_length_time_axis=int(self.sys.buffer_size) // self.B,
            sample_sequence_length=self.seq_len,
            device=device,
            seed=int(config.arch.seed) + 31,
        )
        self.batch_size = int(self.sys.batch_size)
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    @torch.no_grad()
    def _rollout_into_buffer(self) -> None:
        ts = self.ts
        obs_l, act_l, rew_l, disc_l, next_l = [], [], [], [], []
        for _ in range(self.T):
            obs = ts.observation
            action = self.actor(obs).sample(self.gen)
            next_ts = self.env.step(action)
            obs_l.append(obs.clone())
            act_l.append(action)
            rew_l.append(next_ts.reward)
            disc_l.append(next_ts.discount)
            next_l.append(next_ts.extras["next_obs"].clone())
            ts = next_ts
        self.ts = ts
        # rows = envs, time axis = rollout steps
        self.buffer.add(
            {
                "obs": torch.stack(obs_l, dim=1),
                "action": torch.stack(act_l, dim=1),
                "reward": torch.stack(rew_l, dim=1),
                "discount": torch.stack(disc_l, dim=1),
                "next_obs": torch.stack(next_l, dim=1),
            }
        )
        em = ts.extras["episode_metrics"]
        final, has = get_final_step_metrics(em)
        if has:
            self.episode_metrics = {k: v.mean() for k, v in final.items()}

    def _lambda_targets(self, batch: Dict[str, Tensor], critic: nn.Module) -> Tensor:
        """TD(lambda) targets over [batch, L] windows, time-major inside."""
        L = self.seq_len
        obs = batch["next_obs"]  # [batch, L, D]
        with torch.no_grad():
            v_next = critic(obs.reshape(-1, obs.shape[-1])).reshape(-1, L)
        r = batch["reward"].transpose(0, 1)  # [L, batch]
        d = batch["discount"].transpose(0, 1) * self.gamma
        vt = v_next.transpose(0, 1)
        return multistep.batch_lambda_returns(r, d, vt, float(self.sys.td_lambda)).transpose(0, 1)

    def update_step(self) -> Dict[str, Tensor]:
        self._rollout_into_buffer()
        if not self.buffer.can_sample:
            return {}
        metrics: Dict[str, Tensor] = {}
        frozen_critic = copy.deepcopy(self.critic)

        for _ in range(int(self.sys.num_critic_steps)):
            batch = self.buffer.sample(self.batch_size)
            targets = self._lambda_targets(batch, frozen_critic)
            obs = batch["obs"].reshape(-1, batch["obs"].shape[-1])
            v = self.critic(obs).reshape(targets.shape)
            v_loss = 0.5 * ((v - targets) ** 2).mean()
            self.critic_opt.zero_grad(set_to_none=True)
            v_loss.backward()
            self.reducer.reduce()
            self.reducer.wait()
            self.critic_opt.step()
            metrics["value_loss"] = v_loss.detach()

        beta = float(self.sys.beta)
        clip = float(self.sys.weight_clip)
        for _ in range(int(self.sys.num_actor_steps)):
            batch = self.buffer.sample(self.batch_size)
            targets = self._lambda_targets(batch, self.critic)
            obs = batch["obs"].reshape(-1, batch["obs"].shape[-1])
            with torch.no_grad():
                v = self.critic(obs).reshape(targets.shape)
                w = torch.exp((targets - v) / beta).clamp(max=clip)
            dist = self.actor(obs)
            act = batch["action"].reshape(-1, *batch["action"].shape[2:])
            logp = dist.log_prob(act).reshape(targets.shape)
            a_loss = -(w * logp).mean()
            self.actor_opt.zero_grad(set_to_none=True)
            a_loss.backward()
            self.reducer.reduce()
            self.reducer.wait()
            self.actor_opt.step()
            metrics["actor_loss"] = a_loss.detach()
        return metrics

    def state_for_checkpoint(self):
        return {"actor": dict(self.actor.state_dict()), "critic": dict(self.critic.state_dict())}

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])


def learner_factory(config, env, device) -> AWRLearner:
    return AWRLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_awr.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
