"""Anakin REINFORCE with baseline (parity: /root/reference/stoix/systems/vpg/
ff_reinforce.py and ff_reinforce_continuous.py — the head comes from the
network config): one update per rollout using n-step/discounted returns as
targets, advantage = G - V(s), policy-gradient + entropy + critic L2."""
from __future__ import annotations

import sys
from typing import Dict

import torch
import torch.nn as nn

from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class ReinforceLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs

        obs_space, act_space = env.observation_space, env.action_space
        self.actor = build_actor(config.network.actor_network, obs_space, act_space).to(device)
        self.critic = build_critic(config.network.critic_network, obs_space).to(device)
        broadcast_module(self.actor)
        broadcast_module(self.critic)
        self.actor_opt = torch.optim.Adam(self.actor.parameters(), lr=float(self.sys.actor_lr), eps=1e-5)
        self.critic_opt = torch.optim.Adam(self.critic.parameters(), lr=float(self.sys.critic_lr), eps=1e-5)
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.critic.parameters()), device
        )
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 7)
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    def update_step(self) -> Dict[str, Tensor]:
        T, B = self.T, self.B
        obs_buf = []
        act_buf = []
        rew_buf = []
        disc_buf = []
        boot_buf = []
        ts = self.ts
        with torch.no_grad():
            for _ in range(T):
                obs = ts.observation
                dist = self.actor(obs)
                action = dist.sample(self.gen)
                next_ts = self.env.step(action)
                obs_buf.append(obs)
                act_buf.append(action)
                rew_buf.append(next_ts.reward)
                disc_buf.append(next_ts.discount)
                boot_buf.append(self.critic(next_ts.extras["next_obs"]))
                ts = next_ts
            self.ts = ts
            em = ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

        obs_t = torch.stack([o.clone() for o in obs_buf])
        act_t = torch.stack(act_buf)
        r_t = torch.stack(rew_buf)
        d_t = torch.stack(disc_buf) * float(self.sys.gamma)
        boot_t = torch.stack(boot_buf)
        targets = multistep.batch_discounted_returns(r_t, d_t, boot_t)

        flat_obs = obs_t.reshape(T * B, *obs_t.shape[2:])
        value = self.critic(flat_obs).reshape(T, B)
        dist = self.actor(flat_obs)
        log_prob = dist.log_prob(act_t.reshape(T * B, *act_t.shape[2:])).reshape(T, B)
        entropy = dist.entropy().mean()
        adv = (targets - value).detach()
        if bool(getattr(self.sys, "standardize_advantages", False)):
            adv = (adv - adv.mean()) / (adv.std(unbiased=False) + 1e-8)
        actor_loss = -(log_prob * adv).mean() - float(self.sys.ent_coef) * entropy
        critic_loss = float(self.sys.vf_coef) * 0.5 * ((value - targets.detach()) ** 2).mean()

        self.actor_opt.zero_grad(set_to_none=True)
        self.critic_opt.zero_grad(set_to_none=True)
        (actor_loss + critic_loss).backward()
        self.reducer.reduce()
        self.reducer.wait()
        nn.utils.clip_grad_norm_(self.actor.parameters(), float(self.sys.max_grad_norm))
        nn.utils.clip_grad_norm_(self.critic.parameters(), float(self.sys.max_grad_norm))
        self.actor_opt.step()
        self.critic_opt.step()
        return {
            "actor_loss": actor_loss.detach(),
            "value_loss": critic_loss.detach(),
            "entropy": entropy.detach(),
        }

    def state_for_checkpoint(self):
        return {"actor": dict(self.actor.state_dict()), "critic": dict(self.critic.state_dict())}

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])


def learner_factory(config, env, device) -> ReinforceLearner:
    return ReinforceLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_reinforce.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
