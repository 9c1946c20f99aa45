"""Anakin V-MPO (discrete and continuous by action-space detection).

Parity: /root/reference/stoix/systems/mpo/ff_vmpo.py / ff_vmpo_continuous.py —
on-policy (no buffer), advantages from GAE on the fresh rollout (:184-208),
E-step over the TOP-HALF advantages with a temperature dual, M-step weighted
CE with KL Lagrange penalties (rlax.vmpo_loss semantics, :130-150), target
actor updated every ``target_update_period`` updates (:269), V-critic on
importance-corrected TD errors (multistep.py:452).
"""
from __future__ import annotations

import math
import copy
import sys
from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.envs.spaces import DiscreteSpace
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment
from stoix_amd.systems.mpo.ff_mpo import softplus_dual

Tensor = torch.Tensor


class VMPOLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)
        obs_space, act_space = env.observation_space, env.action_space
        self.discrete = isinstance(act_space, DiscreteSpace)

        self.actor = build_actor(config.network.actor_network, obs_space, act_space).to(device)
        self.critic = build_critic(config.network.critic_network, obs_space).to(device)
        broadcast_module(self.actor)
        broadcast_module(self.critic)
        self.actor_target = copy.deepcopy(self.actor)
        for p in self.actor_target.parameters():
            p.requires_grad_(False)

        init = float(getattr(self.sys, "init_log_temperature", 10.0))
        init_a = float(getattr(self.sys, "init_log_alpha", 10.0))
        # the reference's continuous configs pin the std dual much higher
        # (init_log_alpha_stddev 500 vs alpha_mean 10) to freeze sigma early
        init_as = float(getattr(self.sys, "init_log_alpha_stddev", init_a))
        self.log_temperature = nn.Parameter(torch.tensor(init, device=device))
        if self.discrete:
            self.log_alpha = nn.Parameter(torch.tensor(init_a, device=device))
            duals = [self.log_temperature, self.log_alpha]
        else:
            act_dim = act_space.shape[0]
            self.log_alpha_mean = nn.Parameter(torch.full((act_dim,), init_a, device=device))
            self.log_alpha_std = nn.Parameter(torch.full((act_dim,), init_as, device=device))
            duals = [self.log_temperature, self.log_alpha_mean, self.log_alpha_std]
        self.duals = duals

        self.actor_opt = torch.optim.Adam(self.actor.parameters(), lr=float(self.sys.actor_lr))
        self.critic_opt = torch.optim.Adam(self.critic.parameters(), lr=float(self.sys.critic_lr))
        self.dual_opt = torch.optim.Adam(duals, lr=float(self.sys.dual_lr))
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.critic.parameters()) + duals, device
        )
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 461)
        self.update_count = 0
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    def update_step(self) -> Dict[str, Tensor]:
        T, B = self.T, self.B
        obs_l, act_l, logp_l, rew_l, disc_l, trunc_l, next_l = [], [], [], [], [], [], []
        ts = self.ts
        with torch.no_grad():
            for _ in range(T):
                obs = ts.observation
                dist = self.actor(obs)
                action = dist.sample(self.gen)
                logp_l.append(dist.log_prob(action))
                next_ts = self.env.step(action)
                obs_l.append(obs.clone())
                act_l.append(action)
                rew_l.append(next_ts.reward)
                disc_l.append(next_ts.discount)
                trunc_l.append(next_ts.truncated())
                next_l.append(next_ts.extras["next_obs"].clone())
                ts = next_ts
            self.ts = ts
            em = ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

        obs_t = torch.stack(obs_l)
        act_t = torch.stack(act_l)
        logp_b = torch.stack(logp_l)
        r_t = torch.stack(rew_l)
        d_t = torch.stack(disc_l) * self.gamma
        trunc = torch.stack(trunc_l)

        flat_obs = obs_t.reshape(T * B, -1)
        value = self.critic(flat_obs).reshape(T, B)
        with torch.no_grad():
            boot = self.critic(torch.stack(next_l).reshape(T * B, -1)).reshape(T, B)
        adv, targets = multistep.batch_truncated_generalized_advantage_estimation(
            r_t, d_t, float(getattr(self.sys, "gae_lambda", 0.95)), value.detach(), boot, truncation_t=trunc
        )
        critic_loss = 0.5 * ((value - targets.detach()) ** 2).mean()

        # --- E-step: top-half advantages
        eps_temp = float(getattr(self.sys, "epsilon", 0.1))
        temperature = softplus_dual(self.log_temperature)
        flat_adv = adv.reshape(-1).detach()
        k = flat_adv.numel() // 2
        top_adv, top_idx = torch.topk(flat_adv, k)
        w_logits = top_adv / temperature.detach()
        weights = torch.softmax(w_logits, dim=0).detach()
        temp_loss = temperature * (
            eps_temp + torch.logsumexp(top_adv / temperature, dim=0) - math.log(k)
        )

        dist_o = self.actor(flat_obs)
        flat_act = act_t.reshape(T * B, *act_t.shape[2:])
        logp_o = dist_o.log_prob(flat_act)
        ce_loss = -(weights * logp_o[top_idx]).sum()

        # --- KL constraint vs target policy
        with torch.no_grad():
            dist_t = self.actor_target(flat_obs)
        if self.discrete:
            eps_pol = float(getattr(self.sys, "epsilon_policy", 0.1))
            kl = (dist_t.probs * (dist_t.logits - dist_o.logits)).sum(-1).mean()
            alpha = softplus_dual(self.log_alpha)
            kl_loss = alpha.detach() * kl + alpha * (eps_pol - kl.detach())
        else:
            eps_mean = float(getattr(self.sys, "epsilon_mean", 0.01))
            eps_std = float(getattr(self.sys, "epsilon_stddev", 1e-5))
            mu_o, sig_o = dist_o._n.loc, dist_o._n.scale
            mu_t, sig_t = dist_t._n.loc, dist_t._n.scale
            kl_mean = (((mu_o - mu_t) ** 2) / (2 * sig_t**2)).mean(0)
            kl_std = ((sig_o / sig_t).log() * -1.0 + sig_o**2 / (2 * sig_t**2) - 0.5).mean(0)
            a_m = softplus_dual(self.log_alpha_mean)
            a_s = softplus_dual(self.log_alpha_std)
            kl_loss = (
                (a_m.detach() * kl_mean).sum()
                + (a_s.detach() * kl_std).sum()
                + (a_m * (eps_mean - kl_mean.detach())).sum()
                + (a_s * (eps_std - kl_std.detach())).sum()
            )
            kl = kl_mean.mean()

        loss = ce_loss + kl_loss + temp_loss + critic_loss
        self.actor_opt.zero_grad(set_to_none=True)
        self.critic_opt.zero_grad(set_to_none=True)
        self.dual_opt.zero_grad(set_to_none=True)
        loss.backward()
        self.reducer.reduce()
        self.reducer.wait()
        if getattr(self.sys, "max_grad_norm", None):
            nn.utils.clip_grad_norm_(self.actor.parameters(), float(self.sys.max_grad_norm))
            nn.utils.clip_grad_norm_(self.critic.parameters(), float(self.sys.max_grad_norm))
        self.actor_opt.step()
        self.critic_opt.step()
        self.dual_opt.step()

        self.update_count += 1
        if self.update_count % int(getattr(self.sys, "target_update_period", 100)) == 0:
            self.actor_target.load_state_dict(self.actor.state_dict())

        return {
            "policy_loss": ce_loss.detach(),
            "value_loss": critic_loss.detach(),
            "temperature": temperature.detach(),
            "kl": kl.detach(),
        }

    def state_for_checkpoint(self):
        return {"actor": dict(self.actor.state_dict()), "critic": dict(self.critic.state_dict())}

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])


def learner_factory(config, env, device) -> VMPOLearner:
    return VMPOLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_vmpo.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
