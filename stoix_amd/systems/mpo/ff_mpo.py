"""Anakin MPO — Maximum a Posteriori Policy Optimisation (discrete and
continuous by action-space detection).

Parity surface: /root/reference/stoix/systems/mpo/ff_mpo.py (discrete),
ff_mpo_continuous.py and the shared losses (continuous_loss.py:26-303,
discrete_loss.py:20-120):
  * sequences with stored behaviour log-probs in a trajectory buffer
    (ff_mpo.py:539-545)
  * critic: retrace(lambda) targets from target nets (ff_mpo.py:263-285)
  * E-step: temperature-tempered weights over sampled (continuous) or all
    (discrete) actions + temperature dual loss
  * M-step: weighted CE with decoupled fixed-mean/fixed-std distributions
    and per-dim KL duals alpha_mean/alpha_std (continuous), single alpha
    (discrete); duals in softplus space clipped at log -18
    (continuous_loss.py:18-20, 143-151)
  * target nets updated periodically (polyak here), joint actor+dual grads.
"""
from __future__ import annotations

import copy
import math
import sys
from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.buffers import TrajectoryBuffer
from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.envs.spaces import DiscreteSpace
from stoix_amd.networks.factory import build_actor, build_critic, build_q_network
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor

_MIN_LOG_DUAL = -18.0


def softplus_dual(param: Tensor) -> Tensor:
    return F.softplus(param.clamp(min=_MIN_LOG_DUAL)) + 1e-8


class MPOLearner:
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
        broadcast_module(self.actor)
        self.actor_target = copy.deepcopy(self.actor)
        if self.discrete:
            self.num_actions = act_space.num_values
            self.q = build_q_network(
                config.network.critic_network, obs_space, act_space, epsilon=0.0
            ).to(device)
        else:
            self.act_dim = act_space.shape[0]
            self.q = build_critic(
                config.network.critic_network, obs_space, act_space, obs_action_input=True
            ).to(device)
        broadcast_module(self.q)
        self.q_target = copy.deepcopy(self.q)
        for p in list(self.actor_target.parameters()) + list(self.q_target.parameters()):
            p.requires_grad_(False)

        # dual variables (softplus space)
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
            self.log_alpha_mean = nn.Parameter(torch.full((self.act_dim,), init_a, device=device))
            self.log_alpha_std = nn.Parameter(torch.full((self.act_dim,), init_as, device=device))
            duals = [self.log_temperature, self.log_alpha_mean, self.log_alpha_std]

        self.actor_opt = torch.optim.Adam(self.actor.parameters(), lr=float(self.sys.actor_lr))
        self.q_opt = torch.optim.Adam(self.q.parameters(), lr=float(self.sys.q_lr))
        self.dual_opt = torch.optim.Adam(duals, lr=float(self.sys.dual_lr))
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.q.parameters()) + duals, device
        )
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 401)

        self.seq_len = int(getattr(self.sys, "sample_sequence_length", 8))
        self.buffer = TrajectoryBuffer(
            add_batch_size=self.B,
            max_length_time_axis=max(self.seq_len + 1, int(self.sys.buffer_size) // self.B),
            sample_sequence_length=self.seq_len,
            device=device,
            seed=int(config.arch.seed) + 37,
        )
        self.batch_size = int(self.sys.batch_size)
        self.num_samples = int(getattr(self.sys, "num_samples", 20))
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    # --------------------------------------------------------------- acting

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    # -------------------------------------------------------------- rollout

    @torch.no_grad()
    def _rollout_into_buffer(self) -> None:
        ts = self.ts
        obs_l, act_l, rew_l, disc_l, logp_l, next_l = [], [], [], [], [], []
        for _ in range(self.T):
            obs = ts.observation
            dist = self.actor(obs)
            action = dist.sample(self.gen)
            logp = dist.log_prob(action)
            next_ts = self.env.step(action)
            obs_l.append(obs.clone())
            act_l.append(action)
            logp_l.append(logp)
            rew_l.append(next_ts.reward)
            disc_l.append(next_ts.discount)
            next_l.append(next_ts.extras["next_obs"].clone())
            ts = next_ts
        self.ts = ts
        self.buffer.add(
            {
                "obs": torch.stack(obs_l, 1),
                "action": torch.stack(act_l, 1),
                "log_prob": torch.stack(logp_l, 1),
                "reward": torch.stack(rew_l, 1),
                "discount": torch.stack(disc_l, 1),
                "next_obs": torch.stack(next_l, 1),
            }
        )
        em = ts.extras["episode_metrics"]
        final, has = get_final_step_metrics(em)
        if has:
            self.episode_metrics = {k: v.mean() for k, v in final.items()}

    # ---------------------------------------------------------------- critic

    def _q_values(self, net, obs: Tensor, action: Tensor) -> Tensor:
        if self.discrete:
            out = net(obs)
            q = out.preferences if hasattr(out, "preferences") else out
            return q.gather(-1, action.long().unsqueeze(-1)).squeeze(-1)
        return net(obs, action)

    def _critic_update(self, batch: Dict[str, Tensor]) -> Tensor:
        L = self.seq_len
        obs = batch["obs"]  # [batch, L, D]
        nxt = batch["next_obs"]
        act = batch["action"]
        with torch.no_grad():
            # fresh actions/values from target nets for retrace
            flat_n = nxt.reshape(-1, nxt.shape[-1])
            dist_t = self.actor_target(flat_n)
            a_t = dist_t.sample(self.gen)
            q_next = self._q_values(self.q_target, flat_n, a_t).reshape(-1, L)
            if self.discrete:
                out = self.q_target(flat_n)
                qv = out.preferences if hasattr(out, "preferences") else out
                probs = dist_t.probs
                v_next = (probs * qv).sum(-1).reshape(-1, L)
            else:
                v_next = q_next  # E_a' Q approximated with one sample
            # behaviour-correction ratios: pi_target(a|s') vs stored logp
            flat_o = obs.reshape(-1, obs.shape[-1])
            dist_now = self.actor_target(flat_o)
            logp_now = dist_now.log_prob(act.reshape(-1, *act.shape[2:])).reshape(-1, L)
            log_rho = (logp_now - batch["log_prob"]).clamp(-10, 10)
            q_sel = self._q_values(self.q_target, flat_o, act.reshape(-1, *act.shape[2:])).reshape(-1, L)
            # retrace expects q_t[t]=Q(s_{t+1}, a_{t+1}); use shifted windows
            targets = multistep.batch_retrace_continuous(
                q_sel.transpose(0, 1),
                q_next.transpose(0, 1),
                v_next.transpose(0, 1),
                batch["reward"].transpose(0, 1),
                batch["discount"].transpose(0, 1) * self.gamma,
                log_rho.transpose(0, 1),
                float(getattr(self.sys, "retrace_lambda", 0.95)),
            ).transpose(0, 1)
        q_pred = self._q_values(self.q, obs.reshape(-1, obs.shape[-1]), act.reshape(-1, *act.shape[2:])).reshape(-1, L)
        loss = 0.5 * ((q_pred - targets) ** 2).mean()
        self.q_opt.zero_grad(set_to_none=True)
        loss.backward()
        return loss.detach()

    # ----------------------------------------------------------- actor/dual

    def _policy_update_continuous(self, obs: Tensor) -> Dict[str, Tensor]:
        eps = float(getattr(self.sys, "epsilon", 0.01))
        eps_mean = float(getattr(self.sys, "epsilon_mean", 1e-3))
        eps_std = float(getattr(self.sys, "epsilon_stddev", 1e-6))
        M = self.num_samples
        Bz = obs.shape[0]
        with torch.no_grad():
            dist_t = self.actor_target(obs)
            mu_t, sigma_t = dist_t._n.loc, dist_t._n.scale
            a_samp = mu_t.unsqueeze(0) + sigma_t.unsqueeze(0) * torch.randn(
                (M, Bz, self.act_dim), device=obs.device, generator=self.gen
            )
            obs_rep = obs.unsqueeze(0).expand(M, -1, -1).reshape(M * Bz, -1)
            q_samp = self._q_values(self.q_target, obs_rep, a_samp.reshape(M * Bz, -1)).reshape(M, Bz)

        temperature = softplus_dual(self.log_temperature)
        # E-step weights + temperature dual
        q_detached = q_samp
        logits = q_detached / temperature.detach()
        weights = torch.softmax(logits, dim=0).detach()
        temp_loss = temperature * (
            eps + (torch.logsumexp(q_detached / temperature, dim=0) - math.log(M)).mean()
        )

        # M-step: decoupled mean/std weighted CE
        dist_o = self.actor(obs)
        mu_o, sigma_o = dist_o._n.loc, dist_o._n.scale

        def normal_logp(mu, sigma, a):
            var = sigma**2
            return (-((a - mu) ** 2) / (2 * var) - sigma.log() - 0.5 * math.log(2 * math.pi)).sum(-1)

        lp_mean = normal_logp(mu_o.unsqueeze(0), sigma_t.unsqueeze(0), a_samp)  # grad wrt mu
        lp_std = normal_logp(mu_t.unsqueeze(0), sigma_o.unsqueeze(0), a_samp)  # grad wrt sigma
        ce_loss = -(weights * (lp_mean + lp_std)).sum(0).mean()

        # per-dim KL constraints with duals
        kl_mean = (((mu_o - mu_t) ** 2) / (2 * sigma_t**2)).mean(0)  # [act_dim]
        kl_std = (
            (sigma_o / sigma_t).log() * -1.0 + (sigma_o**2) / (2 * sigma_t**2) - 0.5
        ).mean(0)
        alpha_mean = softplus_dual(self.log_alpha_mean)
        alpha_std = softplus_dual(self.log_alpha_std)
        loss_kl_mean = (alpha_mean.detach() * kl_mean).sum()
        loss_kl_std = (alpha_std.detach() * kl_std).sum()
        loss_alpha = (alpha_mean * (eps_mean - kl_mean.detach())).sum() + (
            alpha_std * (eps_std - kl_std.detach())
        ).sum()

        loss = ce_loss + loss_kl_mean + loss_kl_std + loss_alpha + temp_loss
        self.actor_opt.zero_grad(set_to_none=True)
        self.dual_opt.zero_grad(set_to_none=True)
        loss.backward()
        return {
            "policy_loss": ce_loss.detach(),
            "temperature": temperature.detach(),
            "kl_mean": kl_mean.mean().detach(),
        }

    def _policy_update_discrete(self, obs: Tensor) -> Dict[str, Tensor]:
        eps = float(getattr(self.sys, "epsilon", 0.01))
        eps_pol = float(getattr(self.sys, "epsilon_policy", 1e-3))
        with torch.no_grad():
            out = self.q_target(obs)
            q_all = out.preferences if hasattr(out, "preferences") else out  # [B, A]
            dist_t = self.actor_target(obs)
            logits_t = dist_t.logits
        temperature = softplus_dual(self.log_temperature)
        weights = torch.softmax(q_all / temperature.detach() + logits_t, dim=-1).detach()
        temp_loss = temperature * (
            eps
            + (torch.logsumexp(q_all / temperature + logits_t, dim=-1)).mean()
        )
        dist_o = self.actor(obs)
        ce_loss = -(weights * dist_o.logits).sum(-1).mean()
        kl = (logits_t.exp() * (logits_t - dist_o.logits)).sum(-1).mean()
        alpha = softplus_dual(self.log_alpha)
        loss = ce_loss + alpha.detach() * kl + alpha * (eps_pol - kl.detach()) + temp_loss
        self.actor_opt.zero_grad(set_to_none=True)
        self.dual_opt.zero_grad(set_to_none=True)
        loss.backward()
        return {"policy_loss": ce_loss.detach(), "temperature": temperature.detach(), "kl": kl.detach()}

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        self._rollout_into_buffer()
        if not self.buffer.can_sample:
            return {}
        metrics: Dict[str, Tensor] = {}
        tau = float(self.sys.tau)
        for _ in range(int(self.sys.epochs)):
            batch = self.buffer.sample(self.batch_size)
            q_loss = self._critic_update(batch)
            obs = batch["obs"].reshape(-1, batch["obs"].shape[-1])
            if self.discrete:
                pm = self._policy_update_discrete(obs)
            else:
                pm = self._policy_update_continuous(obs)
            self.reducer.reduce()
            self.reducer.wait()
            if getattr(self.sys, "max_grad_norm", None):
                nn.utils.clip_grad_norm_(self.actor.parameters(), float(self.sys.max_grad_norm))
                nn.utils.clip_grad_norm_(self.q.parameters(), float(self.sys.max_grad_norm))
            self.q_opt.step()
            self.actor_opt.step()
            self.dual_opt.step()
            with torch.no_grad():
                from stoix_amd.parallel.dist import polyak_update

                polyak_update(self.q.parameters(), self.q_target.parameters(), tau)
                polyak_update(self.actor.parameters(), self.actor_target.parameters(), tau)
            metrics = {"q_loss": q_loss, **pm}
        return metrics

    def state_for_checkpoint(self):
        return {"actor": dict(self.actor.state_dict()), "q": dict(self.q.state_dict())}

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])


def learner_factory(config, env, device) -> MPOLearner:
    return MPOLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_mpo.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
