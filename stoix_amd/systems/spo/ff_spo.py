"""Anakin SPO — Sequential-Monte-Carlo Policy Optimisation (parity:
/root/reference/stoix/systems/spo/ff_spo.py and ff_spo_continuous.py; one
file here, the head type selects discrete vs continuous).

Acting = SMC search over the REAL env model: a particle set
[B, P] (state clones + the ROOT action each particle first took +
TD-advantage log-weights) is rolled ``search_depth`` steps — sample action
from pi, step env._step_fn, add the temperature-scaled TD error
(r + gamma*V(s') - V(s))/eta to the log-weight, systematic-resample whenever
the effective sample size falls below ``ess_threshold`` (reference
ff_spo.py:342-983). Readout: a particle drawn per env proportional to the
final weights supplies the executed root action; the weight distribution
over root particles is the search policy target.

Adaptive temperature: eta minimises the MPO-style dual
  g(eta) = eta*epsilon + eta*log E[exp(A/eta)]
by SGD on log_eta per update (reference's adaptive temperature dual,
ff_spo.py:431-466).

Training: actor CE towards the particle weights (weighted log-likelihood of
root actions — sampled-MPO E-step form), critic on GAE over the
particle-weighted search values; epochs x minibatches like PPO.
"""
from __future__ import annotations

import math
import sys
from typing import Dict

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.ops import multistep
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


def systematic_resample(logw: Tensor, generator=None) -> Tensor:
    """Systematic resampling indices per env: [B, P] -> [B, P] int64."""
    B, P = logw.shape
    w = torch.softmax(logw, dim=-1)
    cdf = torch.cumsum(w, dim=-1)
    u0 = torch.rand((B, 1), device=logw.device, generator=generator) / P
    pts = u0 + torch.arange(P, device=logw.device).unsqueeze(0) / P
    return torch.searchsorted(cdf, pts.contiguous()).clamp(max=P - 1)


class SPOLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.P = int(self.sys.num_particles)
        self.depth = int(self.sys.search_depth)
        self.gamma = float(self.sys.gamma)
        self.ess_threshold = float(getattr(self.sys, "ess_threshold", 0.5))

        obs_space, act_space = env.observation_space, env.action_space
        self._discrete = not hasattr(act_space, "shape") or len(getattr(act_space, "shape", ())) == 0
        self.num_actions = act_space.num_values if self._discrete else act_space.shape[0]
        self.actor = build_actor(config.network.actor_network, obs_space, act_space).to(device)
        self.critic = build_critic(config.network.critic_network, obs_space).to(device)
        broadcast_module(self.actor)
        broadcast_module(self.critic)
        self.actor_opt = torch.optim.Adam(self.actor.parameters(), lr=float(self.sys.actor_lr))
        self.critic_opt = torch.optim.Adam(self.critic.parameters(), lr=float(self.sys.critic_lr))
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.critic.parameters()), device
        )
        # adaptive SMC temperature dual (log-parameterised, positive)
        self.log_eta = torch.tensor(
            math.log(float(getattr(self.sys, "init_temperature", 1.0))), device=device
        ).requires_grad_(True)
        self.eta_opt = torch.optim.Adam([self.log_eta], lr=float(getattr(self.sys, "dual_lr", 1e-2)))
        self.kl_epsilon = float(getattr(self.sys, "kl_epsilon", 0.1))

        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 787)
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}
        self._adv_buffer: Tensor | None = None  # advantages seen by the dual

    # --------------------------------------------------------------- search

    @torch.no_grad()
    def _smc_search(self, obs: Tensor, state: Dict[str, Tensor]):
        """Returns (root_actions [B,P,(adim)], weights [B,P],
        search_value [B])."""
        B, P = self.B if obs.shape[0] == self.B else obs.shape[0], self.P
        eta = self.log_eta.exp().detach()
        bidx = torch.arange(B, device=self.device)

        # replicate env state per particle: [B*P, ...]
        pstate = {k: v.repeat_interleave(P, dim=0) for k, v in state.items()}
        pobs = obs.repeat_interleave(P, dim=0)
        v = self.critic(pobs)  # [B*P]
        logw = torch.zeros(B, P, device=self.device)
        # un-reset advantage accumulator for the temperature dual: follows
        # particle lineage through resampling but is NOT zeroed by it (the
        # dual must see the whole search's advantages, not the post-resample
        # tail — ADVICE r1)
        adv_acc = torch.zeros(B, P, device=self.device)
        alive = torch.ones(B * P, device=self.device)
        root_action = None

        for d in range(self.depth):
            dist = self.actor(pobs)
            action = dist.sample(self.gen)
            if root_action is None:
                root_action = action.view(B, P, *action.shape[1:]).clone()
            pstate, reward, terminated = self.env._step_fn(pstate, action)
            pobs = self.env._obs_fn(pstate)
            v_next = self.critic(pobs)
            td = (reward + self.gamma * v_next * (~terminated).float() - v) * alive
            logw = logw + (td / eta).view(B, P)
            adv_acc = adv_acc + td.view(B, P)
            alive = alive * (~terminated).float()
            v = v_next
            # ESS-triggered systematic resampling (per env)
            w = torch.softmax(logw, dim=-1)
            ess = 1.0 / (w.pow(2).sum(-1) * P)  # normalised ESS in (0, 1]
            need = ess < self.ess_threshold
            if bool(need.any()) and d < self.depth - 1:
                idx = systematic_resample(logw, self.gen)  # [B, P]
                flat_idx = (bidx.unsqueeze(1) * P + idx).view(-1)
                sel = need.repeat_interleave(P)
                for k in list(pstate.keys()):
                    pstate[k] = torch.where(
                        _expand(sel, pstate[k]), pstate[k][flat_idx], pstate[k]
                    )
                pobs = torch.where(_expand(sel, pobs), pobs[flat_idx], pobs)
                v = torch.where(sel, v[flat_idx], v)
                alive = torch.where(sel, alive[flat_idx], alive)
                adv_acc = torch.where(
                    need.unsqueeze(1), adv_acc.view(-1)[flat_idx].view(B, P), adv_acc
                )
                ra_flat = root_action.view(B * P, *root_action.shape[2:])
                ra_new = torch.where(_expand(sel, ra_flat), ra_flat[flat_idx], ra_flat)
                root_action = ra_new.view_as(root_action)
                logw = torch.where(need.unsqueeze(1), torch.zeros_like(logw), logw)

        weights = torch.softmax(logw, dim=-1)  # [B, P]
        # particle-weighted search value at the root: V(s) + weighted
        # advantage signal is approximated by the weighted bootstrap values
        search_value = (weights * v.view(B, P)).sum(-1)
        # record the full search's advantages for the temperature dual
        # (lineage-tracked accumulator, unaffected by resample resets)
        self._adv_buffer = adv_acc.detach()
        return root_action, weights, search_value

    @torch.no_grad()
    def _search_root(self, obs: Tensor, state: Dict[str, Tensor], greedy: bool = False):
        root_action, weights, search_value = self._smc_search(obs, state)
        if greedy:
            arm = weights.argmax(dim=-1)
        else:
            arm = torch.multinomial(weights.clamp(min=1e-9), 1, generator=self.gen).squeeze(-1)
        bidx = torch.arange(weights.shape[0], device=self.device)
        action = root_action[bidx, arm]
        return action, root_action, weights, search_value

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        eval_env = getattr(self, "_eval_env_ref", None)
        if eval_env is not None and eval_env.observation_space.shape == obs.shape[1:]:
            state = {k: v.clone() for k, v in eval_env._state.items()}
            action, *_ = self._search_root(obs, state, greedy=True)
            return action
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        T, B, P = self.T, self.B, self.P
        obs_l, ra_l, w_l, sv_l, rew_l, disc_l, trunc_l = [], [], [], [], [], [], []
        adv_l = []
        ts = self.ts
        with torch.no_grad():
            for _ in range(T):
                obs = ts.observation
                root_state = {k: v.clone() for k, v in self.env._state.items()}
                action, root_actions, weights, search_value = self._search_root(obs, root_state)
                adv_l.append(self._adv_buffer)
                next_ts = self.env.step(action)
                obs_l.append(obs.clone())
                ra_l.append(root_actions)
                w_l.append(weights)
                sv_l.append(search_value)
                rew_l.append(next_ts.reward)
                disc_l.append(next_ts.discount)
                trunc_l.append(next_ts.truncated())
                ts = next_ts
            bootstrap = self.critic(ts.extras["next_obs"])
            self.ts = ts
            em = ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

        obs_t = torch.stack(obs_l)
        ra_t = torch.stack(ra_l)  # [T, B, P, (adim)]
        w_t = torch.stack(w_l)  # [T, B, P]
        sv_t = torch.stack(sv_l)
        r_t = torch.stack(rew_l)
        d_t = torch.stack(disc_l) * self.gamma
        trunc_t = torch.stack(trunc_l)
        sv_next = torch.cat([sv_t[1:], bootstrap.unsqueeze(0)], dim=0)
        adv, targets = multistep.batch_truncated_generalized_advantage_estimation(
            r_t, d_t, float(self.sys.gae_lambda), sv_t, sv_next, truncation_t=trunc_t
        )

        # ---- temperature dual step on the search advantages, aggregated
        # over ALL T per-step searches of the rollout (not just the last)
        if adv_l:
            eta = self.log_eta.exp()
            A = torch.cat(adv_l, dim=0)  # [T*B, P] raw advantage sums
            dual = eta * self.kl_epsilon + eta * torch.logsumexp(
                A / eta - math.log(A.shape[-1]), dim=-1
            ).mean()
            self.eta_opt.zero_grad(set_to_none=True)
            dual.backward()
            self.eta_opt.step()

        TB = T * B
        flat_obs = obs_t.reshape(TB, *obs_t.shape[2:])
        flat_ra = ra_t.reshape(TB, P, *ra_t.shape[3:])
        flat_w = w_t.reshape(TB, P)
        flat_tgt = targets.reshape(TB)
        n_mb = int(self.sys.num_minibatches)
        mb = TB // n_mb
        metrics: Dict[str, Tensor] = {}
        for _ in range(int(self.sys.epochs)):
            perm = torch.randperm(TB, device=self.device, generator=self.gen)
            for i in range(n_mb):
                idx = perm[i * mb : (i + 1) * mb]
                dist = self.actor(flat_obs[idx])
                logp = torch.stack(
                    [dist.log_prob(flat_ra[idx][:, p]) for p in range(P)], dim=1
                )  # [mb, P]
                ce = -(flat_w[idx] * logp).sum(-1).mean()
                entropy = dist.entropy().mean()
                actor_loss = ce - float(self.sys.ent_coef) * entropy
                v = self.critic(flat_obs[idx])
                v_loss = 0.5 * ((v - flat_tgt[idx]) ** 2).mean()
                self.actor_opt.zero_grad(set_to_none=True)
                self.critic_opt.zero_grad(set_to_none=True)
                (actor_loss + float(self.sys.vf_coef) * v_loss).backward()
                self.reducer.reduce()
                self.reducer.wait()
                nn.utils.clip_grad_norm_(self.actor.parameters(), float(self.sys.max_grad_norm))
                nn.utils.clip_grad_norm_(self.critic.parameters(), float(self.sys.max_grad_norm))
                self.actor_opt.step()
                self.critic_opt.step()
                metrics = {
                    "policy_ce": ce.detach(),
                    "value_loss": v_loss.detach(),
                    "entropy": entropy.detach(),
                    "temperature": self.log_eta.exp().detach(),
                }
        return metrics

    def state_for_checkpoint(self):
        return {
            "actor": dict(self.actor.state_dict()),
            "critic": dict(self.critic.state_dict()),
            "log_eta": self.log_eta.detach(),
        }

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])


def _expand(mask: Tensor, like: Tensor) -> Tensor:
    """Broadcast a [N] bool mask over trailing dims of `like` [N, ...]."""
    return mask.view(-1, *([1] * (like.dim() - 1))).expand_as(like)


def learner_factory(config, env, device) -> SPOLearner:
    return SPOLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_spo.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
