"""Anakin Sampled AlphaZero — continuous-action search (parity:
/root/reference/stoix/systems/search/ff_sampled_az.py).

Like ff_az but over continuous actions: each search node samples K candidate
actions from the tanh-normal actor and the batched sampled-MCTS searches over
the K arms (stoix_amd/search/mcts.py sampled_mcts_search; mctx sampled-policy
equivalent). The learner trains the policy towards the search weights over
the ROOT's sampled actions (weighted log-likelihood, the Sampled MuZero
policy target) and the critic towards GAE over the search-value trace.
"""
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
from stoix_amd.search.mcts import sampled_mcts_search
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class SampledAZLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)
        self.K = int(getattr(self.sys, "num_sampled_actions", 8))

        obs_space, act_space = env.observation_space, env.action_space
        self.act_dim = act_space.shape[0]
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
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 919)
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    # ------------------------------------------------------- search plumbing

    def _sample_candidates(self, obs: Tensor) -> Tensor:
        """K actions per env from the current policy: [B, K, act_dim]."""
        dist = self.actor(obs)
        return torch.stack([dist.sample(self.gen) for _ in range(self.K)], dim=1)

    def _recurrent_fn(self, embedding: Dict[str, Tensor], action: Tensor):
        state, reward, terminated = self.env._step_fn(dict(embedding), action)
        obs = self.env._obs_fn(state)
        discount = self.gamma * (~terminated).to(torch.float32)
        with torch.no_grad():
            cand = self._sample_candidates(obs)
            value = self.critic(obs)
        return state, reward, discount, cand, value

    @torch.no_grad()
    def _search(self, obs: Tensor, state: Dict[str, Tensor], greedy: bool = False):
        cand = self._sample_candidates(obs)
        value = self.critic(obs)
        return sampled_mcts_search(
            obs,
            state,
            cand,
            value,
            self._recurrent_fn,
            num_simulations=int(self.sys.num_simulations),
            c_puct=float(getattr(self.sys, "c_puct", 1.25)),
            temperature=0.0 if greedy else float(getattr(self.sys, "search_temperature", 1.0)),
            generator=self.gen,
        )

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        eval_env = getattr(self, "_eval_env_ref", None)
        if eval_env is not None and eval_env.observation_space.shape == obs.shape[1:]:
            state = {k: v.clone() for k, v in eval_env._state.items()}
            return self._search(obs, state, greedy=True).action
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        T, B = self.T, self.B
        obs_l, cand_l, w_l, sv_l, rew_l, disc_l, trunc_l = [], [], [], [], [], [], []
        ts = self.ts
        with torch.no_grad():
            for _ in range(T):
                obs = ts.observation
                root_state = {k: v.clone() for k, v in self.env._state.items()}
                out = self._search(obs, root_state)
                next_ts = self.env.step(out.action)
                obs_l.append(obs.clone())
                cand_l.append(out.sampled_actions)
                w_l.append(out.action_weights)
                sv_l.append(out.search_value)
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
        cand_t = torch.stack(cand_l)  # [T, B, K, act_dim]
        w_t = torch.stack(w_l)  # [T, B, K]
        sv_t = torch.stack(sv_l)
        r_t = torch.stack(rew_l)
        d_t = torch.stack(disc_l) * self.gamma
        trunc_t = torch.stack(trunc_l)
        sv_next = torch.cat([sv_t[1:], bootstrap.unsqueeze(0)], dim=0)
        adv, targets = multistep.batch_truncated_generalized_advantage_estimation(
            r_t, d_t, float(self.sys.gae_lambda), sv_t, sv_next, truncation_t=trunc_t
        )

        TB = T * B
        flat_obs = obs_t.reshape(TB, *obs_t.shape[2:])
        flat_cand = cand_t.reshape(TB, self.K, self.act_dim)
        flat_w = w_t.reshape(TB, self.K)
        flat_tgt = targets.reshape(TB)
        n_mb = int(self.sys.num_minibatches)
        mb = TB // n_mb
        metrics: Dict[str, Tensor] = {}
        for _ in range(int(self.sys.epochs)):
            perm = torch.randperm(TB, device=self.device, generator=self.gen)
            for i in range(n_mb):
                idx = perm[i * mb : (i + 1) * mb]
                dist = self.actor(flat_obs[idx])
                # weighted log-likelihood of the root's sampled actions
                # towards the search visit weights (sampled ExIt target)
                logp = torch.stack(
                    [dist.log_prob(flat_cand[idx][:, k]) for k in range(self.K)], dim=1
                )  # [mb, K]
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
                }
        return metrics

    def state_for_checkpoint(self):
        return {"actor": dict(self.actor.state_dict()), "critic": dict(self.critic.state_dict())}

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])


def learner_factory(config, env, device) -> SampledAZLearner:
    return SampledAZLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose(
        "default/anakin/default_ff_sampled_az.yaml", argv if argv is not None else sys.argv[1:]
    )
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
