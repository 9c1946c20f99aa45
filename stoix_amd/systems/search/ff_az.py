"""Anakin AlphaZero (parity: /root/reference/stoix/systems/search/ff_az.py).

Acting = batched MCTS over the REAL environment model: the recurrent_fn
steps ``env._step_fn`` on embedded env-state tensors (ff_az.py:79-103), the
root comes from the actor/critic networks (:51-71). Stores ExIt transitions
with the search visit distribution and search value; the learner trains the
policy towards the search policy (CE) and the critic towards GAE targets
computed over the SEARCH value trace (:238-283).
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
from stoix_amd.search.mcts import mcts_search
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class AZLearner:
    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)

        obs_space, act_space = env.observation_space, env.action_space
        self.num_actions = act_space.num_values
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
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 911)
        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}

    # ------------------------------------------------------- search plumbing

    def _recurrent_fn(self, embedding: Dict[str, Tensor], action: Tensor):
        """Real-env-model expansion (reference ff_az.py:79-103)."""
        state, reward, terminated = self.env._step_fn(dict(embedding), action)
        obs = self.env._obs_fn(state)
        discount = self.gamma * (~terminated).to(torch.float32)
        with torch.no_grad():
            prior_logits = self.actor(obs).logits
            value = self.critic(obs)
        return state, reward, discount, prior_logits, value

    @torch.no_grad()
    def _search(self, obs: Tensor, state: Dict[str, Tensor], greedy: bool = False):
        prior_logits = self.actor(obs).logits
        value = self.critic(obs)
        return mcts_search(
            obs,
            state,
            prior_logits,
            value,
            self._recurrent_fn,
            num_simulations=int(self.sys.num_simulations),
            c_puct=float(getattr(self.sys, "c_puct", 1.25)),
            dirichlet_alpha=None if greedy else float(getattr(self.sys, "dirichlet_alpha", 0.3)),
            temperature=0.0 if greedy else float(getattr(self.sys, "search_temperature", 1.0)),
            generator=self.gen,
        )

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        """Search evaluator: full search per eval step (reference
        systems/search/evaluator.py)."""
        eval_env = getattr(self, "_eval_env_ref", None)
        if eval_env is not None and eval_env.observation_space.shape == obs.shape[1:]:
            state = {k: v.clone() for k, v in eval_env._state.items()}
            return self._search(obs, state, greedy=True).action
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        T, B = self.T, self.B
        obs_l, w_l, sv_l, rew_l, disc_l, trunc_l = [], [], [], [], [], []
        ts = self.ts
        with torch.no_grad():
            for _ in range(T):
                obs = ts.observation
                root_state = {k: v.clone() for k, v in self.env._state.items()}
                out = self._search(obs, root_state)
                next_ts = self.env.step(out.action)
                obs_l.append(obs.clone())
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
        w_t = torch.stack(w_l)  # [T, B, A] search policies
        sv_t = torch.stack(sv_l)
        r_t = torch.stack(rew_l)
        d_t = torch.stack(disc_l) * self.gamma
        trunc_t = torch.stack(trunc_l)
        # GAE over the SEARCH value trace
        sv_next = torch.cat([sv_t[1:], bootstrap.unsqueeze(0)], dim=0)
        adv, targets = multistep.batch_truncated_generalized_advantage_estimation(
            r_t, d_t, float(self.sys.gae_lambda), sv_t, sv_next, truncation_t=trunc_t
        )

        TB = T * B
        flat_obs = obs_t.reshape(TB, *obs_t.shape[2:])
        flat_w = w_t.reshape(TB, self.num_actions)
        flat_tgt = targets.reshape(TB)
        n_mb = int(self.sys.num_minibatches)
        mb = TB // n_mb
        metrics: Dict[str, Tensor] = {}
        for _ in range(int(self.sys.epochs)):
            perm = torch.randperm(TB, device=self.device, generator=self.gen)
            for i in range(n_mb):
                idx = perm[i * mb : (i + 1) * mb]
                dist = self.actor(flat_obs[idx])
                log_pi = torch.log_softmax(dist.logits, dim=-1)
                ce = -(flat_w[idx] * log_pi).sum(-1).mean()
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


def learner_factory(config, env, device) -> AZLearner:
    return AZLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_az.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
