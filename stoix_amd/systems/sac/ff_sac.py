"""Anakin SAC (parity: /root/reference/stoix/systems/sac/ff_sac.py).

Twin Q critics (MultiNetwork of obs-action critics, ff_sac.py:374-380),
soft target ``r + gamma*d*(min Q' - alpha*logpi(a'))`` with a fresh sample
(:173-199), actor loss ``alpha*logpi - min Q`` via rsample (:201-221),
autotuned temperature with ``target_entropy = -scale*action_dim`` and dual
loss ``-log_alpha * sg(logpi + H*)`` (:151-171, 406-411).
"""
from __future__ import annotations

import copy
import sys
from typing import Dict

import torch
import torch.nn as nn

from stoix_amd.buffers import ItemBuffer
from stoix_amd.config import compose
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.base import MultiNetwork
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module
from stoix_amd.systems.anakin import run_anakin_experiment

Tensor = torch.Tensor


class SACLearner:
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
        q_cfg = config.network.critic_network

        def make_q():
            return build_critic(q_cfg, obs_space, act_space, obs_action_input=True)

        self.q_online = MultiNetwork([make_q(), make_q()]).to(device)
        broadcast_module(self.actor)
        broadcast_module(self.q_online)
        self.q_target = copy.deepcopy(self.q_online)
        for p in self.q_target.parameters():
            p.requires_grad_(False)

        act_dim = act_space.shape[0]
        self.target_entropy = -float(getattr(self.sys, "target_entropy_scale", 1.0)) * act_dim
        init_alpha = float(getattr(self.sys, "init_alpha", 1.0))
        self.log_alpha = torch.nn.Parameter(
            torch.tensor(float(torch.log(torch.tensor(init_alpha))), device=device)
        )
        self.autotune = bool(getattr(self.sys, "autotune", True))

        self.actor_opt = torch.optim.Adam(self.actor.parameters(), lr=float(self.sys.actor_lr), capturable=device.type == "cuda")
        self.q_opt = torch.optim.Adam(self.q_online.parameters(), lr=float(self.sys.q_lr), capturable=device.type == "cuda")
        self.alpha_opt = torch.optim.Adam([self.log_alpha], lr=float(self.sys.alpha_lr), capturable=device.type == "cuda")
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.q_online.parameters()) + [self.log_alpha],
            device,
        )
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 97)

        cap = int(self.sys.buffer_size) // max(1, int(config.arch.n_devices))
        self.buffer = ItemBuffer(cap, device=device, seed=int(config.arch.seed) + 23)
        self.batch_size = int(self.sys.batch_size)

        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}
        self.collect_metrics = True
        self._warmup()

    # --------------------------------------------------------------- acting

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        dist = self.actor(obs)
        return dist.mode() if greedy else dist.sample(self.gen)

    # -------------------------------------------------------------- rollout

    @torch.no_grad()
    def _rollout_into_buffer(self, steps: int, random_actions: bool = False) -> None:
        ts = self.ts
        for _ in range(steps):
            obs = ts.observation
            if random_actions:
                action = self.env.action_space.sample(self.B, self.device, self.gen)
            else:
                action = self.actor(obs).sample(self.gen)
            next_ts = self.env.step(action)
            self.buffer.add(
                {
                    "obs": obs,
                    "action": action,
                    "reward": next_ts.reward,
                    "discount": next_ts.discount,
                    "next_obs": next_ts.extras["next_obs"],
                }
            )
            ts = next_ts
        self.ts = ts
        if self.collect_metrics:
            em = ts.extras["episode_metrics"]
            final, has = get_final_step_metrics(em)
            if has:
                self.episode_metrics = {k: v.mean() for k, v in final.items()}

    def _warmup(self) -> None:
        steps = max(1, int(getattr(self.sys, "warmup_steps", 64)) // self.B + 1)
        self._rollout_into_buffer(steps, random_actions=True)

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        self._rollout_into_buffer(self.T)
        metrics: Dict[str, Tensor] = {}
        tau = float(self.sys.tau)
        for _ in range(int(self.sys.epochs)):
            batch = self.buffer.sample(self.batch_size)
            alpha = self.log_alpha.exp().detach()

            # ---- critic update
            with torch.no_grad():
                next_dist = self.actor(batch["next_obs"])
                a_next, logp_next = next_dist.sample_and_log_prob(self.gen)
                q_next = self.q_target(batch["next_obs"], a_next).min(dim=0).values
                target = batch["reward"] + self.gamma * batch["discount"] * (
                    q_next - alpha * logp_next
                )
            q_pred = self.q_online(batch["obs"], batch["action"])  # [2, B]
            q_loss = 0.5 * ((q_pred - target.unsqueeze(0)) ** 2).mean()

            # ---- actor update (pathwise, twin-min)
            dist = self.actor(batch["obs"])
            a_new, logp_new = dist.sample_and_log_prob(self.gen)
            q_new = self.q_online(batch["obs"], a_new).min(dim=0).values
            actor_loss = (alpha * logp_new - q_new).mean()

            # ---- temperature dual
            if self.autotune:
                alpha_loss = -(self.log_alpha * (logp_new + self.target_entropy).detach()).mean()
            else:
                alpha_loss = torch.zeros((), device=self.device)

            self.actor_opt.zero_grad(set_to_none=True)
            self.q_opt.zero_grad(set_to_none=True)
            self.alpha_opt.zero_grad(set_to_none=True)
            (q_loss + actor_loss + alpha_loss).backward()
            self.reducer.reduce()
            self.reducer.wait()
            if getattr(self.sys, "max_grad_norm", None):
                nn.utils.clip_grad_norm_(self.actor.parameters(), float(self.sys.max_grad_norm))
                nn.utils.clip_grad_norm_(self.q_online.parameters(), float(self.sys.max_grad_norm))
            self.actor_opt.step()
            self.q_opt.step()
            if self.autotune:
                self.alpha_opt.step()

            with torch.no_grad():
                from stoix_amd.parallel.dist import polyak_update

                polyak_update(self.q_online.parameters(), self.q_target.parameters(), tau)
            metrics = {
                "q_loss": q_loss.detach(),
                "actor_loss": actor_loss.detach(),
                "alpha": self.log_alpha.exp().detach(),
                "entropy": -logp_new.mean().detach(),
            }
        return metrics

    # ------------------------------------------------------------ checkpoint


    # ------------------------------------------------------- graph support

    @property
    def graph_capturable(self) -> bool:
        return (
            getattr(self.env, "_hip", None) is not None
            or getattr(self.env, "capture_safe", False)
        )

    def prepare_for_graph_capture(self) -> None:
        """Capture-safe modes: default (graph-aware) CUDA RNG, inline
        all-reduce, no host-side metric reads, graph-safe buffer RNG."""
        self.gen = None
        self.collect_metrics = False
        self.reducer._stream = None
        self.buffer.graph_safe_rng = True

    def after_graph_replay(self) -> None:
        from stoix_amd.envs.env import latched_episode_metrics

        self.episode_metrics = latched_episode_metrics(self.env, self)

    # ------------------------------------------------------------ checkpoint

    def state_for_checkpoint(self):
        return {
            "actor": dict(self.actor.state_dict()),
            "q_online": dict(self.q_online.state_dict()),
            "q_target": dict(self.q_target.state_dict()),
            "log_alpha": self.log_alpha.detach(),
        }

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])


def learner_factory(config, env, device) -> SACLearner:
    return SACLearner(config, env, device)


def run(config) -> float:
    return run_anakin_experiment(config, learner_factory)


def hydra_entry_point(argv=None) -> float:
    cfg = compose("default/anakin/default_ff_sac.yaml", argv if argv is not None else sys.argv[1:])
    return run(cfg)


if __name__ == "__main__":
    hydra_entry_point()
