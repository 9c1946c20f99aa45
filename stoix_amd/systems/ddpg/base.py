"""Shared DDPG-family machinery (parity: /root/reference/stoix/systems/ddpg/
ff_ddpg.py / ff_td3.py / ff_d4pg.py common shape): deterministic actor with
Gaussian exploration noise at rollout, obs-action critics with targets,
polyak, item buffer, delayed/periodic actor updates."""
from __future__ import annotations

import copy
from typing import Dict, Tuple

import torch
import torch.nn as nn

from stoix_amd.buffers import ItemBuffer
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.base import MultiNetwork
from stoix_amd.networks.factory import build_actor, build_critic
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module

Tensor = torch.Tensor


class DDPGFamilyLearner:
    n_critics = 1

    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)

        obs_space, act_space = env.observation_space, env.action_space
        self.act_min = float(act_space.minimum.min())
        self.act_max = float(act_space.maximum.max())
        self.actor = build_actor(config.network.actor_network, obs_space, act_space).to(device)
        q_cfg = config.network.critic_network

        def make_q():
            return build_critic(q_cfg, obs_space, act_space, obs_action_input=True)

        if self.n_critics > 1:
            self.q_online = MultiNetwork([make_q() for _ in range(self.n_critics)]).to(device)
        else:
            self.q_online = make_q().to(device)
        broadcast_module(self.actor)
        broadcast_module(self.q_online)
        self.actor_target = copy.deepcopy(self.actor)
        self.q_target = copy.deepcopy(self.q_online)
        for p in list(self.actor_target.parameters()) + list(self.q_target.parameters()):
            p.requires_grad_(False)

        self.actor_opt = torch.optim.Adam(self.actor.parameters(), lr=float(self.sys.actor_lr), capturable=device.type == "cuda")
        self.q_opt = torch.optim.Adam(self.q_online.parameters(), lr=float(self.sys.q_lr), capturable=device.type == "cuda")
        self.reducer = FlatGradReducer(
            list(self.actor.parameters()) + list(self.q_online.parameters()), device
        )
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 131)

        cap = int(self.sys.buffer_size) // max(1, int(config.arch.n_devices))
        self.buffer = ItemBuffer(cap, device=device, seed=int(config.arch.seed) + 29)
        self.batch_size = int(self.sys.batch_size)
        self.exploration_sigma = float(getattr(self.sys, "exploration_sigma", 0.1))
        self.update_count = 0

        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}
        self.collect_metrics = True
        self._warmup()

    # --------------------------------------------------------------- acting

    def _actor_action(self, net: nn.Module, obs: Tensor) -> Tensor:
        out = net(obs)
        return out if isinstance(out, Tensor) else out.mode()

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        """EVALUATION action: deterministic either way — the reference's
        deterministic policies "sample" to their mean at eval
        (exploration noise belongs to the ROLLOUT path only,
        _explore_action); greedy is accepted for evaluator-API parity."""
        del greedy
        return self._actor_action(self.actor, obs)

    @torch.no_grad()
    def _explore_action(self, obs: Tensor) -> Tensor:
        a = self._actor_action(self.actor, obs)
        noise = torch.randn(a.shape, device=a.device, generator=self.gen) * self.exploration_sigma
        return (a + noise).clamp(self.act_min, self.act_max)

    # -------------------------------------------------------------- rollout

    @torch.no_grad()
    def _rollout_into_buffer(self, steps: int, random_actions: bool = False) -> None:
        ts = self.ts
        for _ in range(steps):
            obs = ts.observation
            if random_actions:
                action = self.env.action_space.sample(self.B, self.device, self.gen)
            else:
                action = self._explore_action(obs)
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

    # ---------------------------------------------------------------- losses

    def critic_loss(self, batch: Dict[str, Tensor]) -> Tuple[Tensor, Dict[str, Tensor]]:
        raise NotImplementedError

    def actor_loss(self, batch: Dict[str, Tensor]) -> Tensor:
        a = self._actor_action(self.actor, batch["obs"])
        q = self.q_online(batch["obs"], a)
        if q.dim() > 1 and self.n_critics > 1:
            q = q[0]
        return -self._scalar_q(q).mean()

    def _scalar_q(self, q_out) -> Tensor:
        return q_out

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        self._rollout_into_buffer(self.T)
        metrics: Dict[str, Tensor] = {}
        tau = float(self.sys.tau)
        freq = int(getattr(self.sys, "policy_frequency", 1))
        for _ in range(int(self.sys.epochs)):
            batch = self.buffer.sample(self.batch_size)
            c_loss, metrics = self.critic_loss(batch)
            self.q_opt.zero_grad(set_to_none=True)
            c_loss.backward()

            do_actor = self.update_count % freq == 0
            if do_actor:
                a_loss = self.actor_loss(batch)
                self.actor_opt.zero_grad(set_to_none=True)
                a_loss.backward()
                metrics = {**metrics, "actor_loss": a_loss.detach()}

            self.reducer.reduce()
            self.reducer.wait()
            if getattr(self.sys, "max_grad_norm", None):
                nn.utils.clip_grad_norm_(self.q_online.parameters(), float(self.sys.max_grad_norm))
                nn.utils.clip_grad_norm_(self.actor.parameters(), float(self.sys.max_grad_norm))
            self.q_opt.step()
            from stoix_amd.parallel.dist import polyak_update

            if do_actor:
                self.actor_opt.step()
                with torch.no_grad():
                    polyak_update(self.actor.parameters(), self.actor_target.parameters(), tau)
            with torch.no_grad():
                polyak_update(self.q_online.parameters(), self.q_target.parameters(), tau)
            self.update_count += 1
        return metrics


    # ------------------------------------------------------- graph support

    @property
    def graph_capturable(self) -> bool:
        # the delayed-actor branch (update_count % policy_frequency) varies
        # per call; a single captured graph is only valid at frequency 1
        return (
            getattr(self.env, "_hip", None) is not None
            or getattr(self.env, "capture_safe", False)
        ) and int(getattr(self.sys, "policy_frequency", 1)) == 1

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
        }

    def snapshot_params(self):
        return {"actor": {k: v.clone() for k, v in self.actor.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.actor.load_state_dict(snap["actor"])
