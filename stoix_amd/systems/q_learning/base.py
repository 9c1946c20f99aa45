"""Shared off-policy Anakin machinery for the DQN family.

Structure parity with /root/reference/stoix/systems/q_learning/ff_dqn.py:
warmup buffer fill (get_warmup_fn :37-89), short rollouts feeding an
HBM-resident item buffer (:135-142), many sampled update epochs per rollout
(:185-205), polyak target update every update (:207-209), reward clipping
(:159-161), epsilon-greedy baked into the Q-head with separate train/eval
epsilon (:276-290). Each concrete algorithm overrides ``loss_fn``.
"""
from __future__ import annotations

import copy
from typing import Dict

import torch
import torch.nn as nn

from stoix_amd.buffers import ItemBuffer
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics
from stoix_amd.networks.distributions import EpsilonGreedy
from stoix_amd.networks.factory import build_q_network
from stoix_amd.parallel.dist import FlatGradReducer, broadcast_module

Tensor = torch.Tensor


class OffPolicyQLearner:
    """Base learner: subclasses implement ``loss_fn(batch) -> (loss, metrics)``
    and optionally ``q_values(net, obs)`` / act distribution."""

    def __init__(self, config, env: StatefulVecEnv, device: torch.device):
        self.cfg = config
        self.sys = config.system
        self.env = env
        self.device = device
        self.T = int(self.sys.rollout_length)
        self.B = env.num_envs
        self.gamma = float(self.sys.gamma)

        self.q_online = self.build_network().to(device)
        broadcast_module(self.q_online)
        self.q_target = copy.deepcopy(self.q_online)
        for p in self.q_target.parameters():
            p.requires_grad_(False)

        self.opt = torch.optim.Adam(self.q_online.parameters(), lr=float(self.sys.q_lr), eps=1e-5, capturable=device.type == "cuda")
        self.reducer = FlatGradReducer(self.q_online.parameters(), device)
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(int(config.arch.seed) * 7919 + 41)

        cap = int(self.sys.buffer_size) // max(1, int(config.arch.n_devices))
        self.buffer = ItemBuffer(cap, device=device, seed=int(config.arch.seed) + 17)
        self.batch_size = int(self.sys.batch_size)
        self.train_eps = float(self.sys.training_epsilon)
        self.eval_eps = float(getattr(self.sys, "evaluation_epsilon", 0.0))
        self.max_abs_reward = float(getattr(self.sys, "max_abs_reward", 1e6))

        self.ts = env.reset()
        self.episode_metrics: Dict[str, Tensor] = {}
        self.collect_metrics = True
        self._warmup()

    # ------------------------------------------------------------- networks

    def build_network(self) -> nn.Module:
        return build_q_network(
            self.cfg.network.actor_network,
            self.env.observation_space,
            self.env.action_space,
            epsilon=float(self.sys.training_epsilon),
        )

    def q_values(self, net: nn.Module, obs: Tensor) -> Tensor:
        """Scalar per-action Q values for acting/argmax."""
        out = net(obs)
        if isinstance(out, EpsilonGreedy):
            return out.preferences
        if hasattr(out, "q_values"):
            return out.q_values
        return out

    # --------------------------------------------------------------- acting

    @torch.no_grad()
    def _act(self, obs: Tensor, epsilon: float) -> Tensor:
        q = self.q_values(self.q_online, obs)
        return EpsilonGreedy(q, epsilon).sample(self.gen)

    @torch.no_grad()
    def act_fn(self, obs: Tensor, greedy: bool) -> Tensor:
        q = self.q_values(self.q_online, obs)
        if greedy or self.eval_eps == 0.0:
            return q.argmax(dim=-1)
        return EpsilonGreedy(q, self.eval_eps).sample(self.gen)

    # -------------------------------------------------------------- rollout

    @torch.no_grad()
    def _rollout_into_buffer(self, steps: int, random_actions: bool = False) -> None:
        ts = self.ts
        for _ in range(steps):
            obs = ts.observation
            if random_actions:
                action = self.env.action_space.sample(self.B, self.device, self.gen)
            else:
                action = self._act(obs, self.train_eps)
            next_ts = self.env.step(action)
            reward = next_ts.reward.clamp(-self.max_abs_reward, self.max_abs_reward)
            self.buffer.add(
                {
                    "obs": obs if isinstance(obs, Tensor) else obs,
                    "action": action,
                    "reward": reward,
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
        steps = max(1, int(getattr(self.sys, "warmup_steps", 16)) // self.B + 1)
        self._rollout_into_buffer(steps, random_actions=True)

    # ---------------------------------------------------------------- losses

    def loss_fn(self, batch: Dict[str, Tensor]):
        raise NotImplementedError

    # ---------------------------------------------------------------- update

    def update_step(self) -> Dict[str, Tensor]:
        self._rollout_into_buffer(self.T)
        metrics: Dict[str, Tensor] = {}
        tau = float(self.sys.tau)
        for _ in range(int(self.sys.epochs)):
            batch = self.buffer.sample(self.batch_size)
            loss, metrics = self.loss_fn(batch)
            self.opt.zero_grad(set_to_none=True)
            loss.backward()
            self.reducer.reduce()
            self.reducer.wait()
            if getattr(self.sys, "max_grad_norm", None):
                nn.utils.clip_grad_norm_(self.q_online.parameters(), float(self.sys.max_grad_norm))
            self.opt.step()
            self._polyak(tau)
        return metrics

    @torch.no_grad()
    def _polyak(self, tau: float) -> None:
        from stoix_amd.parallel.dist import polyak_update

        polyak_update(self.q_online.parameters(), self.q_target.parameters(), tau)
        for bo, bt in zip(self.q_online.buffers(), self.q_target.buffers()):
            if bt.dtype.is_floating_point:
                bt.mul_(1.0 - tau).add_(bo, alpha=tau)
            else:
                bt.copy_(bo)


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
            "q_online": dict(self.q_online.state_dict()),
            "q_target": dict(self.q_target.state_dict()),
        }

    def snapshot_params(self):
        return {"q_online": {k: v.clone() for k, v in self.q_online.state_dict().items()}}

    def load_params(self, snap) -> None:
        self.q_online.load_state_dict(snap["q_online"])
