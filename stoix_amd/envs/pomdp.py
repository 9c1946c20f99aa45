"""PopJym-class POMDP envs: velocity-masked classic control with the
start-flag + previous-action observation augmentation.

Restores the capability class of the reference's popjym suite
(/root/reference/stoix/utils/make_env.py:363-364 wraps popjym envs in
``AddStartFlagAndPrevAction``): partially observable tasks where the
optimal policy NEEDS memory — a feed-forward policy caps out, a recurrent
one solves them. The canonical instance is StatelessCartPole (velocities
masked out of the observation), the standard POMDP-ification used across
the POPGym line.

Observation layout: [masked_obs..., start_flag, prev_action_onehot...] —
the reference wrapper's exact augmentation semantics.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.classic import CartPole, Pendulum
from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace


class _MaskedObsEnv(StatefulVecEnv):
    """Base: wrap a fully-observable env class, keep only OBS_KEEP indices
    of its observation, append start flag + previous-action one-hot."""

    INNER_CLS = None
    OBS_KEEP: Tuple[int, ...] = ()

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self._inner = self.INNER_CLS(num_envs, device=device, seed=seed, **kw)
        # disable the inner HIP fast path: the wrapper drives _step_fn
        # directly (functional), so the augmentation composes on both CPU
        # and GPU torch paths
        self._inner._hip = None
        self.max_episode_steps = self._inner.max_episode_steps
        self.action_space = self._inner.action_space
        self._n_act = getattr(self.action_space, "num_values", 0) or 0
        base_dim = len(self.OBS_KEEP)
        self.observation_space = BoxSpace((base_dim + 1 + self._n_act,), -5.0, 5.0)
        self._keep = torch.tensor(self.OBS_KEEP, device=self.device)

    def _reset_fn(self, n: int) -> State:
        inner = self._inner._reset_fn(n)
        return {
            **inner,
            "_prev_a": torch.zeros(n, dtype=torch.long, device=self.device),
            "_is_start": torch.ones(n, device=self.device),
        }

    def _obs_fn(self, state: State) -> Tensor:
        inner_state = {k: v for k, v in state.items() if not k.startswith("_")}
        full = self._inner._obs_fn(inner_state)
        masked = full.index_select(-1, self._keep)
        onehot = torch.nn.functional.one_hot(
            state["_prev_a"].clamp(0, max(self._n_act - 1, 0)), max(self._n_act, 1)
        ).float()
        return torch.cat([masked, state["_is_start"].unsqueeze(-1), onehot], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        inner_state = {k: v for k, v in state.items() if not k.startswith("_")}
        new_inner, reward, terminated = self._inner._step_fn(inner_state, action)
        return (
            {
                **new_inner,
                "_prev_a": action.long(),
                "_is_start": torch.zeros_like(state["_is_start"]),
            },
            reward,
            terminated,
        )


class StatelessCartPole(_MaskedObsEnv):
    """CartPole with the two velocity components hidden (obs = [x, theta]
    + flag + prev action): the POPGym StatelessCartPole task."""

    INNER_CLS = CartPole
    OBS_KEEP = (0, 2)  # x, theta (drop x_dot, theta_dot)
    max_episode_steps = 500
    solved_return_threshold = 450.0


class NoisyStatelessCartPole(StatelessCartPole):
    """StatelessCartPole with observation noise (POPGym 'noisy' variant)."""

    NOISE = 0.1

    def _obs_fn(self, state: State) -> Tensor:
        obs = super()._obs_fn(state)
        n = obs.shape[0]
        noise = torch.randn(n, 2, device=self.device, generator=self.gen) * self.NOISE
        obs = obs.clone()
        obs[:, :2] = obs[:, :2] + noise
        return obs


class StatelessPendulum(_MaskedObsEnv):
    """Pendulum with angular velocity hidden (obs = [cos, sin] + flag +
    prev-action placeholder; continuous action -> no one-hot)."""

    INNER_CLS = Pendulum
    OBS_KEEP = (0, 1)  # cos(theta), sin(theta); drop theta_dot
    max_episode_steps = 200

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed, **kw)
        base_dim = len(self.OBS_KEEP)
        # continuous action: append the raw previous action instead
        adim = self.action_space.shape[0]
        self.observation_space = BoxSpace((base_dim + 1 + adim,), -5.0, 5.0)

    def _reset_fn(self, n: int) -> State:
        inner = self._inner._reset_fn(n)
        adim = self.action_space.shape[0]
        return {
            **inner,
            "_prev_a": torch.zeros(n, adim, device=self.device),
            "_is_start": torch.ones(n, device=self.device),
        }

    def _obs_fn(self, state: State) -> Tensor:
        inner_state = {k: v for k, v in state.items() if not k.startswith("_")}
        full = self._inner._obs_fn(inner_state)
        masked = full.index_select(-1, self._keep)
        return torch.cat(
            [masked, state["_is_start"].unsqueeze(-1), state["_prev_a"]], dim=-1
        )

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        inner_state = {k: v for k, v in state.items() if not k.startswith("_")}
        new_inner, reward, terminated = self._inner._step_fn(inner_state, action)
        return (
            {
                **new_inner,
                "_prev_a": action.float().reshape(state["_prev_a"].shape),
                "_is_start": torch.zeros_like(state["_is_start"]),
            },
            reward,
            terminated,
        )
