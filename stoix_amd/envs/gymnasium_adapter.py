"""Gymnasium vector-env adapter + factory (Sebulba CPU on-ramp).

Parity with the reference's second CPU path
(/root/reference/stoix/wrappers/gymnasium.py:12 ``VecGymToStoa`` and
/root/reference/stoix/utils/env_factory.py:71-86 ``GymnasiumFactory``):
wraps anything speaking the gymnasium ``VectorEnv`` API (reset(seed)->
(obs, info); step(actions)->(obs, reward, terminated, truncated, info) with
autoreset + ``final_observation`` in info) and reconstructs the TimeStep
contract of §8.7: termination => discount 0, truncation => discount 1 with
step_type TRUNCATED, autoreset observation in ``observation`` and the true
final observation in ``extras["next_obs"]``, device-side episode metrics.

gymnasium itself is not installed in this offline image; the adapter is
duck-typed (tested against a mock vector env in tests/test_sebulba.py) and
``GymnasiumFactory`` imports gymnasium lazily with a clear error.
"""
from __future__ import annotations

import threading

import numpy as np
import torch

from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace
from stoix_amd.types import StepType, TimeStep

Tensor = torch.Tensor


def _to_tensor(x, dtype=torch.float32) -> Tensor:
    if isinstance(x, torch.Tensor):
        return x.to(dtype)
    return torch.as_tensor(np.asarray(x)).to(dtype)


class VecGymToStoa:
    """Stateful CPU env over a gymnasium ``VectorEnv``-shaped object.

    Exposes the same surface the Sebulba actors use from the native pool
    envs: ``num_envs``, ``observation_space``/``action_space``, ``reset()``
    and ``step(action)`` returning TimeSteps with the §8.7 contract.
    """

    def __init__(self, venv, seed: int = 0):
        self._venv = venv
        self.num_envs = int(venv.num_envs)
        self.seed = int(seed)
        self.device = torch.device("cpu")
        single_obs = getattr(venv, "single_observation_space", None)
        single_act = getattr(venv, "single_action_space", None)
        self.observation_space = self._convert_obs_space(single_obs)
        self.action_space = self._convert_act_space(single_act)
        self._discrete = isinstance(self.action_space, DiscreteSpace)
        B = self.num_envs
        self._ep_return = torch.zeros(B)
        self._ep_length = torch.zeros(B, dtype=torch.int32)
        self._last_ep_return = torch.zeros(B)
        self._last_ep_length = torch.zeros(B, dtype=torch.int32)

    @staticmethod
    def _convert_obs_space(sp) -> BoxSpace:
        shape = tuple(getattr(sp, "shape", ()) or ())
        lo = getattr(sp, "low", None)
        hi = getattr(sp, "high", None)
        lo_f = float(np.min(lo)) if lo is not None else -np.inf
        hi_f = float(np.max(hi)) if hi is not None else np.inf
        return BoxSpace(shape, lo_f, hi_f)

    @staticmethod
    def _convert_act_space(sp):
        n = getattr(sp, "n", None)
        if n is not None:
            return DiscreteSpace(int(n))
        shape = tuple(getattr(sp, "shape", ()) or ())
        lo = getattr(sp, "low", None)
        hi = getattr(sp, "high", None)
        return BoxSpace(shape, float(np.min(lo)), float(np.max(hi)))

    # ----------------------------------------------------------------- api

    def reset(self) -> TimeStep:
        obs, _info = self._venv.reset(seed=self.seed)
        obs_t = _to_tensor(obs)
        B = self.num_envs
        self._ep_return.zero_()
        self._ep_length.zero_()
        return TimeStep(
            step_type=torch.full((B,), StepType.FIRST, dtype=torch.uint8),
            reward=torch.zeros(B),
            discount=torch.ones(B),
            observation=obs_t,
            extras={
                "next_obs": obs_t.clone(),
                "episode_metrics": {
                    "episode_return": self._last_ep_return.clone(),
                    "episode_length": self._last_ep_length.to(torch.float32),
                    "is_terminal_step": torch.zeros(B, dtype=torch.bool),
                },
            },
        )

    def step(self, action: Tensor) -> TimeStep:
        if isinstance(action, torch.Tensor):
            a = action.detach().cpu().numpy()
            if self._discrete:
                a = a.astype(np.int64)
        else:
            a = np.asarray(action)
        obs, reward, terminated, truncated, info = self._venv.step(a)
        obs_t = _to_tensor(obs)
        reward_t = _to_tensor(reward)
        term_t = _to_tensor(terminated, torch.bool)
        trunc_t = _to_tensor(truncated, torch.bool) & ~term_t
        done = term_t | trunc_t

        # true final observation: gymnasium's vector autoreset returns the
        # RESET obs in `obs` for done envs and stashes the terminal one in
        # info (key differs by version; accept both spellings)
        next_obs = obs_t.clone()
        finals = None
        for key in ("final_observation", "final_obs"):
            if isinstance(info, dict) and key in info and info[key] is not None:
                finals = info[key]
                break
        if finals is not None:
            for i, f in enumerate(finals):
                if f is not None:
                    next_obs[i] = _to_tensor(f)

        self._ep_return += reward_t
        self._ep_length += 1
        self._last_ep_return = torch.where(done, self._ep_return, self._last_ep_return)
        self._last_ep_length = torch.where(done, self._ep_length, self._last_ep_length)
        self._ep_return = torch.where(done, torch.zeros_like(self._ep_return), self._ep_return)
        self._ep_length = torch.where(done, torch.zeros_like(self._ep_length), self._ep_length)

        step_type = torch.where(
            term_t,
            torch.tensor(StepType.TERMINATED, dtype=torch.uint8),
            torch.where(
                trunc_t,
                torch.tensor(StepType.TRUNCATED, dtype=torch.uint8),
                torch.tensor(StepType.MID, dtype=torch.uint8),
            ),
        )
        discount = torch.where(term_t, 0.0, 1.0).to(torch.float32)
        return TimeStep(
            step_type=step_type,
            reward=reward_t,
            discount=discount,
            observation=obs_t,
            extras={
                "next_obs": next_obs,
                "episode_metrics": {
                    "episode_return": self._last_ep_return.clone(),
                    "episode_length": self._last_ep_length.to(torch.float32),
                    "is_terminal_step": done,
                },
            },
        )

    def close(self) -> None:
        close = getattr(self._venv, "close", None)
        if close is not None:
            close()


class GymnasiumFactory:
    """Thread-safe factory of fresh ``VecGymToStoa`` envs with unique seeds
    (reference env_factory.py:71-86). Needs the real gymnasium package."""

    def __init__(self, task_id: str, seed: int = 0, async_envs: bool = False, **env_kwargs):
        self.task_id = task_id
        self.base_seed = int(seed)
        self.async_envs = async_envs
        self.env_kwargs = env_kwargs
        self._lock = threading.Lock()
        self._count = 0
        try:
            import gymnasium  # noqa: F401
        except ImportError as e:  # pragma: no cover - offline image
            raise ImportError(
                "GymnasiumFactory needs the 'gymnasium' package, which is "
                "not installed in this offline image. The adapter itself "
                "(VecGymToStoa) works with any VectorEnv-shaped object."
            ) from e

    def __call__(self, num_envs: int) -> VecGymToStoa:
        import gymnasium as gym

        with self._lock:
            idx = self._count
            self._count += 1
        seed = self.base_seed + 7919 * (idx + 1)
        ctor = gym.vector.AsyncVectorEnv if self.async_envs else gym.vector.SyncVectorEnv
        venv = ctor(
            [lambda: gym.make(self.task_id, **self.env_kwargs) for _ in range(num_envs)]
        )
        return VecGymToStoa(venv, seed=seed)
