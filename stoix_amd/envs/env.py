"""Vectorised environment engine.

The reference composes a functional JAX env with core wrappers
AddRNGKey -> RecordEpisodeMetrics -> AutoReset(next_obs_in_extras=True) ->
VmapWrapper (/root/reference/stoix/utils/make_env.py:29-61). This build is
MI355X-first: an env IS batched (struct-of-arrays state tensors resident on
one device), and the wrapper semantics are implemented once in
``StatefulVecEnv`` — autoreset with the true final observation exposed in
``extras["next_obs"]``, episode metrics, and step-limit truncation (the
semantics every Anakin system's bootstrap-value logic depends on, SURVEY.md
§8.7).

Subclasses implement two methods:
  * ``_reset_fn(n)`` -> state dict of [n, ...] tensors
  * ``_step_fn(state, action)`` -> (state, reward, terminated) where
    ``state`` is updated in place or replaced.
and one observation function ``_obs_fn(state)`` -> [B, obs_dim].

GPU fast path: envs that have a HIP kernel implementation (CartPole, Ant)
override ``step`` wholesale on CUDA devices via ``stoix_amd.ops`` — the
kernel fuses dynamics + termination + autoreset + metrics in one launch.
"""
from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import torch

from stoix_amd.envs.spaces import Space
from stoix_amd.types import StepType, TimeStep

Tensor = torch.Tensor
State = Dict[str, Tensor]


class StatefulVecEnv:
    """Batched stateful environment with built-in autoreset + metrics."""

    observation_space: Space
    action_space: Space
    max_episode_steps: int = 10**9
    solved_return_threshold: Optional[float] = None
    # subclass opt-in: _step_fn/_reset_fn/_obs_fn are free of host-synced
    # branches, so step() can run under hip-graph capture (graph_mode takes
    # the unconditional-autoreset path and the default graph-aware RNG)
    capture_safe: bool = False
    graph_mode: bool = False

    def __init__(self, num_envs: int, device: torch.device | str = "cpu", seed: int = 0):
        self.num_envs = int(num_envs)
        self.device = torch.device(device)
        self.gen = torch.Generator(device=self.device)
        self.gen.manual_seed(int(seed))
        self._state: State = {}
        self._step_count = torch.zeros(self.num_envs, dtype=torch.int32, device=self.device)
        self._ep_return = torch.zeros(self.num_envs, dtype=torch.float32, device=self.device)
        self._ep_length = torch.zeros(self.num_envs, dtype=torch.int32, device=self.device)
        self._last_ep_return = torch.zeros(self.num_envs, dtype=torch.float32, device=self.device)
        self._last_ep_length = torch.zeros(self.num_envs, dtype=torch.int32, device=self.device)
        # device-resident step-type constants: creating them per step would
        # be a pageable H2D copy — illegal inside hip-graph capture
        self._st_terminated = torch.tensor(StepType.TERMINATED, dtype=torch.uint8, device=self.device)
        self._st_truncated = torch.tensor(StepType.TRUNCATED, dtype=torch.uint8, device=self.device)
        self._st_mid = torch.tensor(StepType.MID, dtype=torch.uint8, device=self.device)
        # running count of completed episodes (device scalar; graph-legal):
        # lets graph-replay learners tell fresh latched metrics from stale
        # ones without a host sync
        self._done_count = torch.zeros((), dtype=torch.long, device=self.device)

    # ------------------------------------------------------- subclass hooks

    def _reset_fn(self, n: int) -> State:
        raise NotImplementedError

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        raise NotImplementedError

    def _obs_fn(self, state: State) -> Tensor:
        raise NotImplementedError

    # --------------------------------------------------------------- public

    def reset(self) -> TimeStep:
        self._state = self._reset_fn(self.num_envs)
        self._step_count.zero_()
        self._ep_return.zero_()
        self._ep_length.zero_()
        obs = self._obs_fn(self._state)
        B = self.num_envs
        dev = self.device
        return TimeStep(
            step_type=torch.full((B,), StepType.FIRST, dtype=torch.uint8, device=dev),
            reward=torch.zeros(B, dtype=torch.float32, device=dev),
            discount=torch.ones(B, dtype=torch.float32, device=dev),
            observation=obs,
            extras=self._make_extras(obs, torch.zeros(B, dtype=torch.bool, device=dev)),
        )

    def step(self, action: Tensor) -> TimeStep:
        self._state, reward, terminated = self._step_fn(self._state, action)
        reward = reward.to(torch.float32)
        self._step_count += 1
        truncated = (self._step_count >= self.max_episode_steps) & ~terminated
        done = terminated | truncated

        self._ep_return += reward
        self._ep_length += 1
        self._done_count += done.sum()
        # Latch completed-episode metrics at the terminal step.
        self._last_ep_return = torch.where(done, self._ep_return, self._last_ep_return)
        self._last_ep_length = torch.where(done, self._ep_length, self._last_ep_length)

        final_obs = self._obs_fn(self._state)

        # Autoreset: regenerate state for done envs; observation returned is
        # the RESET obs there, the true final obs goes to extras["next_obs"].
        # graph_mode takes this branch unconditionally (a host-synced
        # bool(done.any()) is not capture-legal).
        if self.graph_mode or bool(done.any()):
            fresh = self._reset_fn(self.num_envs)
            mask = done
            for k in self._state:
                m = mask.view(-1, *([1] * (self._state[k].dim() - 1)))
                self._state[k] = torch.where(m, fresh[k], self._state[k])
            self._step_count = torch.where(mask, torch.zeros_like(self._step_count), self._step_count)
            self._ep_return = torch.where(mask, torch.zeros_like(self._ep_return), self._ep_return)
            self._ep_length = torch.where(mask, torch.zeros_like(self._ep_length), self._ep_length)
            obs = self._obs_fn(self._state)
        else:
            obs = final_obs

        step_type = torch.where(
            terminated,
            self._st_terminated,
            torch.where(truncated, self._st_truncated, self._st_mid),
        )
        discount = torch.where(terminated, 0.0, 1.0).to(torch.float32)
        return TimeStep(
            step_type=step_type,
            reward=reward,
            discount=discount,
            observation=obs,
            extras=self._make_extras(final_obs, done),
        )

    def _make_extras(self, next_obs: Tensor, done: Tensor) -> Dict[str, Any]:
        return {
            "next_obs": next_obs,
            "episode_metrics": {
                "episode_return": self._last_ep_return.clone(),
                "episode_length": self._last_ep_length.clone().to(torch.float32),
                "is_terminal_step": done.clone(),
            },
        }

    def prepare_for_graph_capture(self) -> None:
        """Switch to capture-safe modes: unconditional autoreset and the
        default (graph-aware) CUDA generator. Only valid when the subclass
        declares ``capture_safe`` (or has a HIP step kernel)."""
        self.graph_mode = True
        self.gen = None

    # ------------------------------------------------------------- utility

    def rand(self, *shape, lo: float = 0.0, hi: float = 1.0) -> Tensor:
        u = torch.rand(shape, device=self.device, generator=self.gen)
        return lo + u * (hi - lo)

    def randn(self, *shape) -> Tensor:
        return torch.randn(shape, device=self.device, generator=self.gen)

    def randint(self, high: int, *shape) -> Tensor:
        return torch.randint(0, high, shape, device=self.device, generator=self.gen)


def latched_episode_metrics(env: "StatefulVecEnv", learner) -> Dict[str, Tensor]:
    """Episode metrics for graph-replay learners, read from the env's
    latched device buffers with a freshness flag: ``has_final`` is a device
    bool that is True only when at least one episode completed since the
    previous call (no host sync; the flag is resolved host-side at log
    time). Closes the stale-latched-metrics gap from ADVICE r1."""
    dc = env._done_count.clone()
    prev = getattr(learner, "_dc_prev", None)
    fresh = (dc > prev) if prev is not None else (dc > 0)
    learner._dc_prev = dc
    return {
        "episode_return": env._last_ep_return,
        "episode_length": env._last_ep_length.to(torch.float32),
        "has_final": fresh,
    }


def get_final_step_metrics(metrics: Dict[str, Tensor]) -> Tuple[Dict[str, Tensor], bool]:
    """Filter episode metrics to completed episodes (reference
    stoa.get_final_step_metrics; consumed at ff_ppo.py:624)."""
    mask = metrics["is_terminal_step"]
    has_final = bool(mask.any())
    out = {}
    for k, v in metrics.items():
        if k == "is_terminal_step":
            continue
        out[k] = v[mask] if has_final else v[:0]
    return out, has_final
