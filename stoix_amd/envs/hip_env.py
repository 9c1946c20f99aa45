"""HIP fast-path mixin for environments with fused gfx950 step kernels.

On a CUDA (ROCm) device, ``StatefulVecEnv.step`` routes through
``hip_step``: ONE kernel launch performs dynamics + termination +
truncation + metrics + autoreset for all B envs (plus a 1-thread RNG-counter
bump so the sequence is hip-graph-replayable). The extension is REQUIRED on
GPU — no silent eager fallback (stoix_amd.ops.ext(required=True)).
"""
from __future__ import annotations

from typing import Dict

import torch

from stoix_amd.types import TimeStep


class HipStepMixin:
    """Mixin for StatefulVecEnv subclasses with a fused HIP step kernel.

    Subclasses define ``HIP_KERNEL`` ('cartpole_step' / 'ant_step'),
    ``OBS_DIM``, ``STATE_KEY`` and ``_hip_action(action)`` casting."""

    HIP_KERNEL: str = ""
    OBS_DIM: int = 0
    STATE_KEY: str = "s"

    def _init_hip(self) -> None:
        self._hip = None
        if self.device.type != "cuda":
            return
        from stoix_amd import ops

        self._hip = ops.ext(required=True)
        B = self.num_envs
        dev = self.device

        def mk() -> Dict[str, torch.Tensor]:
            return {
                "obs": torch.zeros(B, self.OBS_DIM, device=dev),
                "next_obs": torch.zeros(B, self.OBS_DIM, device=dev),
                "reward": torch.zeros(B, device=dev),
                "discount": torch.zeros(B, device=dev),
                "steptype": torch.zeros(B, dtype=torch.uint8, device=dev),
                "done": torch.zeros(B, dtype=torch.uint8, device=dev),
            }

        # The kernels write into ONE stable buffer set, but step() returns
        # CLONES: returning the buffers themselves silently aliased — a
        # learner doing ``obs = ts.observation; env.step(a); buf[t] = obs``
        # stored the NEXT step's observation (stream-ordered overwrite),
        # putting actions and observations off by one. That broke PPO
        # learning on GPU while every single-step numerics test passed
        # (forensics recorded in profiles/r01_learning_curves.md). The clones cost ~6 small copies
        # per eager step; the fused rollout path (FusedPPOEngine) bypasses
        # step() entirely and keeps the zero-copy pipeline.
        self._hb: Dict[str, torch.Tensor] = mk()
        self._hb["draw"] = torch.zeros(1, dtype=torch.int32, device=dev)
        self._hip_seed = int(torch.randint(0, 2**31 - 1, (1,), generator=self.gen, device=dev).item())

    def _hip_action(self, action: torch.Tensor) -> torch.Tensor:
        return action

    def hip_step_into(
        self,
        action: torch.Tensor,
        reward_out: torch.Tensor,
        discount_out: torch.Tensor,
        steptype_out: torch.Tensor,
        draw_offset: int = 0,
        do_bump: bool = True,
    ) -> None:
        """Fused-rollout variant of step(): the kernel writes reward /
        discount / step-type straight into the caller's rollout storage
        (e.g. buf_reward[t]) instead of the stable _hb buffers — no
        per-step copy kernels. obs/next_obs still land in _hb."""
        hb = self._hb
        kern = getattr(self._hip, self.HIP_KERNEL)
        kern(
            self._state[self.STATE_KEY],
            self._hip_action(action),
            self._step_count,
            self._ep_return,
            self._ep_length,
            self._last_ep_return,
            self._last_ep_length,
            hb["obs"],
            hb["next_obs"],
            reward_out,
            discount_out,
            steptype_out,
            hb["done"],
            self.max_episode_steps,
            self._hip_seed,
            hb["draw"],
            draw_offset,
            1 if do_bump else 0,
        )
        self._done_count += hb["done"].sum()

    def step(self, action: torch.Tensor) -> TimeStep:  # type: ignore[override]
        if getattr(self, "_hip", None) is None:
            return super().step(action)  # type: ignore[misc]
        hb = self._hb
        kern = getattr(self._hip, self.HIP_KERNEL)
        kern(
            self._state[self.STATE_KEY],
            self._hip_action(action),
            self._step_count,
            self._ep_return,
            self._ep_length,
            self._last_ep_return,
            self._last_ep_length,
            hb["obs"],
            hb["next_obs"],
            hb["reward"],
            hb["discount"],
            hb["steptype"],
            hb["done"],
            self.max_episode_steps,
            self._hip_seed,
            hb["draw"],
            0,
            1,
        )
        self._done_count += hb["done"].sum()
        return TimeStep(
            step_type=hb["steptype"].clone(),
            reward=hb["reward"].clone(),
            discount=hb["discount"].clone(),
            observation=hb["obs"].clone(),
            extras={
                "next_obs": hb["next_obs"].clone(),
                "episode_metrics": {
                    "episode_return": self._last_ep_return.clone(),
                    "episode_length": self._last_ep_length.to(torch.float32),
                    "is_terminal_step": hb["done"].bool(),
                },
            },
        )
