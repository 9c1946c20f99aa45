"""Kinetix-class procedural 2D articulated-physics task (vectorised torch).

Partially restores the capability class of the reference's kinetix suite
(/root/reference/stoix/utils/make_env.py:211-274; JAX-only external):
torque-controlled articulated 2D bodies with PER-EPISODE PROCEDURAL
variation the policy must read from its observation. This is the
reacher-class slice of that space — a 2-link arm with RANDOMISED link
lengths each episode and a random goal; kinetix's full procedural
morphology/scene generality remains gated (PARITY.md).

Dynamics: torque-driven joints with viscous damping (top-down plane, no
gravity), semi-implicit Euler. Observation: [cos q1, sin q1, cos q2,
sin q2, dq1, dq2, goal_x, goal_y, tip_x, tip_y, L1, L2] (12). Action:
2 torques in [-1, 1]. Reward: -distance(tip, goal) per step, +5 touch
bonus and termination inside the goal radius.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace

DT = 0.05
DAMP = 4.0
TORQUE = 4.0
GOAL_R = 0.10


class ProceduralReacher(StatefulVecEnv):
    max_episode_steps = 150
    capture_safe = True
    solved_return_threshold = -10.0

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((12,), -10.0, 10.0)
        self.action_space = BoxSpace((2,), -1.0, 1.0)

    def _reset_fn(self, n: int) -> State:
        q = self.rand(n, 2, lo=-3.14159, hi=3.14159)
        dq = torch.zeros(n, 2, device=self.device)
        # procedural morphology: per-episode link lengths
        lengths = self.rand(n, 2, lo=0.3, hi=0.7)
        # goal inside the annulus the arm can actually reach
        reach_max = lengths.sum(-1)
        reach_min = (lengths[:, 0] - lengths[:, 1]).abs() + 0.05
        r = reach_min + self.rand(n) * (0.95 * reach_max - reach_min).clamp(min=0.01)
        ang = self.rand(n, lo=-3.14159, hi=3.14159)
        goal = torch.stack([r * torch.cos(ang), r * torch.sin(ang)], dim=-1)
        return {"q": q, "dq": dq, "len": lengths, "goal": goal}

    @staticmethod
    def _tip(q: Tensor, lengths: Tensor) -> Tensor:
        a1 = q[:, 0]
        a2 = q[:, 0] + q[:, 1]
        x = lengths[:, 0] * torch.cos(a1) + lengths[:, 1] * torch.cos(a2)
        y = lengths[:, 0] * torch.sin(a1) + lengths[:, 1] * torch.sin(a2)
        return torch.stack([x, y], dim=-1)

    def _obs_fn(self, state: State) -> Tensor:
        q, dq = state["q"], state["dq"]
        tip = self._tip(q, state["len"])
        return torch.cat(
            [
                torch.cos(q[:, :1]), torch.sin(q[:, :1]),
                torch.cos(q[:, 1:]), torch.sin(q[:, 1:]),
                dq, state["goal"], tip, state["len"],
            ],
            dim=-1,
        )

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        q, dq = state["q"], state["dq"]
        torque = action.reshape(-1, 2).clamp(-1.0, 1.0) * TORQUE
        # inertia grows with the link lengths the joint must swing
        lengths = state["len"]
        inertia1 = (lengths[:, 0] + lengths[:, 1]).pow(2).unsqueeze(-1)
        inertia2 = lengths[:, 1:].pow(2)
        inertia = torch.cat([inertia1, inertia2], dim=-1).clamp(min=0.05)
        ddq = torque / inertia - DAMP * dq
        dq = (dq + DT * ddq).clamp(-8.0, 8.0)
        q = q + DT * dq
        tip = self._tip(q, lengths)
        dist = (tip - state["goal"]).norm(dim=-1)
        touched = dist < GOAL_R
        reward = -dist + 5.0 * touched.float()
        return (
            {"q": q, "dq": dq, "len": lengths, "goal": state["goal"]},
            reward,
            touched,
        )
