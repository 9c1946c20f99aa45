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
    # morphology tier knobs (the reference's kinetix env_size tiers
    # small/medium/large scale scene complexity; here they scale the
    # articulation: link count and per-episode length ranges)
    NLINK = 2
    LEN_LO, LEN_HI = 0.3, 0.7
    max_episode_steps = 150
    capture_safe = True
    solved_return_threshold = -10.0

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        L = self.NLINK
        # obs: [cos q_i, sin q_i] per joint, dq (L), goal (2), tip (2), len (L)
        self.observation_space = BoxSpace((3 * L + L + 4,), -10.0, 10.0)
        self.action_space = BoxSpace((L,), -1.0, 1.0)

    def _reset_fn(self, n: int) -> State:
        L = self.NLINK
        q = self.rand(n, L, lo=-3.14159, hi=3.14159)
        dq = torch.zeros(n, L, device=self.device)
        # procedural morphology: per-episode link lengths
        lengths = self.rand(n, L, lo=self.LEN_LO, hi=self.LEN_HI)
        # goal inside the annulus the arm can actually reach
        reach_max = lengths.sum(-1)
        reach_min = (2.0 * lengths.max(dim=-1).values - reach_max).clamp(min=0.0) + 0.05
        r = reach_min + self.rand(n) * (0.95 * reach_max - reach_min).clamp(min=0.01)
        ang = self.rand(n, lo=-3.14159, hi=3.14159)
        goal = torch.stack([r * torch.cos(ang), r * torch.sin(ang)], dim=-1)
        return {"q": q, "dq": dq, "len": lengths, "goal": goal}

    @staticmethod
    def _tip(q: Tensor, lengths: Tensor) -> Tensor:
        a = torch.cumsum(q, dim=-1)  # absolute link angles
        x = (lengths * torch.cos(a)).sum(-1)
        y = (lengths * torch.sin(a)).sum(-1)
        return torch.stack([x, y], dim=-1)

    def _obs_fn(self, state: State) -> Tensor:
        q, dq = state["q"], state["dq"]
        tip = self._tip(q, state["len"])
        trig = torch.stack([torch.cos(q), torch.sin(q)], dim=-1).flatten(1)
        return torch.cat([trig, dq, state["goal"], tip, state["len"]], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        q, dq = state["q"], state["dq"]
        torque = action.reshape(-1, self.NLINK).clamp(-1.0, 1.0) * TORQUE
        # inertia per joint grows with the outboard link lengths it swings
        lengths = state["len"]
        outboard = lengths.flip(-1).cumsum(-1).flip(-1)  # sum of links j..L-1
        inertia = outboard.pow(2).clamp(min=0.05)
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


class ProceduralReacherSmall(ProceduralReacher):
    """kinetix env_size small tier: 2 links, narrow morphology range."""

    LEN_LO, LEN_HI = 0.4, 0.6


class ProceduralReacher3(ProceduralReacher):
    """kinetix env_size large tier: 3-link arm, full morphology range —
    redundant kinematics the policy must resolve per episode."""

    NLINK = 3
    max_episode_steps = 200
