"""LunarLander-class 2-D rocket-landing environment (vectorised).

Fills the role of envpool/gym LunarLander-v2 in the reference's env suite
(/root/reference/stoix/configs/env/envpool/lunarlander.yaml).
Box2D is not installable offline; this is an original, fully tensorised
rigid-body implementation with the gym contract: 8-dim observation
(x, y, vx, vy, angle, vangle, left-contact, right-contact), 4 discrete
actions (noop / left engine / main engine / right engine), potential-based
shaping reward with fuel costs, +100 for landing at rest on the pad /
-100 for crashing.

Dynamics: planar rigid body under gravity; the main engine thrusts along
the body axis, side engines apply lateral force + torque; two legs at
body-frame offsets make ground contacts (y=0 plane, pad between
x = +-PAD_W). Crash = body/leg touching ground with excess speed or
attitude; success = both legs down, near-zero velocity.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace

DT = 0.05
GRAV = -1.6
MAIN_F = 4.2
SIDE_F = 0.9
SIDE_TORQUE = 1.4
ANG_DAMP = 0.2
LEG_X = 0.22
PAD_W = 0.3
X_LIM = 1.5
MAIN_COST = 0.30
SIDE_COST = 0.03


class LunarLander(StatefulVecEnv):
    max_episode_steps = 1000

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((8,), -float("inf"), float("inf"))
        self.action_space = DiscreteSpace(4)

    def _reset_fn(self, n: int) -> State:
        s = torch.zeros(n, 6, device=self.device)
        s[:, 0] = self.rand(n, lo=-0.3, hi=0.3)   # x
        s[:, 1] = 1.3                              # y
        s[:, 2] = self.rand(n, lo=-0.2, hi=0.2)   # vx
        s[:, 3] = self.rand(n, lo=-0.1, hi=0.0)   # vy
        s[:, 4] = self.rand(n, lo=-0.1, hi=0.1)   # angle
        s[:, 5] = self.rand(n, lo=-0.05, hi=0.05)  # vangle
        return {"s": s}

    @staticmethod
    def _legs(s: Tensor) -> Tuple[Tensor, Tensor]:
        """World y of the two leg tips (body-frame x = +-LEG_X, y = -0.1)."""
        ca, sa = torch.cos(s[:, 4]), torch.sin(s[:, 4])
        ly = s[:, 1] + (-LEG_X) * sa + (-0.1) * ca
        ry = s[:, 1] + (LEG_X) * sa + (-0.1) * ca
        return ly, ry

    def _potential(self, s: Tensor, lc: Tensor, rc: Tensor) -> Tensor:
        """Gym-style shaping potential (higher = closer to a good landing)."""
        dist = torch.sqrt(s[:, 0] ** 2 + s[:, 1] ** 2)
        speed = torch.sqrt(s[:, 2] ** 2 + s[:, 3] ** 2)
        return (
            -100.0 * dist - 100.0 * speed - 100.0 * s[:, 4].abs()
            + 10.0 * lc.float() + 10.0 * rc.float()
        )

    def _obs_fn(self, state: State) -> Tensor:
        s = state["s"]
        ly, ry = self._legs(s)
        return torch.cat(
            [s, (ly <= 0.0).float().unsqueeze(-1), (ry <= 0.0).float().unsqueeze(-1)],
            dim=-1,
        )

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        s = state["s"].clone()
        a = action.long().clamp(0, 3)
        ly0, ry0 = self._legs(s)
        phi0 = self._potential(s, ly0 <= 0, ry0 <= 0)

        main = (a == 2).float()
        left = (a == 1).float()
        right = (a == 3).float()
        ca, sa = torch.cos(s[:, 4]), torch.sin(s[:, 4])
        # main engine thrusts along +body-y; side engines push laterally
        # and torque the body (left engine fires rightward -> rotates CCW)
        ax = main * MAIN_F * (-sa) + (right - left) * SIDE_F * ca * (-1.0)
        ay = main * MAIN_F * ca + GRAV + (right - left) * SIDE_F * (-sa)
        aang = (left - right) * SIDE_TORQUE - ANG_DAMP * s[:, 5]
        s[:, 2] = s[:, 2] + DT * ax
        s[:, 3] = s[:, 3] + DT * ay
        s[:, 5] = s[:, 5] + DT * aang
        s[:, 0] = s[:, 0] + DT * s[:, 2]
        s[:, 1] = s[:, 1] + DT * s[:, 3]
        s[:, 4] = s[:, 4] + DT * s[:, 5]

        ly, ry = self._legs(s)
        lc, rc = ly <= 0.0, ry <= 0.0
        # shaping potential BEFORE the contact clamps: zeroing the descent
        # velocity on touchdown is a discontinuous state change, and
        # evaluating the potential after it would hand out a free
        # +100*|vy| for every leg tap (a farmable reward exploit a CPU
        # PPO run actually found: +2874-return hover-tap cycles)
        phi1 = self._potential(s, lc, rc)
        # crash condition on the PRE-clamp impact state: the support clamp
        # below zeroes vy for two-leg-supported rows, so checking velocity
        # after it would score a full-speed slam as a soft landing (another
        # farmable exploit of the same family as the potential-order one)
        impact_vy, impact_ang = s[:, 3], s[:, 4]
        supported = lc & rc
        body_down = s[:, 1] <= 0.02
        crash = (
            (body_down & ~supported)
            | ((lc | rc) & ((impact_vy.abs() > 0.6) | (impact_ang.abs() > 0.6)))
            | (s[:, 0].abs() > X_LIM)
        )
        # leg contacts: hold the lander up (simple support: zero downward
        # motion, damp horizontal drift)
        s[:, 3] = torch.where(supported & (s[:, 3] < 0), torch.zeros_like(s[:, 3]), s[:, 3])
        s[:, 2] = torch.where(supported, s[:, 2] * 0.7, s[:, 2])
        s[:, 5] = torch.where(supported, s[:, 5] * 0.7, s[:, 5])
        s[:, 1] = torch.where(supported & (s[:, 1] < 0.1), torch.full_like(s[:, 1], 0.1), s[:, 1])
        on_pad = s[:, 0].abs() <= PAD_W
        rest = (
            supported & on_pad & ~crash
            & (s[:, 2].abs() < 0.05) & (s[:, 3].abs() < 0.05)
            & (s[:, 4].abs() < 0.2) & (s[:, 5].abs() < 0.1)
        )
        reward = (
            (phi1 - phi0)
            - MAIN_COST * main - SIDE_COST * (left + right)
            + 100.0 * rest.float() - 100.0 * crash.float()
        )
        terminated = crash | rest
        return {"s": s}, reward, terminated
