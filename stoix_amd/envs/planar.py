"""Planar locomotion environments: HalfCheetah-class and Hopper-class.

Fill the roles of Brax halfcheetah / hopper in the reference's env suite
(/root/reference/stoix/configs/env/brax/{halfcheetah,hopper}.yaml; SURVEY
§8.8). Brax is JAX-only and cannot be ported; these are original,
self-contained 2-D (sagittal-plane) rigid-body simulations with the MuJoCo
action/observation contracts:

  * HalfCheetah: 6 torque actuators in [-1, 1] (back thigh/shin/foot, front
    thigh/shin/foot), 17-dim observation (z, pitch, qpos 6, vx, vz,
    pitch-rate, qvel 6), no early termination, reward = forward velocity
    - 0.1 * ctrl cost.
  * Hopper: 3 torque actuators (thigh, leg, foot), 11-dim observation
    (z, pitch, qpos 3, vx, vz, pitch-rate, qvel 3), healthy-range
    termination (z and pitch bounds), reward = forward velocity + healthy
    bonus - 1e-3 * ctrl cost.

Physics model (semi-implicit Euler, 4 substeps — the same penalty-contact
formulation as stoix_amd/envs/ant.py, reduced to the x-z plane): the torso
is a planar free body (x, z, pitch); every joint is a damped inertial DOF
with soft angle limits; foot contact points (positions derived from the leg
joint chain, rotated by the torso pitch) make spring-damper ground contacts
whose normal + friction forces and moments react on the torso.

These are completeness-tier envs: they run the generic torch vectorised
path on CPU and GPU (no fused HIP step kernel — only the BASELINE.json
benchmark envs get kernel-side steps)."""
from __future__ import annotations

from typing import List, Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace


class PlanarLocomotion(StatefulVecEnv):
    """Shared planar (x-z) torso + joint-chain legs + penalty contacts.

    State row layout: x, z, pitch, vx, vz, pitchvel, qpos[NJ], qvel[NJ].
    Subclasses define the joint chain via LEGS: a list of
    (hip_x_offset, [joint indices in chain order]) — the chain's segments
    all have length SEG_LEN and the chain's end point is the foot.
    """

    NJ: int = 0
    LEGS: List[Tuple[float, List[int]]] = []
    SEG_LEN = 0.25

    TORSO_MASS = 9.0
    TORSO_INERTIA = 0.35
    TORSO_Z0 = 0.7
    JOINT_INERTIA = 0.06
    JOINT_DAMPING = 1.5
    GEAR = 30.0
    LIMIT = 1.0
    LIMIT_K = 60.0
    CONTACT_KN = 9e3
    CONTACT_KD = 200.0
    FRICTION = 0.9
    GRAVITY = -9.81
    DT = 0.05
    SUBSTEPS = 4
    CTRL_COST = 0.1
    HEALTHY_REWARD = 0.0
    FORWARD_W = 1.0
    Z_MIN, Z_MAX = -float("inf"), float("inf")
    PITCH_MAX = float("inf")
    max_episode_steps = 1000

    def __init__(self, num_envs, device="cpu", seed=0, dtype=torch.float32, **kw):
        super().__init__(num_envs, device, seed)
        self.dtype = dtype
        self.observation_space = BoxSpace((2 + self.NJ + 3 + self.NJ,), -float("inf"), float("inf"))
        self.action_space = BoxSpace((self.NJ,), -1.0, 1.0)

    # ------------------------------------------------------------ state ops

    def _reset_fn(self, n: int) -> State:
        s = torch.zeros(n, 6 + 2 * self.NJ, dtype=self.dtype, device=self.device)
        s[:, 1] = self.TORSO_Z0
        s[:, 6 : 6 + self.NJ] = self.rand(n, self.NJ, lo=-0.1, hi=0.1)
        s[:, 6 + self.NJ :] = self.rand(n, self.NJ, lo=-0.05, hi=0.05)
        return {"s": s}

    def _obs_fn(self, state: State) -> Tensor:
        s = state["s"]
        # z, pitch, qpos, vx, vz, pitchvel, qvel (x itself is excluded,
        # matching the MuJoCo convention of position-agnostic observations)
        return torch.cat([s[:, 1:3], s[:, 6 : 6 + self.NJ], s[:, 3:6], s[:, 6 + self.NJ :]], dim=-1)

    # -------------------------------------------------------------- physics

    def _feet(self, qpos: Tensor) -> Tensor:
        """Body-frame foot points [B, n_feet, 2] from the joint chains."""
        feet = []
        for hip_x, chain in self.LEGS:
            ang = torch.zeros_like(qpos[:, 0])
            fx = torch.full_like(ang, hip_x)
            fz = torch.zeros_like(ang)
            for j in chain:
                ang = ang + qpos[:, j]
                fx = fx + self.SEG_LEN * torch.sin(ang)
                fz = fz - self.SEG_LEN * torch.cos(ang)
            feet.append(torch.stack([fx, fz], dim=-1))
        return torch.stack(feet, dim=1)

    def _substep(self, s: Tensor, torque: Tensor, dt: float) -> Tensor:
        NJ = self.NJ
        x, z, pitch = s[:, 0], s[:, 1], s[:, 2]
        vx, vz, pv = s[:, 3], s[:, 4], s[:, 5]
        qpos = s[:, 6 : 6 + NJ]
        qvel = s[:, 6 + NJ :]

        limit_tau = -self.LIMIT_K * (
            torch.relu(qpos - self.LIMIT) - torch.relu(-self.LIMIT - qpos)
        )
        qacc = (self.GEAR * torque - self.JOINT_DAMPING * qvel + limit_tau) / self.JOINT_INERTIA
        qvel = qvel + dt * qacc
        qpos = qpos + dt * qvel

        # foot contacts: rotate body-frame points by pitch into the world
        body = self._feet(qpos)  # [B, F, 2]
        c, sn = torch.cos(pitch).unsqueeze(-1), torch.sin(pitch).unsqueeze(-1)
        rx = c * body[..., 0] + sn * body[..., 1]
        rz = -sn * body[..., 0] + c * body[..., 1]
        foot_z = z.unsqueeze(-1) + rz
        # point velocity = torso vel + omega x r (planar: omega x r = (-w*rz, w*rx))
        foot_vx = vx.unsqueeze(-1) - pv.unsqueeze(-1) * rz
        foot_vz = vz.unsqueeze(-1) + pv.unsqueeze(-1) * rx
        pen = (-foot_z).clamp(min=0.0)
        fn = (self.CONTACT_KN * pen - self.CONTACT_KD * foot_vz).clamp(min=0.0)
        fn = torch.where(pen > 0, fn, torch.zeros_like(fn))
        ft = -self.FRICTION * fn * torch.tanh(4.0 * foot_vx)
        total_fx = ft.sum(-1)
        total_fz = fn.sum(-1)
        total_tau = (rx * fn - rz * ft).sum(-1)  # planar cross r x F

        vx = vx + dt * total_fx / self.TORSO_MASS
        vz = vz + dt * (total_fz / self.TORSO_MASS + self.GRAVITY)
        pv = pv + dt * (total_tau / self.TORSO_INERTIA - 0.8 * pv)
        x = x + dt * vx
        z = z + dt * vz
        pitch = pitch + dt * pv
        return torch.cat(
            [torch.stack([x, z, pitch, vx, vz, pv], dim=-1), qpos, qvel], dim=-1
        )

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        s = state["s"]
        a = action.clamp(-1.0, 1.0).to(s.dtype)
        x_before = s[:, 0].clone()
        dt = self.DT / self.SUBSTEPS
        for _ in range(self.SUBSTEPS):
            s = self._substep(s, a, dt)
        forward_vel = (s[:, 0] - x_before) / self.DT
        ctrl_cost = self.CTRL_COST * (a**2).sum(-1)
        healthy = (
            (s[:, 1] > self.Z_MIN)
            & (s[:, 1] < self.Z_MAX)
            & (s[:, 2].abs() < self.PITCH_MAX)
            & torch.isfinite(s).all(dim=-1)
        )
        reward = self.FORWARD_W * forward_vel - ctrl_cost + torch.where(
            healthy, self.HEALTHY_REWARD, 0.0
        )
        terminated = ~healthy if self.TERMINATES else torch.zeros_like(healthy)
        s = torch.where(torch.isfinite(s), s, torch.zeros_like(s))
        return {"s": s}, reward.to(torch.float32), terminated

    TERMINATES = True


class HalfCheetah(PlanarLocomotion):
    """6-actuator planar runner; never terminates early (MuJoCo contract)."""

    NJ = 6
    # back leg hangs from x=-0.5 (thigh 0, shin 1, foot 2), front from x=+0.5
    LEGS = [(-0.5, [0, 1, 2]), (0.5, [3, 4, 5])]
    SEG_LEN = 0.16
    TORSO_Z0 = 0.55
    CTRL_COST = 0.1
    TERMINATES = False


class Hopper(PlanarLocomotion):
    """3-actuator one-legged hopper with healthy-range termination."""

    NJ = 3
    LEGS = [(0.0, [0, 1, 2])]
    SEG_LEN = 0.22
    TORSO_MASS = 4.0
    TORSO_INERTIA = 0.15
    TORSO_Z0 = 0.75
    GEAR = 25.0
    CTRL_COST = 1e-3
    HEALTHY_REWARD = 1.0
    Z_MIN = 0.45
    PITCH_MAX = 0.6
    TERMINATES = True
