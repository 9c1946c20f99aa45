"""Humanoid-class biped locomotion environment (vectorised rigid-body
physics).

Fills the role of Brax Humanoid in the reference's configs
(/root/reference/stoix/configs/env/brax/humanoid.yaml; BASELINE.json config
#3: Anakin SAC, HBM-resident replay). Brax is JAX-only and cannot be
ported; this is an original, self-contained biped simulation with the
MuJoCo-Humanoid action contract: 17 torque actuators in [-1, 1]
(abdomen 3, 2x hip 3, 2x knee, 2x shoulder 2, 2x elbow), forward-progress
reward with control cost and healthy-range termination. The observation is
the compact physical state (45 dims: z, quat, linvel, angvel, qpos 17,
qvel 17) rather than Brax's 244-dim cinert-augmented vector.

Physics model (semi-implicit Euler, 4 substeps): free torso rigid body;
all 17 joints are damped inertial DOFs with soft limits; the two FEET
(positions derived from hip-pitch/roll + knee angles) make penalty
spring-damper ground contacts reacting on the torso; arms/abdomen affect
only the control cost and the observation. Same formulation as
stoix_amd/envs/ant.py (and the HIP kernel humanoid_step mirrors it
kernel-side, one env per lane).
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.ant import quat_integrate, quat_rotate
from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.hip_env import HipStepMixin
from stoix_amd.envs.spaces import BoxSpace

POS, QUAT, LINVEL, ANGVEL, QPOS, QVEL = 0, 3, 7, 10, 13, 30
STATE_DIM = 47
OBS_DIM = 45
ACT_DIM = 17

# joint index map (17): 0-2 abdomen(z,y,x); 3-5 R hip(x,z,y); 6 R knee;
# 7-9 L hip(x,z,y); 10 L knee; 11-12 R shoulder; 13 R elbow;
# 14-15 L shoulder; 16 L elbow
R_HIP_X, R_HIP_Y, R_KNEE = 3, 5, 6
L_HIP_X, L_HIP_Y, L_KNEE = 7, 9, 10


class Humanoid(HipStepMixin, StatefulVecEnv):
    """GPU fast path: ops/csrc/envs.hip::humanoid_step_kernel."""

    HIP_KERNEL = "humanoid_step"
    OBS_DIM = OBS_DIM
    max_episode_steps = 1000

    TORSO_MASS = 40.0
    TORSO_INERTIA = 2.0
    TORSO_Z0 = 1.3
    HIP_SEP = 0.12  # lateral hip offset
    LU = 0.45  # upper leg
    LL = 0.45  # lower leg
    JOINT_INERTIA = 0.12
    JOINT_DAMPING = 2.0
    GEAR = 60.0
    LIMIT = 1.2  # generic soft joint limit (rad)
    KNEE_LO, KNEE_HI = 0.02, 2.0
    LIMIT_K = 80.0
    CONTACT_KN = 1.2e4
    CONTACT_KD = 300.0
    FRICTION = 1.0
    GRAVITY = -9.81
    DT = 0.015
    SUBSTEPS = 4
    CTRL_COST = 0.1
    HEALTHY_REWARD = 5.0
    FORWARD_W = 1.25
    Z_MIN, Z_MAX = 0.8, 2.1

    def __init__(self, num_envs, device="cpu", seed=0, dtype=torch.float32, **kw):
        super().__init__(num_envs, device, seed)
        self.dtype = dtype
        self.observation_space = BoxSpace((OBS_DIM,), -float("inf"), float("inf"))
        self.action_space = BoxSpace((ACT_DIM,), -1.0, 1.0)
        self._init_hip()

    def _hip_action(self, action):
        return action.to(torch.float32).contiguous()

    # ------------------------------------------------------------ state ops

    def _reset_fn(self, n: int) -> State:
        s = torch.zeros(n, STATE_DIM, dtype=self.dtype, device=self.device)
        s[:, POS + 2] = self.TORSO_Z0
        s[:, QUAT] = 1.0
        s[:, QPOS : QPOS + 17] = self.rand(n, 17, lo=-0.03, hi=0.03)
        s[:, QPOS + R_KNEE] += 0.15
        s[:, QPOS + L_KNEE] += 0.15
        s[:, QVEL : QVEL + 17] = self.rand(n, 17, lo=-0.02, hi=0.02)
        return {"s": s}

    def _obs_fn(self, state: State) -> Tensor:
        s = state["s"]
        return torch.cat(
            [
                s[:, POS + 2 : POS + 3],
                s[:, QUAT : QUAT + 4],
                s[:, LINVEL : LINVEL + 3],
                s[:, ANGVEL : ANGVEL + 3],
                s[:, QPOS : QPOS + 17],
                s[:, QVEL : QVEL + 17],
            ],
            dim=-1,
        )

    # -------------------------------------------------------------- physics

    def _foot_offsets(self, qpos: Tensor) -> Tensor:
        """Body-frame foot offsets [B, 2, 3] from leg joint angles."""
        outs = []
        for side, (hx, hy, kn) in enumerate(
            [(R_HIP_X, R_HIP_Y, R_KNEE), (L_HIP_X, L_HIP_Y, L_KNEE)]
        ):
            hip_roll = qpos[:, hx]
            hip_pitch = qpos[:, hy]
            knee = qpos[:, kn]
            lx = self.LU * torch.sin(hip_pitch) + self.LL * torch.sin(hip_pitch + knee)
            lz = -(self.LU * torch.cos(hip_pitch) + self.LL * torch.cos(hip_pitch + knee))
            ly = (self.LU + self.LL) * torch.sin(hip_roll) + (
                self.HIP_SEP if side == 0 else -self.HIP_SEP
            )
            outs.append(torch.stack([lx, ly, lz], dim=-1))
        return torch.stack(outs, dim=1)  # [B, 2, 3]

    def _substep(self, s: Tensor, torque: Tensor, dt: float) -> Tensor:
        pos = s[:, POS : POS + 3]
        quat = s[:, QUAT : QUAT + 4]
        linvel = s[:, LINVEL : LINVEL + 3]
        angvel = s[:, ANGVEL : ANGVEL + 3]
        qpos = s[:, QPOS : QPOS + 17]
        qvel = s[:, QVEL : QVEL + 17]

        # joints: damped inertial with soft limits (knees asymmetric)
        lo = torch.full_like(qpos, -self.LIMIT)
        hi = torch.full_like(qpos, self.LIMIT)
        lo[:, R_KNEE] = self.KNEE_LO
        lo[:, L_KNEE] = self.KNEE_LO
        hi[:, R_KNEE] = self.KNEE_HI
        hi[:, L_KNEE] = self.KNEE_HI
        limit_tau = -self.LIMIT_K * (torch.relu(qpos - hi) - torch.relu(lo - qpos))
        qacc = (self.GEAR * torque - self.JOINT_DAMPING * qvel + limit_tau) / self.JOINT_INERTIA
        qvel = qvel + dt * qacc
        qpos = qpos + dt * qvel

        # feet contacts
        body_off = self._foot_offsets(qpos)  # [B, 2, 3]
        r = quat_rotate(quat.unsqueeze(1).expand(-1, 2, -1), body_off)
        foot_w = pos.unsqueeze(1) + r
        foot_vel = linvel.unsqueeze(1) + torch.cross(
            angvel.unsqueeze(1).expand(-1, 2, -1), r, dim=-1
        )
        pen = (-foot_w[..., 2]).clamp(min=0.0)
        in_contact = pen > 0
        fn = (self.CONTACT_KN * pen - self.CONTACT_KD * foot_vel[..., 2]).clamp(min=0.0)
        fn = torch.where(in_contact, fn, torch.zeros_like(fn))
        ft = -self.FRICTION * fn.unsqueeze(-1) * torch.tanh(4.0 * foot_vel[..., :2])
        contact_f = torch.cat([ft, fn.unsqueeze(-1)], dim=-1)
        total_f = contact_f.sum(dim=1)
        total_tau = torch.cross(r, contact_f, dim=-1).sum(dim=1)

        acc = total_f / self.TORSO_MASS + torch.tensor(
            [0.0, 0.0, self.GRAVITY], dtype=s.dtype, device=s.device
        )
        linvel = linvel + dt * acc
        pos = pos + dt * linvel
        angacc = total_tau / self.TORSO_INERTIA - 0.5 * angvel
        angvel = angvel + dt * angacc
        quat = quat_integrate(quat, angvel, dt)
        return torch.cat([pos, quat, linvel, angvel, qpos, qvel], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        s = state["s"]
        a = action.clamp(-1.0, 1.0).to(s.dtype)
        x_before = s[:, POS].clone()
        dt = self.DT / self.SUBSTEPS
        for _ in range(self.SUBSTEPS):
            s = self._substep(s, a, dt)
        forward_vel = (s[:, POS] - x_before) / self.DT
        ctrl_cost = self.CTRL_COST * (a**2).sum(-1)
        z = s[:, POS + 2]
        healthy = (z > self.Z_MIN) & (z < self.Z_MAX) & torch.isfinite(s).all(dim=-1)
        reward = self.FORWARD_W * forward_vel + self.HEALTHY_REWARD - ctrl_cost
        terminated = ~healthy
        s = torch.where(torch.isfinite(s), s, torch.zeros_like(s))
        return {"s": s}, reward.to(torch.float32), terminated
