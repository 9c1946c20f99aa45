"""Ant-class quadruped locomotion environment (vectorised rigid-body physics).

Fills the role of Brax Ant in the reference's benchmark configs
(/root/reference/stoix/configs/env/brax/ant.yaml; BASELINE.json north-star
config: Anakin PPO, 4096 envs/GPU). Brax itself is JAX-only and cannot be
ported; this is an original, self-contained articulated-ant simulation with
the same interface contract: 27-dim observation, 8-dim torque action in
[-1, 1], forward-progress reward with control cost and healthy-range
termination.

Physics model (semi-implicit Euler, 4 substeps of 12.5 ms per control step):
  * torso: free rigid body (position, quaternion, linear + angular velocity)
  * 4 legs x 2 hinge joints (hip yaw in the torso plane, knee pitch), each a
    damped inertial joint driven by the action torque, with soft angle-limit
    springs
  * feet: point contacts against the ground plane via a penalty spring-damper
    normal force + Coulomb-capped tangential friction, reacting on the torso
    (force + torque at the foot moment arm) — the standard penalty-contact
    formulation Brax's spring pipeline also uses
  * reward = forward x-velocity + healthy bonus - ctrl cost - contact cost
  * terminated when torso z leaves [0.2, 1.0] (unhealthy), like MuJoCo Ant

State layout is one flat [B, 29] tensor (struct-of-arrays within one row:
pos 3 | quat 4 | linvel 3 | angvel 3 | qpos 8 | qvel 8) so that the CDNA4 HIP
kernel (stoix_amd/ops/hip/env_ant.hip) holds one env per lane with the row in
registers.
"""
from __future__ import annotations

import math
from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.hip_env import HipStepMixin
from stoix_amd.envs.spaces import BoxSpace

# state slice offsets
POS, QUAT, LINVEL, ANGVEL, QPOS, QVEL = 0, 3, 7, 10, 13, 21
STATE_DIM = 29
OBS_DIM = 27
ACT_DIM = 8


def quat_rotate(q: Tensor, v: Tensor) -> Tensor:
    """Rotate vectors v [..., 3] by unit quaternions q [..., 4] (w, x, y, z)."""
    w, x, y, z = q.unbind(-1)
    qv = torch.stack([x, y, z], dim=-1)
    uv = torch.cross(qv, v, dim=-1)
    uuv = torch.cross(qv, uv, dim=-1)
    return v + 2.0 * (w.unsqueeze(-1) * uv + uuv)


def quat_integrate(q: Tensor, omega: Tensor, dt: float) -> Tensor:
    """Integrate quaternion by world-frame angular velocity omega over dt."""
    w, x, y, z = q.unbind(-1)
    ox, oy, oz = omega.unbind(-1)
    dw = 0.5 * (-x * ox - y * oy - z * oz)
    dx = 0.5 * (w * ox + y * oz - z * oy)
    dy = 0.5 * (w * oy + z * ox - x * oz)
    dz = 0.5 * (w * oz + x * oy - y * ox)
    nq = torch.stack([w + dt * dw, x + dt * dx, y + dt * dy, z + dt * dz], dim=-1)
    return nq / nq.norm(dim=-1, keepdim=True).clamp(min=1e-8)


class Ant(HipStepMixin, StatefulVecEnv):
    """GPU fast path: ops/csrc/envs.hip::ant_step_kernel (fused step)."""

    HIP_KERNEL = "ant_step"
    OBS_DIM = OBS_DIM
    max_episode_steps = 1000

    # body parameters
    TORSO_MASS = 10.0
    TORSO_INERTIA = 0.4  # isotropic
    TORSO_Z0 = 0.55
    HIP_RADIUS = 0.2  # attachment distance from torso centre
    L1 = 0.2  # upper leg length
    L2 = 0.4  # lower leg length
    JOINT_INERTIA = 0.08
    JOINT_DAMPING = 1.2
    GEAR = 15.0
    HIP_LIMIT = 0.6  # rad, soft limit around nominal
    KNEE_LO, KNEE_HI = 0.4, 1.4  # knee angle range (downward bend)
    LIMIT_K = 40.0
    # contact
    CONTACT_KN = 2.0e3
    CONTACT_KD = 40.0
    FRICTION = 1.0
    GRAVITY = -9.81
    DT = 0.05
    SUBSTEPS = 4
    # reward
    CTRL_COST = 0.5
    CONTACT_COST = 5e-4
    HEALTHY_REWARD = 1.0
    Z_MIN, Z_MAX = 0.2, 1.0

    def __init__(self, num_envs, device="cpu", seed=0, dtype=torch.float32, **kw):
        super().__init__(num_envs, device, seed)
        self.dtype = dtype
        self.observation_space = BoxSpace((OBS_DIM,), -float("inf"), float("inf"))
        self.action_space = BoxSpace((ACT_DIM,), -1.0, 1.0)
        # hip attachment directions (diagonals, in torso frame)
        ang = torch.tensor([math.pi / 4 + i * math.pi / 2 for i in range(4)], dtype=dtype)
        self._hip_dir = torch.stack([torch.cos(ang), torch.sin(ang), torch.zeros(4)], dim=-1).to(
            self.device
        )  # [4, 3]
        self._init_hip()

    def _hip_action(self, action):
        return action.to(torch.float32).contiguous()

    # ------------------------------------------------------------ state ops

    def _reset_fn(self, n: int) -> State:
        s = torch.zeros(n, STATE_DIM, dtype=self.dtype, device=self.device)
        s[:, POS + 2] = self.TORSO_Z0
        s[:, QUAT] = 1.0  # identity quaternion
        # small random perturbations on joints and pose, like MuJoCo reset noise
        s[:, QPOS : QPOS + 8] = self.rand(n, 8, lo=-0.1, hi=0.1)
        s[:, QPOS + 4 : QPOS + 8] += 0.9  # knees start bent inside limits
        s[:, QVEL : QVEL + 8] = self.rand(n, 8, lo=-0.05, hi=0.05)
        s[:, LINVEL : LINVEL + 3] = self.rand(n, 3, lo=-0.05, hi=0.05)
        return {"s": s}

    def _obs_fn(self, state: State) -> Tensor:
        s = state["s"]
        return torch.cat(
            [
                s[:, POS + 2 : POS + 3],  # z height (1)
                s[:, QUAT : QUAT + 4],  # orientation (4)
                s[:, QPOS : QPOS + 8],  # joint angles (8)
                s[:, LINVEL : LINVEL + 3],  # linear velocity (3)
                s[:, ANGVEL : ANGVEL + 3],  # angular velocity (3)
                s[:, QVEL : QVEL + 8],  # joint velocities (8)
            ],
            dim=-1,
        )

    # -------------------------------------------------------------- physics

    def _foot_positions(self, s: Tensor) -> Tuple[Tensor, Tensor]:
        """World foot positions [B, 4, 3] and body-frame offsets [B, 4, 3]."""
        B = s.shape[0]
        quat = s[:, QUAT : QUAT + 4]
        pos = s[:, POS : POS + 3]
        hip = s[:, QPOS : QPOS + 4]  # [B, 4] yaw about torso z
        knee = s[:, QPOS + 4 : QPOS + 8]  # [B, 4] downward pitch
        hd = self._hip_dir.unsqueeze(0)  # [1, 4, 3]
        base_ang = torch.atan2(hd[..., 1], hd[..., 0])  # [1, 4]
        leg_ang = base_ang + hip  # [B, 4]
        cos_a, sin_a = torch.cos(leg_ang), torch.sin(leg_ang)
        dir_xy = torch.stack([cos_a, sin_a, torch.zeros_like(cos_a)], dim=-1)  # [B,4,3]
        attach = hd * self.HIP_RADIUS  # [1, 4, 3]
        ck, sk = torch.cos(knee), torch.sin(knee)
        # upper leg horizontal; lower leg pitched down by knee angle
        upper = dir_xy * self.L1
        lower = dir_xy * (self.L2 * ck).unsqueeze(-1) + torch.stack(
            [torch.zeros_like(sk), torch.zeros_like(sk), -self.L2 * sk], dim=-1
        )
        body_off = attach + upper + lower  # [B, 4, 3]
        world = pos.unsqueeze(1) + quat_rotate(quat.unsqueeze(1).expand(-1, 4, -1), body_off)
        return world, body_off

    def _substep(self, s: Tensor, torque: Tensor, dt: float) -> Tuple[Tensor, Tensor]:
        """One physics substep; returns (new_state, contact_force_magnitude)."""
        pos = s[:, POS : POS + 3]
        quat = s[:, QUAT : QUAT + 4]
        linvel = s[:, LINVEL : LINVEL + 3]
        angvel = s[:, ANGVEL : ANGVEL + 3]
        qpos = s[:, QPOS : QPOS + 8]
        qvel = s[:, QVEL : QVEL + 8]

        # --- joints: damped inertial, soft limits
        hip, knee = qpos[:, :4], qpos[:, 4:]
        limit_tau_hip = -self.LIMIT_K * (
            torch.relu(hip - self.HIP_LIMIT) - torch.relu(-self.HIP_LIMIT - hip)
        )
        limit_tau_knee = -self.LIMIT_K * (
            torch.relu(knee - self.KNEE_HI) - torch.relu(self.KNEE_LO - knee)
        )
        limit_tau = torch.cat([limit_tau_hip, limit_tau_knee], dim=-1)
        qacc = (self.GEAR * torque - self.JOINT_DAMPING * qvel + limit_tau) / self.JOINT_INERTIA
        qvel = qvel + dt * qacc
        qpos = qpos + dt * qvel

        # --- contacts at feet
        foot_w, body_off = self._foot_positions(s)
        r = quat_rotate(quat.unsqueeze(1).expand(-1, 4, -1), body_off)  # world moment arm
        foot_vel = linvel.unsqueeze(1) + torch.cross(
            angvel.unsqueeze(1).expand(-1, 4, -1), r, dim=-1
        )
        pen = (-foot_w[..., 2]).clamp(min=0.0)  # penetration depth
        in_contact = pen > 0
        fn = (self.CONTACT_KN * pen - self.CONTACT_KD * foot_vel[..., 2]).clamp(min=0.0)
        fn = torch.where(in_contact, fn, torch.zeros_like(fn))
        ft = -self.FRICTION * fn.unsqueeze(-1) * torch.tanh(4.0 * foot_vel[..., :2])
        contact_f = torch.cat([ft, fn.unsqueeze(-1)], dim=-1)  # [B, 4, 3]
        total_f = contact_f.sum(dim=1)
        total_tau = torch.cross(r, contact_f, dim=-1).sum(dim=1)

        # --- torso integration (semi-implicit Euler)
        acc = total_f / self.TORSO_MASS
        acc = acc + torch.tensor([0.0, 0.0, self.GRAVITY], dtype=s.dtype, device=s.device)
        linvel = linvel + dt * acc
        pos = pos + dt * linvel
        angacc = total_tau / self.TORSO_INERTIA - 0.2 * angvel  # small rotational damping
        angvel = angvel + dt * angacc
        quat = quat_integrate(quat, angvel, dt)

        # keep torso itself off the floor (soft)
        torso_pen = (0.12 - pos[:, 2]).clamp(min=0.0)
        linvel = torch.cat(
            [linvel[:, :2], linvel[:, 2:3] + dt * self.CONTACT_KN / self.TORSO_MASS * torso_pen.unsqueeze(-1)],
            dim=-1,
        )

        ns = torch.cat([pos, quat, linvel, angvel, qpos, qvel], dim=-1)
        return ns, contact_f.abs().sum(dim=(1, 2))

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        s = state["s"]
        a = action.clamp(-1.0, 1.0).to(s.dtype)
        x_before = s[:, POS].clone()
        dt = self.DT / self.SUBSTEPS
        contact_mag = torch.zeros(s.shape[0], dtype=s.dtype, device=s.device)
        for _ in range(self.SUBSTEPS):
            s, cf = self._substep(s, a, dt)
            contact_mag = contact_mag + cf
        x_after = s[:, POS]
        forward_vel = (x_after - x_before) / self.DT
        ctrl_cost = self.CTRL_COST * (a**2).sum(-1)
        contact_cost = self.CONTACT_COST * (contact_mag / self.SUBSTEPS) ** 2
        z = s[:, POS + 2]
        healthy = (z > self.Z_MIN) & (z < self.Z_MAX) & torch.isfinite(s).all(dim=-1)
        reward = forward_vel + self.HEALTHY_REWARD - ctrl_cost - contact_cost
        terminated = ~healthy
        # scrub non-finite states so autoreset replaces them cleanly
        s = torch.where(torch.isfinite(s), s, torch.zeros_like(s))
        return {"s": s}, reward.to(torch.float32), terminated
