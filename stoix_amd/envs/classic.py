"""Classic-control environments, vectorised (gymnax-equivalent suite).

Scenario parity with the reference's gymnax env configs
(/root/reference/stoix/configs/env/gymnax/*.yaml: cartpole, pendulum,
mountain_car, mountain_car_continuous, acrobot). Dynamics follow the
standard published OpenAI-Gym / classic-control equations — implemented from
the textbook definitions, batched over torch tensors.
"""
from __future__ import annotations

import math
from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.hip_env import HipStepMixin
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace


class CartPole(HipStepMixin, StatefulVecEnv):
    """CartPole-v1: discrete 2 actions, 4-dim obs, solved at 500.

    GPU fast path: ops/csrc/envs.hip::cartpole_step_kernel (fused step)."""

    HIP_KERNEL = "cartpole_step"
    OBS_DIM = 4

    max_episode_steps = 500
    solved_return_threshold = 500.0

    GRAVITY = 9.8
    MASSCART = 1.0
    MASSPOLE = 0.1
    LENGTH = 0.5  # half pole length
    FORCE_MAG = 10.0
    TAU = 0.02
    THETA_LIMIT = 12 * 2 * math.pi / 360
    X_LIMIT = 2.4

    def __init__(self, num_envs, device="cpu", seed=0, **kwargs):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((4,), -4.8, 4.8)
        self.action_space = DiscreteSpace(2)
        self._init_hip()

    def _hip_action(self, action):
        return action.long().contiguous()

    def _reset_fn(self, n: int) -> State:
        return {"s": self.rand(n, 4, lo=-0.05, hi=0.05)}

    def _obs_fn(self, state: State) -> Tensor:
        return state["s"].clone()

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        s = state["s"]
        x, x_dot, theta, theta_dot = s.unbind(-1)
        force = torch.where(action.to(torch.float32) > 0.5, self.FORCE_MAG, -self.FORCE_MAG)
        costheta = torch.cos(theta)
        sintheta = torch.sin(theta)
        total_mass = self.MASSCART + self.MASSPOLE
        polemass_length = self.MASSPOLE * self.LENGTH
        temp = (force + polemass_length * theta_dot**2 * sintheta) / total_mass
        thetaacc = (self.GRAVITY * sintheta - costheta * temp) / (
            self.LENGTH * (4.0 / 3.0 - self.MASSPOLE * costheta**2 / total_mass)
        )
        xacc = temp - polemass_length * thetaacc * costheta / total_mass
        x = x + self.TAU * x_dot
        x_dot = x_dot + self.TAU * xacc
        theta = theta + self.TAU * theta_dot
        theta_dot = theta_dot + self.TAU * thetaacc
        new_s = torch.stack([x, x_dot, theta, theta_dot], dim=-1)
        terminated = (x.abs() > self.X_LIMIT) | (theta.abs() > self.THETA_LIMIT)
        reward = torch.ones_like(x)
        return {"s": new_s}, reward, terminated


class Pendulum(StatefulVecEnv):
    """Pendulum-v1: continuous 1-dim torque in [-2, 2], obs [cos, sin, thdot]."""

    max_episode_steps = 200
    MAX_SPEED = 8.0
    MAX_TORQUE = 2.0
    DT = 0.05
    G = 10.0
    M = 1.0
    L = 1.0

    def __init__(self, num_envs, device="cpu", seed=0, **kwargs):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((3,), -8.0, 8.0)
        self.action_space = BoxSpace((1,), -self.MAX_TORQUE, self.MAX_TORQUE)

    def _reset_fn(self, n: int) -> State:
        th = self.rand(n, lo=-math.pi, hi=math.pi)
        thdot = self.rand(n, lo=-1.0, hi=1.0)
        return {"th": th, "thdot": thdot}

    def _obs_fn(self, state: State) -> Tensor:
        return torch.stack([torch.cos(state["th"]), torch.sin(state["th"]), state["thdot"]], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        th, thdot = state["th"], state["thdot"]
        u = action.reshape(th.shape).clamp(-self.MAX_TORQUE, self.MAX_TORQUE)
        norm_th = ((th + math.pi) % (2 * math.pi)) - math.pi
        cost = norm_th**2 + 0.1 * thdot**2 + 0.001 * u**2
        newthdot = thdot + (3 * self.G / (2 * self.L) * torch.sin(th) + 3.0 / (self.M * self.L**2) * u) * self.DT
        newthdot = newthdot.clamp(-self.MAX_SPEED, self.MAX_SPEED)
        newth = th + newthdot * self.DT
        terminated = torch.zeros_like(th, dtype=torch.bool)
        return {"th": newth, "thdot": newthdot}, -cost, terminated


class MountainCar(StatefulVecEnv):
    """MountainCar-v0 (discrete 3 actions)."""

    max_episode_steps = 200

    def __init__(self, num_envs, device="cpu", seed=0, continuous: bool = False, **kwargs):
        super().__init__(num_envs, device, seed)
        self.continuous = continuous
        self.observation_space = BoxSpace((2,), -1.2, 0.6)
        self.action_space = BoxSpace((1,), -1.0, 1.0) if continuous else DiscreteSpace(3)
        if continuous:
            self.max_episode_steps = 999

    def _reset_fn(self, n: int) -> State:
        pos = self.rand(n, lo=-0.6, hi=-0.4)
        vel = torch.zeros(n, device=self.device)
        return {"pos": pos, "vel": vel}

    def _obs_fn(self, state: State) -> Tensor:
        return torch.stack([state["pos"], state["vel"]], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        pos, vel = state["pos"], state["vel"]
        if self.continuous:
            force = action.reshape(pos.shape).clamp(-1.0, 1.0)
            vel = vel + force * 0.0015 - 0.0025 * torch.cos(3 * pos)
            vel = vel.clamp(-0.07, 0.07)
            pos = (pos + vel).clamp(-1.2, 0.6)
            vel = torch.where((pos <= -1.2) & (vel < 0), torch.zeros_like(vel), vel)
            terminated = (pos >= 0.45) & (vel >= 0.0)
            reward = torch.where(terminated, 100.0, 0.0) - 0.1 * force**2
        else:
            force = action.to(torch.float32) - 1.0
            vel = vel + force * 0.001 - 0.0025 * torch.cos(3 * pos)
            vel = vel.clamp(-0.07, 0.07)
            pos = (pos + vel).clamp(-1.2, 0.6)
            vel = torch.where((pos <= -1.2) & (vel < 0), torch.zeros_like(vel), vel)
            terminated = (pos >= 0.5) & (vel >= 0.0)
            reward = -torch.ones_like(pos)
        return {"pos": pos, "vel": vel}, reward, terminated


class Acrobot(StatefulVecEnv):
    """Acrobot-v1 (discrete 3 actions, 6-dim obs)."""

    max_episode_steps = 500
    DT = 0.2
    LINK_LENGTH_1 = 1.0
    LINK_MASS_1 = 1.0
    LINK_MASS_2 = 1.0
    LINK_COM_POS_1 = 0.5
    LINK_COM_POS_2 = 0.5
    LINK_MOI = 1.0
    MAX_VEL_1 = 4 * math.pi
    MAX_VEL_2 = 9 * math.pi

    def __init__(self, num_envs, device="cpu", seed=0, **kwargs):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((6,), -1.0, 1.0)
        self.action_space = DiscreteSpace(3)

    def _reset_fn(self, n: int) -> State:
        return {"s": self.rand(n, 4, lo=-0.1, hi=0.1)}

    def _obs_fn(self, state: State) -> Tensor:
        th1, th2, dth1, dth2 = state["s"].unbind(-1)
        return torch.stack(
            [torch.cos(th1), torch.sin(th1), torch.cos(th2), torch.sin(th2), dth1, dth2], dim=-1
        )

    def _dsdt(self, s: Tensor, torque: Tensor) -> Tensor:
        m1, m2 = self.LINK_MASS_1, self.LINK_MASS_2
        l1 = self.LINK_LENGTH_1
        lc1, lc2 = self.LINK_COM_POS_1, self.LINK_COM_POS_2
        I1 = I2 = self.LINK_MOI
        g = 9.8
        th1, th2, dth1, dth2 = s.unbind(-1)
        d1 = m1 * lc1**2 + m2 * (l1**2 + lc2**2 + 2 * l1 * lc2 * torch.cos(th2)) + I1 + I2
        d2 = m2 * (lc2**2 + l1 * lc2 * torch.cos(th2)) + I2
        phi2 = m2 * lc2 * g * torch.cos(th1 + th2 - math.pi / 2.0)
        phi1 = (
            -m2 * l1 * lc2 * dth2**2 * torch.sin(th2)
            - 2 * m2 * l1 * lc2 * dth2 * dth1 * torch.sin(th2)
            + (m1 * lc1 + m2 * l1) * g * torch.cos(th1 - math.pi / 2)
            + phi2
        )
        ddth2 = (torque + d2 / d1 * phi1 - m2 * l1 * lc2 * dth1**2 * torch.sin(th2) - phi2) / (
            m2 * lc2**2 + I2 - d2**2 / d1
        )
        ddth1 = -(d2 * ddth2 + phi1) / d1
        return torch.stack([dth1, dth2, ddth1, ddth2], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        s = state["s"]
        torque = action.to(torch.float32) - 1.0
        # RK4 integration over dt=0.2 (standard acrobot)
        k1 = self._dsdt(s, torque)
        k2 = self._dsdt(s + 0.5 * self.DT * k1, torque)
        k3 = self._dsdt(s + 0.5 * self.DT * k2, torque)
        k4 = self._dsdt(s + self.DT * k3, torque)
        ns = s + self.DT / 6.0 * (k1 + 2 * k2 + 2 * k3 + k4)
        th1 = ((ns[..., 0] + math.pi) % (2 * math.pi)) - math.pi
        th2 = ((ns[..., 1] + math.pi) % (2 * math.pi)) - math.pi
        dth1 = ns[..., 2].clamp(-self.MAX_VEL_1, self.MAX_VEL_1)
        dth2 = ns[..., 3].clamp(-self.MAX_VEL_2, self.MAX_VEL_2)
        ns = torch.stack([th1, th2, dth1, dth2], dim=-1)
        terminated = (-torch.cos(th1) - torch.cos(th2 + th1)) > 1.0
        reward = torch.where(terminated, 0.0, -1.0)
        return {"s": ns}, reward, terminated


class CartPoleSwingUp(CartPole):
    """dm_control-style cartpole swingup (mujoco_playground /
    dm_control capability class): the pole STARTS HANGING (theta ~ pi),
    the action is a continuous force, and the reward is the smooth
    upright/centred product dm_control uses — no termination, fixed
    horizon. Same physics core as CartPole."""

    max_episode_steps = 500
    solved_return_threshold = 350.0

    def __init__(self, num_envs, device="cpu", seed=0, **kwargs):
        super().__init__(num_envs, device, seed, **kwargs)
        from stoix_amd.envs.spaces import BoxSpace

        self.observation_space = BoxSpace((5,), -5.0, 5.0)
        self.action_space = BoxSpace((1,), -1.0, 1.0)
        self._hip = None  # torch path (the HIP kernel is the discrete game)

    def _reset_fn(self, n: int):
        s = self.rand(n, 4, lo=-0.05, hi=0.05)
        s[:, 2] = s[:, 2] + math.pi  # hanging down
        return {"s": s}

    def _obs_fn(self, state):
        s = state["s"]
        # dm_control-style: [x, x_dot, cos(theta), sin(theta), theta_dot]
        return torch.stack(
            [s[:, 0], s[:, 1], torch.cos(s[:, 2]), torch.sin(s[:, 2]), s[:, 3]],
            dim=-1,
        )

    def _step_fn(self, state, action):
        s = state["s"]
        x, x_dot, th, th_dot = s.unbind(-1)
        force = action.reshape(-1).clamp(-1.0, 1.0) * self.FORCE_MAG
        costheta = torch.cos(th)
        sintheta = torch.sin(th)
        total_mass = self.MASSCART + self.MASSPOLE
        pml = self.MASSPOLE * self.LENGTH
        temp = (force + pml * th_dot * th_dot * sintheta) / total_mass
        thacc = (self.GRAVITY * sintheta - costheta * temp) / (
            self.LENGTH * (4.0 / 3.0 - self.MASSPOLE * costheta * costheta / total_mass)
        )
        xacc = temp - pml * thacc * costheta / total_mass
        x = x + self.TAU * x_dot
        x_dot = x_dot + self.TAU * xacc
        th = th + self.TAU * th_dot
        th_dot = th_dot + self.TAU * thacc
        # keep the cart on the track (elastic walls, dm_control style slide limit)
        hit = x.abs() > self.X_LIMIT
        x = x.clamp(-self.X_LIMIT, self.X_LIMIT)
        x_dot = torch.where(hit, torch.zeros_like(x_dot), x_dot)
        new_s = torch.stack([x, x_dot, th, th_dot], dim=-1)
        # smooth dm_control reward: upright * centred in [0, 1]
        upright = (1.0 + torch.cos(th)) / 2.0
        centred = 1.0 - (x.abs() / self.X_LIMIT) ** 2 * 0.5
        reward = upright * centred
        terminated = torch.zeros_like(hit)  # horizon-only
        return {"s": new_s}, reward, terminated


class CartPoleBalance(CartPoleSwingUp):
    """dm_control cartpole BALANCE task (reference
    mjc_playground/dm_control/cartpole_balance.yaml): same continuous-force
    cartpole and smooth upright/centred reward as swing-up, but the pole
    STARTS UPRIGHT and must be kept there."""

    solved_return_threshold = 450.0

    def _reset_fn(self, n: int):
        s = self.rand(n, 4, lo=-0.05, hi=0.05)  # upright start
        return {"s": s}
