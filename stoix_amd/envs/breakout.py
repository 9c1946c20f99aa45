"""Breakout-class pixel game (vectorised; envpool/Atari-suite stand-in).

Fills the role of envpool Atari Breakout in the reference's Sebulba configs
(/root/reference/stoix/configs/env/envpool/breakout.yaml; BASELINE.json
config #4: Sebulba PPO with CPU envs feeding GPU learners). ALE/envpool are
not installable offline; this is an original, fully tensorised paddle/ball/
bricks game with the Atari pixel contract: 84x84 grayscale observations
(channel-last [84, 84, 1] float in [0,1]), 4 actions (NOOP, FIRE, RIGHT,
LEFT), +1 reward per brick, episode ends when the ball is missed or all
bricks are cleared. All state and rendering are batched torch ops, so a
factory of CPU instances drives the Sebulba actor threads exactly like
envpool's batched CPU envs.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace

H, W = 84, 84
BRICK_ROWS, BRICK_COLS = 6, 12
BRICK_W, BRICK_H = W // BRICK_COLS, 3
BRICK_TOP = 12
PADDLE_W, PADDLE_Y = 12, 80
PADDLE_SPEED = 3.0
BALL_SPEED = 1.8


class Breakout(StatefulVecEnv):
    max_episode_steps = 3000

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((H, W, 1), 0.0, 1.0)
        self.action_space = DiscreteSpace(4)
        # precomputed brick masks [rows*cols, H, W]
        masks = torch.zeros(BRICK_ROWS * BRICK_COLS, H, W, device=self.device)
        for r in range(BRICK_ROWS):
            for c in range(BRICK_COLS):
                y0 = BRICK_TOP + r * BRICK_H
                x0 = c * BRICK_W
                masks[r * BRICK_COLS + c, y0 : y0 + BRICK_H - 1, x0 : x0 + BRICK_W - 1] = 0.6
        self._brick_masks = masks

    def _reset_fn(self, n: int) -> State:
        return {
            "paddle_x": torch.full((n,), W / 2.0, device=self.device),
            "ball_x": self.rand(n, lo=W * 0.3, hi=W * 0.7),
            "ball_y": torch.full((n,), 46.0, device=self.device),
            "ball_vx": torch.where(
                self.rand(n) > 0.5,
                torch.full((n,), BALL_SPEED * 0.7, device=self.device),
                torch.full((n,), -BALL_SPEED * 0.7, device=self.device),
            ),
            "ball_vy": torch.full((n,), BALL_SPEED, device=self.device),
            "bricks": torch.ones(n, BRICK_ROWS * BRICK_COLS, device=self.device),
        }

    def _obs_fn(self, state: State) -> Tensor:
        n = state["paddle_x"].shape[0]
        obs = torch.einsum("bk,khw->bhw", state["bricks"], self._brick_masks)
        bidx = torch.arange(n, device=self.device)
        # paddle
        px = state["paddle_x"].long().clamp(PADDLE_W // 2, W - 1 - PADDLE_W // 2)
        for dx in range(-(PADDLE_W // 2), PADDLE_W // 2):
            obs[bidx, PADDLE_Y, (px + dx).clamp(0, W - 1)] = 1.0
            obs[bidx, PADDLE_Y + 1, (px + dx).clamp(0, W - 1)] = 1.0
        # ball (2x2)
        by = state["ball_y"].long().clamp(0, H - 2)
        bx = state["ball_x"].long().clamp(0, W - 2)
        for dy in range(2):
            for dx in range(2):
                obs[bidx, by + dy, bx + dx] = 1.0
        return obs.unsqueeze(-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        a = action.long().clamp(0, 3)
        px = state["paddle_x"] + PADDLE_SPEED * ((a == 2).float() - (a == 3).float())
        px = px.clamp(PADDLE_W / 2, W - PADDLE_W / 2)

        bx = state["ball_x"] + state["ball_vx"]
        by = state["ball_y"] + state["ball_vy"]
        vx = state["ball_vx"].clone()
        vy = state["ball_vy"].clone()

        # side/top walls
        hit_left = bx < 1
        hit_right = bx > W - 2
        vx = torch.where(hit_left | hit_right, -vx, vx)
        bx = bx.clamp(1, W - 2)
        hit_top = by < 1
        vy = torch.where(hit_top, vy.abs(), vy)
        by = torch.where(hit_top, torch.ones_like(by), by)

        # paddle
        on_paddle = (by >= PADDLE_Y - 1) & (by <= PADDLE_Y + 1) & ((bx - px).abs() <= PADDLE_W / 2) & (vy > 0)
        # english: deflect by contact point
        vx = torch.where(on_paddle, vx + 0.4 * (bx - px) / (PADDLE_W / 2), vx)
        vy = torch.where(on_paddle, -vy.abs(), vy)

        # bricks
        in_band = (by >= BRICK_TOP) & (by < BRICK_TOP + BRICK_ROWS * BRICK_H)
        br = ((by - BRICK_TOP) / BRICK_H).long().clamp(0, BRICK_ROWS - 1)
        bc = (bx / BRICK_W).long().clamp(0, BRICK_COLS - 1)
        kidx = br * BRICK_COLS + bc
        n = bx.shape[0]
        bidx = torch.arange(n, device=self.device)
        alive = state["bricks"][bidx, kidx] > 0
        hit_brick = in_band & alive
        bricks = state["bricks"].clone()
        bricks[bidx, kidx] = torch.where(hit_brick, torch.zeros_like(bricks[bidx, kidx]), bricks[bidx, kidx])
        vy = torch.where(hit_brick, -vy, vy)

        reward = hit_brick.float()
        missed = by > H - 2
        cleared = bricks.sum(-1) <= 0
        terminated = missed | cleared
        new_state = {
            "paddle_x": px,
            "ball_x": bx,
            "ball_y": by.clamp(0.0, float(H - 1)),
            "ball_vx": vx.clamp(-2.5, 2.5),
            "ball_vy": vy,
            "bricks": bricks,
        }
        return new_state, reward, terminated
