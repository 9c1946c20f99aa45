"""Game2048 (vectorised; Jumanji-class suite).

Fills the role of Jumanji Game2048-v1 in the reference's configs
(/root/reference/stoix/configs/env/jumanji/game_2048.yaml).
Jumanji is JAX-only; this is an original, fully tensorised torch
implementation: every slide/merge/spawn is batched over B boards with
fixed-iteration tensor ops (no per-env Python), so it runs on CPU and as
pure device tensor work on the GPU.

Board 4x4 of tile EXPONENTS (cell k represents tile 2^k, 0 = empty).
Action in {0: up, 1: right, 2: down, 3: left}. A move slides all tiles,
merges equal neighbours once per pair (2^k + 2^k -> 2^{k+1}), earns
reward = sum of merged tile values, then spawns one new tile (2 with
p=0.9, 4 with p=0.1) on a uniform random empty cell. A move that does not
change the board spawns nothing and earns 0 (Jumanji's illegal-action
no-op). Terminates when no action can change the board.

Observation [4, 4, 16] float32: one-hot of the cell exponent clamped to
15 (up to the 32768 tile) — channel 0 marks empty cells, matching the
one-hot board encoding Jumanji's 2048 networks consume.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace

N = 4
CH = 16


class Game2048(StatefulVecEnv):
    max_episode_steps = 10000
    capture_safe = True  # Gumbel-max spawns, no host-synced branches

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((N, N, CH), 0.0, 1.0)
        self.action_space = DiscreteSpace(4)

    # ------------------------------------------------------------ mechanics

    def _slide_left(self, board: Tensor) -> Tuple[Tensor, Tensor]:
        """Slide+merge every row leftward. board [B, 4, 4] int32 exponents.
        Returns (new_board, merge_reward [B])."""

        def compress(b: Tensor) -> Tensor:
            # 3 bubble passes push zeros right while preserving order
            for _ in range(3):
                for i in range(N - 1):
                    left, right = b[..., i], b[..., i + 1]
                    move = (left == 0) & (right != 0)
                    b = b.clone()
                    b[..., i] = torch.where(move, right, left)
                    b[..., i + 1] = torch.where(move, left, right)
            return b

        b = compress(board)
        reward = torch.zeros(board.shape[0], dtype=torch.float32, device=board.device)
        # single merge sweep left-to-right; a merged right cell becomes 0 so
        # it cannot chain-merge ([2,2,2,2] -> [3,0,3,0], not [4,...])
        for i in range(N - 1):
            left, right = b[..., i], b[..., i + 1]
            merge = (left != 0) & (left == right)
            b = b.clone()
            b[..., i] = torch.where(merge, left + 1, left)
            b[..., i + 1] = torch.where(merge, torch.zeros_like(right), right)
            reward = reward + (merge.float() * torch.pow(2.0, (left + 1).float())).sum(dim=-1)
        return compress(b), reward

    def _apply_move(self, board: Tensor, action: Tensor) -> Tuple[Tensor, Tensor]:
        """Apply per-board actions by orienting every board so its action
        becomes 'left', sliding once, and orienting back."""
        outs, rews = [], []
        for a in range(4):
            ob = self._orient(board, a)
            nb, r = self._slide_left(ob)
            outs.append(self._orient_back(nb, a))
            rews.append(r)
        stacked = torch.stack(outs, dim=0)  # [4, B, 4, 4]
        rstacked = torch.stack(rews, dim=0)  # [4, B]
        bidx = torch.arange(board.shape[0], device=board.device)
        return stacked[action, bidx], rstacked[action, bidx]

    @staticmethod
    def _orient(board: Tensor, action: int) -> Tensor:
        if action == 3:  # left
            return board
        if action == 1:  # right
            return board.flip(-1)
        if action == 0:  # up: columns become rows
            return board.transpose(-1, -2)
        return board.transpose(-1, -2).flip(-1)  # down

    @staticmethod
    def _orient_back(board: Tensor, action: int) -> Tensor:
        if action == 3:
            return board
        if action == 1:
            return board.flip(-1)
        if action == 0:
            return board.transpose(-1, -2)
        return board.flip(-1).transpose(-1, -2)

    def _spawn(self, board: Tensor, mask: Tensor) -> Tensor:
        """Spawn one tile (90% a 2, 10% a 4) on a uniform empty cell of each
        board where mask is True."""
        B = board.shape[0]
        flat = board.reshape(B, N * N)
        # Gumbel-max over the empty mask: uniform over empty cells and
        # capture-legal (torch.multinomial is not graph-capturable)
        empty = flat == 0
        u = torch.rand(B, N * N, device=self.device, generator=self.gen)
        gum = -torch.log(-torch.log(u.clamp(min=1e-12)).clamp(min=1e-12))
        idx = torch.where(empty, gum, torch.full_like(gum, -torch.inf)).argmax(dim=-1)
        val = torch.where(
            torch.rand(B, device=self.device, generator=self.gen) < 0.9, 1, 2
        ).to(board.dtype)
        has_empty = (flat == 0).any(dim=-1)
        do = mask & has_empty
        upd = flat.clone()
        bidx = torch.arange(B, device=self.device)
        upd[bidx, idx] = torch.where(do, val, flat[bidx, idx])
        return upd.reshape(B, N, N)

    def _any_move_possible(self, board: Tensor) -> Tensor:
        """[B] bool: some action changes the board (empty cell, or an equal
        adjacent pair in a row or column)."""
        has_empty = (board == 0).any(dim=(-1, -2))
        row_pair = (board[..., :, :-1] == board[..., :, 1:]) & (board[..., :, :-1] != 0)
        col_pair = (board[..., :-1, :] == board[..., 1:, :]) & (board[..., :-1, :] != 0)
        return has_empty | row_pair.any(dim=(-1, -2)) | col_pair.any(dim=(-1, -2))

    # ------------------------------------------------------------ state ops

    def _reset_fn(self, n: int) -> State:
        board = torch.zeros(n, N, N, dtype=torch.int32, device=self.device)
        all_true = torch.ones(n, dtype=torch.bool, device=self.device)
        board = self._spawn(board, all_true)
        board = self._spawn(board, all_true)
        return {"board": board}

    def _obs_fn(self, state: State) -> Tensor:
        board = state["board"].long().clamp(max=CH - 1)
        return torch.nn.functional.one_hot(board, CH).float()

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        board = state["board"]
        new_board, reward = self._apply_move(board, action.long())
        changed = (new_board != board).any(dim=(-1, -2))
        new_board = self._spawn(new_board, changed)
        terminated = ~self._any_move_possible(new_board)
        return {"board": new_board}, reward, terminated
