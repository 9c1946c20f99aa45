"""Connector-class grid-routing game (vectorised; Jumanji-class suite).

Fills the role of Jumanji Connector-v2 in the reference's configs
(/root/reference/stoix/configs/env/jumanji/connector.yaml: 6x6 grid,
2 agents via the multi-agent single-controller wrapper, flattened grid
observation). Jumanji is JAX-only; this is an original, fully
tensorised torch implementation.

Each agent owns a head and a target on a shared grid and extends a wire
one cell per step (actions per agent: noop/up/right/down/left — the
single controller emits a MultiDiscrete([5, 5]) action). A move is legal
onto an empty cell or the agent's own target; heads leave trails behind
them; reaching the target connects the agent (+1 reward at that step).
A small time penalty applies while any agent is unconnected. The episode
terminates when all agents are connected (step limit truncates
otherwise). Cell encoding matches Jumanji (per agent a: trail=3a+1,
head=3a+2, target=3a+3; 0 empty); the observation is the flattened
one-hot grid (the reference applies FlattenObservationWrapper).
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, MultiDiscreteSpace

N = 6  # grid side
A = 2  # agents
NCLASS = 3 * A + 1
# action deltas: noop, up, right, down, left
_DR = [0, -1, 0, 1, 0]
_DC = [0, 0, 1, 0, -1]
TIME_PENALTY = 0.03


class Connector(StatefulVecEnv):
    max_episode_steps = 36

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((N * N * NCLASS,), 0.0, 1.0)
        self.action_space = MultiDiscreteSpace([5] * A)
        self._dr = torch.tensor(_DR, device=self.device)
        self._dc = torch.tensor(_DC, device=self.device)

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        # sample 2A distinct cells per board (heads + targets)
        scores = torch.rand(n, N * N, device=dev, generator=self.gen)
        picks = scores.topk(2 * A, dim=-1).indices  # [n, 2A] distinct cells
        heads = torch.stack([picks[:, :A] // N, picks[:, :A] % N], dim=-1)  # [n,A,2]
        targets = torch.stack([picks[:, A:] // N, picks[:, A:] % N], dim=-1)
        grid = torch.zeros(n, N, N, dtype=torch.long, device=dev)
        bidx = torch.arange(n, device=dev)
        for a in range(A):
            grid[bidx, heads[:, a, 0], heads[:, a, 1]] = 3 * a + 2
            grid[bidx, targets[:, a, 0], targets[:, a, 1]] = 3 * a + 3
        return {
            "grid": grid,
            "heads": heads,
            "targets": targets,
            "connected": torch.zeros(n, A, dtype=torch.bool, device=dev),
        }

    def _obs_fn(self, state: State) -> Tensor:
        g = state["grid"]
        return torch.nn.functional.one_hot(g, NCLASS).float().reshape(g.shape[0], -1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        n = state["grid"].shape[0]
        dev = self.device
        bidx = torch.arange(n, device=dev)
        grid = state["grid"].clone()
        heads = state["heads"].clone()
        connected = state["connected"].clone()
        reward = torch.full((n,), 0.0, device=dev)
        act = action.long().view(n, A).clamp(0, 4)
        # agents move in index order (deterministic collision resolution,
        # like jumanji's sequential agent stepping)
        for a in range(A):
            da = act[:, a]
            nr = (heads[:, a, 0] + self._dr[da]).clamp(0, N - 1)
            nc = (heads[:, a, 1] + self._dc[da]).clamp(0, N - 1)
            moved_cell = grid[bidx, nr, nc]
            own_target = (nr == state["targets"][:, a, 0]) & (nc == state["targets"][:, a, 1])
            legal = (
                (da != 0)
                & ~connected[:, a]
                & ((moved_cell == 0) | (own_target & (moved_cell == 3 * a + 3)))
                & ~((nr == heads[:, a, 0]) & (nc == heads[:, a, 1]))
            )
            # old head becomes trail, new cell becomes head
            hr, hc = heads[:, a, 0], heads[:, a, 1]
            grid[bidx, hr, hc] = torch.where(legal, torch.full_like(hr, 3 * a + 1), grid[bidx, hr, hc])
            grid[bidx, nr, nc] = torch.where(legal, torch.full_like(nr, 3 * a + 2), grid[bidx, nr, nc])
            heads[:, a, 0] = torch.where(legal, nr, hr)
            heads[:, a, 1] = torch.where(legal, nc, hc)
            newly = legal & own_target
            connected[:, a] = connected[:, a] | newly
            reward = reward + newly.float()
        all_conn = connected.all(dim=-1)
        reward = reward - TIME_PENALTY * (~all_conn).float()
        return (
            {"grid": grid, "heads": heads, "targets": state["targets"], "connected": connected},
            reward,
            all_conn,
        )
