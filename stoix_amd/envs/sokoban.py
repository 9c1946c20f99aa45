"""Sokoban-class box-pushing puzzle (vectorised; Jumanji-class suite).

Fills the role of Jumanji Sokoban-v0 in the reference's configs
(/root/reference/stoix/configs/env/jumanji/sokoban.yaml).
Jumanji's Sokoban loads the boxoban level DATASET (a network download that
does not exist offline); this implementation generates solvable levels
procedurally by REVERSE WALKS instead: boxes start ON their targets and a
simulated agent performs K random reverse-pulls (the time-reversal of a
push), so every generated level is solvable by construction in <= K pushes.

Grid 10x10 (walled border, 8x8 interior), 4 boxes, 4 targets, one agent.
Actions: up/right/down/left. A move into a box pushes it if the cell
behind is free. Rewards (jumanji sokoban shaping): -0.1 per step,
+1 when a box lands on a target / -1 when pushed off, +10 on solving;
terminates when every box sits on a target. Observation is the one-hot
grid [10, 10, 7] (wall, empty, target, box, box-on-target, agent,
agent-on-target).
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace

N = 10
NBOX = 4
GEN_STEPS = 24
# obs channels
WALL, EMPTY, TARGET, BOX, BOX_ON_T, AGENT, AGENT_ON_T = range(7)
_DR = [-1, 0, 1, 0]
_DC = [0, 1, 0, -1]


class Sokoban(StatefulVecEnv):
    max_episode_steps = 120

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((N, N, 7), 0.0, 1.0)
        self.action_space = DiscreteSpace(4)
        self._dr = torch.tensor(_DR, device=self.device)
        self._dc = torch.tensor(_DC, device=self.device)

    # -------------------------------------------------------- level builder

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        # distinct interior cells for the NBOX targets (boxes start on them)
        scores = torch.rand(n, 8 * 8, device=dev, generator=self.gen)
        picks = scores.topk(NBOX + 1, dim=-1).indices
        tr = picks // 8 + 1
        tc = picks % 8 + 1
        targets = torch.stack([tr[:, :NBOX], tc[:, :NBOX]], dim=-1)  # [n,4,2]
        boxes = targets.clone()
        agent = torch.stack([tr[:, NBOX], tc[:, NBOX]], dim=-1)  # [n,2]

        # reverse walk: the agent steps randomly; stepping AWAY from an
        # adjacent box may pull it along (reverse of a push)
        for k in range(GEN_STEPS):
            d = self.randint(4, n)
            dr, dc = self._dr[d], self._dc[d]
            nr = (agent[:, 0] + dr).clamp(1, N - 2)
            nc = (agent[:, 1] + dc).clamp(1, N - 2)
            onto_box = ((boxes[:, :, 0] == nr.unsqueeze(1)) & (boxes[:, :, 1] == nc.unsqueeze(1))).any(-1)
            ok = ~onto_box & ((nr != agent[:, 0]) | (nc != agent[:, 1]))
            # pull: box at agent - d (behind) follows into the agent's old cell
            br = agent[:, 0] - dr
            bc = agent[:, 1] - dc
            is_behind = (boxes[:, :, 0] == br.unsqueeze(1)) & (boxes[:, :, 1] == bc.unsqueeze(1))
            do_pull = ok & is_behind.any(-1) & (self.rand(n) < 0.5)
            pull_mask = is_behind & do_pull.unsqueeze(1)
            boxes[:, :, 0] = torch.where(pull_mask, agent[:, 0].unsqueeze(1), boxes[:, :, 0])
            boxes[:, :, 1] = torch.where(pull_mask, agent[:, 1].unsqueeze(1), boxes[:, :, 1])
            agent[:, 0] = torch.where(ok, nr, agent[:, 0])
            agent[:, 1] = torch.where(ok, nc, agent[:, 1])
        return {"agent": agent, "boxes": boxes, "targets": targets}

    # ------------------------------------------------------------ rendering

    def _on_target(self, state: State) -> Tensor:
        b, t = state["boxes"], state["targets"]
        return (
            (b[:, :, None, 0] == t[:, None, :, 0]) & (b[:, :, None, 1] == t[:, None, :, 1])
        ).any(-1)  # [n, NBOX]

    def _obs_fn(self, state: State) -> Tensor:
        n = state["agent"].shape[0]
        dev = self.device
        obs = torch.zeros(n, N, N, 7, device=dev)
        obs[:, :, :, EMPTY] = 1.0
        obs[:, 0, :, :] = 0.0
        obs[:, -1, :, :] = 0.0
        obs[:, :, 0, :] = 0.0
        obs[:, :, -1, :] = 0.0
        obs[:, 0, :, WALL] = 1.0
        obs[:, -1, :, WALL] = 1.0
        obs[:, :, 0, WALL] = 1.0
        obs[:, :, -1, WALL] = 1.0
        bidx = torch.arange(n, device=dev)
        t = state["targets"]
        for k in range(NBOX):
            obs[bidx, t[:, k, 0], t[:, k, 1], EMPTY] = 0.0
            obs[bidx, t[:, k, 0], t[:, k, 1], TARGET] = 1.0
        on_t = self._on_target(state)
        b = state["boxes"]
        for k in range(NBOX):
            ch = torch.where(on_t[:, k], BOX_ON_T, BOX)
            obs[bidx, b[:, k, 0], b[:, k, 1], EMPTY] = 0.0
            obs[bidx, b[:, k, 0], b[:, k, 1], TARGET] = 0.0
            obs[bidx, b[:, k, 0], b[:, k, 1], ch] = 1.0
        a = state["agent"]
        a_on_t = ((a[:, None, 0] == t[:, :, 0]) & (a[:, None, 1] == t[:, :, 1])).any(-1)
        ach = torch.where(a_on_t, AGENT_ON_T, AGENT)
        obs[bidx, a[:, 0], a[:, 1], EMPTY] = 0.0
        obs[bidx, a[:, 0], a[:, 1], TARGET] = 0.0
        obs[bidx, a[:, 0], a[:, 1], ach] = 1.0
        return obs

    # ------------------------------------------------------------- stepping

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        n = state["agent"].shape[0]
        agent = state["agent"].clone()
        boxes = state["boxes"].clone()
        on_before = self._on_target(state).sum(-1)

        d = action.long().clamp(0, 3)
        dr, dc = self._dr[d], self._dc[d]
        nr = agent[:, 0] + dr
        nc = agent[:, 1] + dc
        in_bounds = (nr >= 1) & (nr <= N - 2) & (nc >= 1) & (nc <= N - 2)
        at_new = (boxes[:, :, 0] == nr.unsqueeze(1)) & (boxes[:, :, 1] == nc.unsqueeze(1))
        pushing = at_new.any(-1)
        # push destination
        pr = nr + dr
        pc = nc + dc
        dest_free = (
            (pr >= 1) & (pr <= N - 2) & (pc >= 1) & (pc <= N - 2)
            & ~((boxes[:, :, 0] == pr.unsqueeze(1)) & (boxes[:, :, 1] == pc.unsqueeze(1))).any(-1)
        )
        can_push = pushing & dest_free & in_bounds
        can_walk = ~pushing & in_bounds
        moved = can_walk | can_push
        push_mask = at_new & can_push.unsqueeze(1)
        boxes[:, :, 0] = torch.where(push_mask, pr.unsqueeze(1), boxes[:, :, 0])
        boxes[:, :, 1] = torch.where(push_mask, pc.unsqueeze(1), boxes[:, :, 1])
        agent[:, 0] = torch.where(moved, nr, agent[:, 0])
        agent[:, 1] = torch.where(moved, nc, agent[:, 1])

        new_state = {"agent": agent, "boxes": boxes, "targets": state["targets"]}
        on_after = self._on_target(new_state).sum(-1)
        solved = on_after == NBOX
        reward = -0.1 + (on_after - on_before).float() + 10.0 * solved.float()
        return new_state, reward, solved
