"""Craftax/Crafter-class crafting env (vectorised torch).

Restores the capability class of the reference's craftax suite
(/root/reference/stoix/utils/make_env.py: craftax is JAX-only external):
an open-world grid with RESOURCES, an INVENTORY, and a sparse
achievement-chain reward — the defining Crafter/Craftax structure
(structured obs + long-horizon milestones). Original design, fully
batched over B worlds.

World 11x11: cell types {0 empty, 1 tree, 2 stone, 3 table}. Actions:
0-3 move, 4 interact (on the faced cell), 5 craft. Mechanics:
  * interact on a TREE  -> +1 wood, tree becomes empty
  * interact on STONE   -> +1 stone, needs a pickaxe
  * craft with wood>=2                 -> place a TABLE on the faced
    empty cell (consumes 2 wood)
  * craft next to a TABLE with wood>=1 and stone>=0 -> PICKAXE
    (consumes 1 wood)
First-time achievements give +1 each (Crafter-style): collect_wood,
place_table, make_pickaxe, collect_stone. Max return 4.0 per episode.

Observation: map one-hot [11, 11, 5] (4 cell types + agent plane)
flattened and concatenated with 7 inventory/achievement scalars.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace

M = 11
N_TREES = 8
N_STONES = 5
_DR = [-1, 0, 1, 0]
_DC = [0, 1, 0, -1]


class Crafting(StatefulVecEnv):
    max_episode_steps = 200
    capture_safe = True
    solved_return_threshold = 3.0

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((M * M * 5 + 7,), 0.0, 10.0)
        self.action_space = DiscreteSpace(6)
        self._dr = torch.tensor(_DR, device=self.device)
        self._dc = torch.tensor(_DC, device=self.device)
        self._one_f = torch.ones((), device=self.device)

    def _sample_free(self, occupied: Tensor, n: int) -> Tensor:
        u = torch.rand(n, M * M, device=self.device, generator=self.gen)
        g = -torch.log(-torch.log(u.clamp(min=1e-12)).clamp(min=1e-12))
        scores = torch.where(occupied, torch.full_like(g, -torch.inf), g)
        return scores.argmax(dim=-1)

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        grid = torch.zeros(n, M * M, dtype=torch.long, device=dev)
        occ = torch.zeros(n, M * M, dtype=torch.bool, device=dev)
        for _ in range(N_TREES):
            cell = self._sample_free(occ, n)
            occ.scatter_(1, cell.unsqueeze(1), True)
            grid.scatter_(1, cell.unsqueeze(1), 1)
        for _ in range(N_STONES):
            cell = self._sample_free(occ, n)
            occ.scatter_(1, cell.unsqueeze(1), True)
            grid.scatter_(1, cell.unsqueeze(1), 2)
        agent = self._sample_free(occ, n)
        z = torch.zeros(n, device=dev)
        return {
            "grid": grid.float(),
            "agent": agent.float(),
            "facing": torch.zeros(n, device=dev),  # last move direction
            "wood": z.clone(),
            "stone": z.clone(),
            "pickaxe": z.clone(),
            # first-time achievement latches
            "ach_wood": z.clone(),
            "ach_table": z.clone(),
            "ach_pick": z.clone(),
            "ach_stone": z.clone(),
        }

    def _obs_fn(self, state: State) -> Tensor:
        n = state["agent"].shape[0]
        dev = self.device
        grid = state["grid"].long().view(n, M, M)
        onehot = torch.nn.functional.one_hot(grid.clamp(0, 3), 4).float()
        agent_plane = torch.zeros(n, M, M, 1, device=dev)
        a = state["agent"].long()
        bidx = torch.arange(n, device=dev)
        agent_plane[bidx, a // M, a % M, 0] = self._one_f
        planes = torch.cat([onehot, agent_plane], dim=-1).reshape(n, -1)
        inv = torch.stack(
            [
                state["wood"], state["stone"], state["pickaxe"],
                state["ach_wood"], state["ach_table"], state["ach_pick"],
                state["ach_stone"],
            ],
            dim=-1,
        )
        return torch.cat([planes, inv], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        n = state["agent"].shape[0]
        dev = self.device
        a = action.long().clamp(0, 5)
        agent = state["agent"].long()
        facing = state["facing"].long().clamp(0, 3)
        grid = state["grid"].long().clone()

        move = a < 4
        dirn = torch.where(move, a, facing)
        r, c = agent // M, agent % M
        tr = (r + self._dr[dirn]).clamp(0, M - 1)
        tc = (c + self._dc[dirn]).clamp(0, M - 1)
        target = tr * M + tc
        target_cell = grid.gather(1, target.unsqueeze(1)).squeeze(1)

        # movement: walk onto empty cells only
        can_walk = move & (target_cell == 0) & (target != agent)
        new_agent = torch.where(can_walk, target, agent)

        wood = state["wood"].clone()
        stone = state["stone"].clone()
        pickaxe = state["pickaxe"].clone()

        interact = a == 4
        chop = interact & (target_cell == 1)
        mine = interact & (target_cell == 2) & (pickaxe > 0)
        wood = wood + chop.float()
        stone = stone + mine.float()
        # consumed resources leave empty cells
        consumed = chop | mine
        grid.scatter_(
            1, target.unsqueeze(1),
            torch.where(consumed, torch.zeros_like(target_cell), target_cell).unsqueeze(1),
        )

        craft = a == 5
        # near-table test: any of the 4 neighbours is a table
        near_table = torch.zeros(n, dtype=torch.bool, device=dev)
        for d in range(4):
            qr = (new_agent // M + self._dr[d]).clamp(0, M - 1)
            qc = (new_agent % M + self._dc[d]).clamp(0, M - 1)
            q = qr * M + qc
            near_table |= grid.gather(1, q.unsqueeze(1)).squeeze(1) == 3
        make_pick = craft & near_table & (wood >= 1) & (pickaxe == 0)
        wood = wood - make_pick.float()
        pickaxe = pickaxe + make_pick.float()
        # place a table on the faced empty cell (when not making a pickaxe)
        target_cell2 = grid.gather(1, target.unsqueeze(1)).squeeze(1)
        place_table = craft & ~make_pick & (wood >= 2) & (target_cell2 == 0) & (target != new_agent)
        wood = wood - 2.0 * place_table.float()
        grid.scatter_(
            1, target.unsqueeze(1),
            torch.where(place_table, torch.full_like(target_cell2, 3), target_cell2).unsqueeze(1),
        )

        # first-time achievements
        new_ach_wood = torch.maximum(state["ach_wood"], chop.float())
        new_ach_table = torch.maximum(state["ach_table"], place_table.float())
        new_ach_pick = torch.maximum(state["ach_pick"], make_pick.float())
        new_ach_stone = torch.maximum(state["ach_stone"], mine.float())
        reward = (
            (new_ach_wood - state["ach_wood"])
            + (new_ach_table - state["ach_table"])
            + (new_ach_pick - state["ach_pick"])
            + (new_ach_stone - state["ach_stone"])
        )
        all_done = (new_ach_wood + new_ach_table + new_ach_pick + new_ach_stone) >= 4.0
        return (
            {
                "grid": grid.float(),
                "agent": new_agent.float(),
                "facing": dirn.float(),
                "wood": wood,
                "stone": stone,
                "pickaxe": pickaxe,
                "ach_wood": new_ach_wood,
                "ach_table": new_ach_table,
                "ach_pick": new_ach_pick,
                "ach_stone": new_ach_stone,
            },
            reward,
            all_done,
        )


class CraftingPixels(Crafting):
    """Pixel-style observation variant (reference craftax pixels.yaml /
    classic_pixels.yaml scenarios feed image obs to CNN torsos): the same
    world/dynamics with obs as [M, M, 12] image planes — 4 cell-type
    one-hots, the agent plane, and the 7 inventory/achievement scalars
    broadcast as constant planes (so any receptive field can read them)."""

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed, **kw)
        self.observation_space = BoxSpace((M, M, 12), 0.0, 10.0)

    def _obs_fn(self, state: State) -> Tensor:
        flat = super()._obs_fn(state)
        n = flat.shape[0]
        planes = flat[:, : M * M * 5].view(n, M, M, 5)
        inv = flat[:, M * M * 5 :].view(n, 1, 1, 7).expand(n, M, M, 7)
        return torch.cat([planes, inv], dim=-1)
