"""XLand-MiniGrid-class goal-conditioned gridworld (vectorised torch).

Restores the capability class of the reference's xland_minigrid suite
(/root/reference/stoix/utils/make_env.py:211-274 routes to the external
JAX-only xminigrid package): procedurally-generated rooms with coloured
objects and a PER-EPISODE GOAL the agent can only infer from its
observation — structured pixel-ish obs + goal conditioning + sparse
reward. Original design (not a port): every op is batched over B boards,
so it runs on CPU and as device tensors on GPU.

Grid 9x9 with border walls plus random interior walls; ``NUM_COLORS``
coloured objects are placed at random free cells; each episode samples a
goal colour. Reaching the goal object's cell gives +1 and terminates;
reaching a WRONG object gives -0.1 and removes it (the agent can recover).
Observation [9, 9, 3 + NUM_COLORS] channels: walls, agent, objects by
colour (one plane per colour)... plus a goal plane broadcasting the goal
colour one-hot over the last NUM_COLORS channels' first row convention is
avoided — the goal is a SEPARATE constant plane set: channel
``3 + goal`` is incremented by a constant 0.5 everywhere, so a CNN torso
can read the goal from any receptive field.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace

N = 9
NUM_COLORS = 4
N_WALLS = 6
_DR = [-1, 0, 1, 0]
_DC = [0, 1, 0, -1]


class XLandGrid(StatefulVecEnv):
    max_episode_steps = 100
    capture_safe = True
    solved_return_threshold = 0.9

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((N, N, 3 + NUM_COLORS), 0.0, 1.5)
        self.action_space = DiscreteSpace(4)
        self._dr = torch.tensor(_DR, device=self.device)
        self._dc = torch.tensor(_DC, device=self.device)
        self._one_f = torch.ones((), device=self.device)

    def _sample_free(self, occupied: Tensor, n: int) -> Tensor:
        """Gumbel-max a free cell per board; occupied [n, N*N] bool."""
        u = torch.rand(n, N * N, device=self.device, generator=self.gen)
        g = -torch.log(-torch.log(u.clamp(min=1e-12)).clamp(min=1e-12))
        scores = torch.where(occupied, torch.full_like(g, -torch.inf), g)
        return scores.argmax(dim=-1)

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        walls = torch.zeros(n, N, N, dtype=torch.bool, device=dev)
        walls[:, 0, :] = walls[:, -1, :] = True
        walls[:, :, 0] = walls[:, :, -1] = True
        occ = walls.reshape(n, N * N).clone()
        # random interior walls (may carve dead ends; goals stay reachable
        # often enough for learning — wrong-object penalties keep signal)
        for _ in range(N_WALLS):
            cell = self._sample_free(occ, n)
            occ.scatter_(1, cell.unsqueeze(1), True)
            walls = occ.reshape(n, N, N).clone()
        obj_pos = torch.zeros(n, NUM_COLORS, dtype=torch.long, device=dev)
        for c in range(NUM_COLORS):
            cell = self._sample_free(occ, n)
            occ.scatter_(1, cell.unsqueeze(1), True)
            obj_pos[:, c] = cell
        agent = self._sample_free(occ, n)
        goal = torch.randint(0, NUM_COLORS, (n,), device=dev, generator=self.gen)
        alive = torch.ones(n, NUM_COLORS, dtype=torch.bool, device=dev)
        return {
            "walls": walls.float(),
            "obj_pos": obj_pos.float(),
            "obj_alive": alive.float(),
            "agent": agent.float(),
            "goal": goal.float(),
        }

    def _obs_fn(self, state: State) -> Tensor:
        n = state["agent"].shape[0]
        dev = self.device
        walls = state["walls"]
        agent = state["agent"].long()
        obj_pos = state["obj_pos"].long()
        alive = state["obj_alive"]
        goal = state["goal"].long()
        bidx = torch.arange(n, device=dev)
        obs = torch.zeros(n, N, N, 3 + NUM_COLORS, device=dev)
        obs[..., 0] = walls
        a_r, a_c = agent // N, agent % N
        obs[bidx, a_r, a_c, 1] = self._one_f  # device scalar: capture-legal
        # channel 2: any-object plane; 3+c: per-colour planes
        for c in range(NUM_COLORS):
            r, cc = obj_pos[:, c] // N, obj_pos[:, c] % N
            obs[bidx, r, cc, 2] = torch.maximum(obs[bidx, r, cc, 2], alive[:, c])
            obs[bidx, r, cc, 3 + c] = alive[:, c]
        # goal conditioning: +0.5 broadcast over the goal colour's plane
        goal_onehot = torch.nn.functional.one_hot(goal, NUM_COLORS).float()
        obs[..., 3:] = obs[..., 3:] + 0.5 * goal_onehot.view(n, 1, 1, NUM_COLORS)
        return obs

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        n = state["agent"].shape[0]
        dev = self.device
        a = action.long().clamp(0, 3)
        agent = state["agent"].long()
        r, c = agent // N, agent % N
        nr = (r + self._dr[a]).clamp(0, N - 1)
        nc = (c + self._dc[a]).clamp(0, N - 1)
        walls = state["walls"] > 0.5
        bidx = torch.arange(n, device=dev)
        blocked = walls[bidx, nr, nc]
        nr = torch.where(blocked, r, nr)
        nc = torch.where(blocked, c, nc)
        new_agent = nr * N + nc

        obj_pos = state["obj_pos"].long()
        alive = state["obj_alive"] > 0.5
        goal = state["goal"].long()
        on_obj = (obj_pos == new_agent.unsqueeze(1)) & alive  # [n, C]
        goal_onehot = torch.nn.functional.one_hot(goal, NUM_COLORS).bool()
        hit_goal = (on_obj & goal_onehot).any(dim=1)
        hit_wrong = (on_obj & ~goal_onehot).any(dim=1)
        reward = hit_goal.float() - 0.1 * hit_wrong.float()
        new_alive = alive & ~on_obj  # consumed on touch
        terminated = hit_goal
        return (
            {
                "walls": state["walls"],
                "obj_pos": state["obj_pos"],
                "obj_alive": new_alive.float(),
                "agent": new_agent.float(),
                "goal": state["goal"],
            },
            reward,
            terminated,
        )
