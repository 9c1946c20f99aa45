"""XLand-MiniGrid-class goal-conditioned gridworld (vectorised torch).

Restores the capability class of the reference's xland_minigrid suite
(/root/reference/stoix/utils/make_env.py:211-274 routes to the external
JAX-only xminigrid package): procedurally-generated rooms with coloured
objects and a PER-EPISODE GOAL the agent can only infer from its
observation — structured pixel-ish obs + goal conditioning + sparse
reward. Original design (not a port): every op is batched over B boards,
so it runs on CPU and as device tensors on GPU.

Grid 9x9 with border walls plus random interior walls; ``C`` (class attr)
coloured objects are placed at random free cells; each episode samples a
goal colour. Reaching the goal object's cell gives +1 and terminates;
reaching a WRONG object gives -0.1 and removes it (the agent can recover).
Observation [N, N, 3 + C] channels: walls, agent, objects by
colour (one plane per colour)... plus a goal plane broadcasting the goal
colour one-hot over the last C channels' first row convention is
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
    # grid size / colour count / interior walls are class attrs so the
    # reference's sized scenarios (empty_5x5, empty_6x6, door_key_5x5, ...)
    # instantiate as subclasses
    N = 9
    C = 4
    N_WALLS = 6
    max_episode_steps = 100
    capture_safe = True
    solved_return_threshold = 0.9

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((self.N, self.N, 3 + self.C), 0.0, 1.5)
        self.action_space = DiscreteSpace(4)
        self._dr = torch.tensor(_DR, device=self.device)
        self._dc = torch.tensor(_DC, device=self.device)
        self._one_f = torch.ones((), device=self.device)

    def _sample_free(self, occupied: Tensor, n: int) -> Tensor:
        """Gumbel-max a free cell per board; occupied [n, self.N*self.N] bool."""
        u = torch.rand(n, self.N * self.N, device=self.device, generator=self.gen)
        g = -torch.log(-torch.log(u.clamp(min=1e-12)).clamp(min=1e-12))
        scores = torch.where(occupied, torch.full_like(g, -torch.inf), g)
        return scores.argmax(dim=-1)

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        walls = torch.zeros(n, self.N, self.N, dtype=torch.bool, device=dev)
        walls[:, 0, :] = walls[:, -1, :] = True
        walls[:, :, 0] = walls[:, :, -1] = True
        occ = walls.reshape(n, self.N * self.N).clone()
        # random interior walls (may carve dead ends; goals stay reachable
        # often enough for learning — wrong-object penalties keep signal)
        for _ in range(self.N_WALLS):
            cell = self._sample_free(occ, n)
            occ.scatter_(1, cell.unsqueeze(1), True)
            walls = occ.reshape(n, self.N, self.N).clone()
        obj_pos = torch.zeros(n, self.C, dtype=torch.long, device=dev)
        for c in range(self.C):
            cell = self._sample_free(occ, n)
            occ.scatter_(1, cell.unsqueeze(1), True)
            obj_pos[:, c] = cell
        agent = self._sample_free(occ, n)
        goal = torch.randint(0, self.C, (n,), device=dev, generator=self.gen)
        alive = torch.ones(n, self.C, dtype=torch.bool, device=dev)
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
        obs = torch.zeros(n, self.N, self.N, 3 + self.C, device=dev)
        obs[..., 0] = walls
        a_r, a_c = agent // self.N, agent % self.N
        obs[bidx, a_r, a_c, 1] = self._one_f  # device scalar: capture-legal
        # channel 2: any-object plane; 3+c: per-colour planes
        for c in range(self.C):
            r, cc = obj_pos[:, c] // self.N, obj_pos[:, c] % self.N
            obs[bidx, r, cc, 2] = torch.maximum(obs[bidx, r, cc, 2], alive[:, c])
            obs[bidx, r, cc, 3 + c] = alive[:, c]
        # goal conditioning: +0.5 broadcast over the goal colour's plane
        goal_onehot = torch.nn.functional.one_hot(goal, self.C).float()
        obs[..., 3:] = obs[..., 3:] + 0.5 * goal_onehot.view(n, 1, 1, self.C)
        return obs

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        n = state["agent"].shape[0]
        dev = self.device
        a = action.long().clamp(0, 3)
        agent = state["agent"].long()
        r, c = agent // self.N, agent % self.N
        nr = (r + self._dr[a]).clamp(0, self.N - 1)
        nc = (c + self._dc[a]).clamp(0, self.N - 1)
        walls = state["walls"] > 0.5
        bidx = torch.arange(n, device=dev)
        blocked = walls[bidx, nr, nc]
        nr = torch.where(blocked, r, nr)
        nc = torch.where(blocked, c, nc)
        new_agent = nr * self.N + nc

        obj_pos = state["obj_pos"].long()
        alive = state["obj_alive"] > 0.5
        goal = state["goal"].long()
        on_obj = (obj_pos == new_agent.unsqueeze(1)) & alive  # [n, C]
        goal_onehot = torch.nn.functional.one_hot(goal, self.C).bool()
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


class DoorKeyGrid(StatefulVecEnv):
    """DoorKey (navix/MiniGrid-class): a wall splits the room; the agent
    must pick up the KEY, open the DOOR, and reach the GOAL. Sparse
    terminal reward 1 - 0.9 * t/T (the MiniGrid shaping), sub-rewards 0.
    Restores the navix suite's capability class (the reference's navix is
    JAX-only, make_env.py). Vectorised torch; capture-safe.

    Observation [9, 9, 6]: walls, agent, key (if not held), door (closed),
    goal, held-key indicator plane (broadcast 0.5 when carrying).
    Actions: 0-3 move (pickup/open happen by walking into the cell).
    """

    N = 9
    WALL_COL_RANGE = (3, 6)  # sampled dividing-wall columns [lo, hi)
    max_episode_steps = 200
    capture_safe = True
    solved_return_threshold = 0.5

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((self.N, self.N, 6), 0.0, 1.5)
        self.action_space = DiscreteSpace(4)
        self._dr = torch.tensor(_DR, device=self.device)
        self._dc = torch.tensor(_DC, device=self.device)
        self._one_f = torch.ones((), device=self.device)

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        # dividing wall at a random column 3..5 with a door at a random row
        lo, hi = self.WALL_COL_RANGE
        wall_col = torch.randint(lo, hi, (n,), device=dev, generator=self.gen)
        door_row = torch.randint(1, self.N - 1, (n,), device=dev, generator=self.gen)
        cols = torch.arange(self.N, device=dev)
        walls = torch.zeros(n, self.N, self.N, dtype=torch.bool, device=dev)
        walls[:, 0, :] = walls[:, -1, :] = True
        walls[:, :, 0] = walls[:, :, -1] = True
        walls |= cols.view(1, 1, self.N) == wall_col.view(n, 1, 1)
        # carve the door cell out of the wall mask (it is tracked separately)
        bidx = torch.arange(n, device=dev)
        walls[bidx, door_row, wall_col] = False
        # key on the LEFT side, goal on the RIGHT side, agent LEFT
        u = torch.rand(n, 4, device=dev, generator=self.gen)
        key_r = 1 + (u[:, 0] * (self.N - 2)).long().clamp(max=self.N - 3)
        key_c = 1 + (u[:, 1] * (wall_col.float() - 1.0)).long().clamp(min=0)
        key_c = torch.minimum(key_c, wall_col - 1).clamp(min=1)
        agent_r = 1 + (u[:, 2] * (self.N - 2)).long().clamp(max=self.N - 3)
        agent_c = torch.ones(n, dtype=torch.long, device=dev)
        goal_r = door_row  # reachable by construction
        goal_c = torch.full((n,), self.N - 2, dtype=torch.long, device=dev)
        return {
            "walls": walls.float(),
            "door_r": door_row.float(), "door_c": wall_col.float(),
            "door_open": torch.zeros(n, device=dev),
            "key_r": key_r.float(), "key_c": key_c.float(),
            "has_key": torch.zeros(n, device=dev),
            "agent_r": agent_r.float(), "agent_c": agent_c.float(),
            "goal_r": goal_r.float(), "goal_c": goal_c.float(),
            # episode time lives IN the state so _step_fn stays functional
            # (search systems drive it with foreign batches); mirrors
            # self._step_count under normal stepping
            "t": torch.zeros(n, device=dev),
        }

    def _obs_fn(self, state: State) -> Tensor:
        n = state["agent_r"].shape[0]
        dev = self.device
        bidx = torch.arange(n, device=dev)
        obs = torch.zeros(n, self.N, self.N, 6, device=dev)
        obs[..., 0] = state["walls"]
        obs[bidx, state["agent_r"].long(), state["agent_c"].long(), 1] = self._one_f
        key_vis = (state["has_key"] < 0.5).float()
        obs[bidx, state["key_r"].long(), state["key_c"].long(), 2] = key_vis
        door_closed = (state["door_open"] < 0.5).float()
        obs[bidx, state["door_r"].long(), state["door_c"].long(), 3] = door_closed
        obs[bidx, state["goal_r"].long(), state["goal_c"].long(), 4] = self._one_f
        obs[..., 5] = 0.5 * state["has_key"].view(n, 1, 1)
        return obs

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        n = state["agent_r"].shape[0]
        dev = self.device
        a = action.long().clamp(0, 3)
        r, c = state["agent_r"].long(), state["agent_c"].long()
        nr = (r + self._dr[a]).clamp(0, self.N - 1)
        nc = (c + self._dc[a]).clamp(0, self.N - 1)
        bidx = torch.arange(n, device=dev)
        walls = state["walls"] > 0.5
        at_door = (nr == state["door_r"].long()) & (nc == state["door_c"].long())
        door_open = state["door_open"] > 0.5
        has_key = state["has_key"] > 0.5
        # a closed door blocks unless the agent carries the key (walking
        # into it with the key OPENS it and moves through)
        blocked = walls[bidx, nr, nc] | (at_door & ~door_open & ~has_key)
        nr = torch.where(blocked, r, nr)
        nc = torch.where(blocked, c, nc)
        new_door_open = door_open | (at_door & has_key & ~blocked)
        on_key = (nr == state["key_r"].long()) & (nc == state["key_c"].long()) & ~has_key
        new_has_key = has_key | on_key
        at_goal = (nr == state["goal_r"].long()) & (nc == state["goal_c"].long())
        # MiniGrid terminal shaping: 1 - 0.9 * t / T (t carried in state)
        t = state["t"] + 1.0
        frac = t / float(self.max_episode_steps)
        reward = at_goal.float() * (1.0 - 0.9 * frac)
        return (
            {
                "walls": state["walls"],
                "door_r": state["door_r"], "door_c": state["door_c"],
                "door_open": new_door_open.float(),
                "key_r": state["key_r"], "key_c": state["key_c"],
                "has_key": new_has_key.float(),
                "agent_r": nr.float(), "agent_c": nc.float(),
                "goal_r": state["goal_r"], "goal_c": state["goal_c"],
                "t": t,
            },
            reward,
            at_goal,
        )


class EmptyGrid5(XLandGrid):
    """MiniGrid Empty-5x5-class (reference xland_minigrid/empty_5x5.yaml,
    navix/empty_5x5.yaml): one object, no interior walls — reach it."""

    N = 5
    C = 1
    N_WALLS = 0
    max_episode_steps = 50


class EmptyGrid6(EmptyGrid5):
    """Empty-6x6 (reference xland_minigrid/empty_6x6.yaml)."""

    N = 6


class DoorKeyGrid5(DoorKeyGrid):
    """DoorKey-5x5 (reference xland_minigrid/door_key_5x5.yaml)."""

    N = 5
    WALL_COL_RANGE = (2, 3)
    max_episode_steps = 100


class DoorKeyGrid8(DoorKeyGrid):
    """DoorKey-8x8 (reference navix/door_key_8x8.yaml)."""

    N = 8
    WALL_COL_RANGE = (2, 6)
