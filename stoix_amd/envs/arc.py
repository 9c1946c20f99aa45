"""ARC-class grid input/output task (jaxarc capability-class slice).

The reference's jaxarc suite (JAX-only external) wraps ARC-style tasks:
the agent sees a TARGET grid and must produce it on a canvas through
primitive editing actions. This is the copy-task slice of that space —
procedurally generated coloured sprites, a cursor, and paint actions;
reward for first-time-correct cells, penalty for wrong paints. The full
ARC generality (abstract transformation inference across demonstration
pairs) remains out of scope and is documented as such (PARITY.md).

Grid 7x7, 3 colours. Observation [7, 7, 3+3+1+1]: target one-hot (3),
canvas one-hot (3), cursor plane, done-progress plane (fraction of
correct cells, broadcast). Actions: 0-3 move cursor, 4-6 paint the
cursor cell with colour k. Episode terminates when every coloured target
cell is painted correctly.
"""
from __future__ import annotations

from typing import Tuple

import torch

from stoix_amd.envs.env import State, StatefulVecEnv, Tensor
from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace

G = 7
K = 3  # colours (0 = empty)
N_SPRITE = 10  # coloured cells per target
_DR = [-1, 0, 1, 0]
_DC = [0, 1, 0, -1]


class GridCopy(StatefulVecEnv):
    max_episode_steps = 150
    capture_safe = True
    solved_return_threshold = 6.0

    def __init__(self, num_envs, device="cpu", seed=0, **kw):
        super().__init__(num_envs, device, seed)
        self.observation_space = BoxSpace((G, G, 2 * K + 2), 0.0, 1.0)
        self.action_space = DiscreteSpace(4 + K)
        self._dr = torch.tensor(_DR, device=self.device)
        self._dc = torch.tensor(_DC, device=self.device)
        self._one_f = torch.ones((), device=self.device)

    def _reset_fn(self, n: int) -> State:
        dev = self.device
        target = torch.zeros(n, G * G, dtype=torch.long, device=dev)
        occ = torch.zeros(n, G * G, dtype=torch.bool, device=dev)
        for _ in range(N_SPRITE):
            u = torch.rand(n, G * G, device=dev, generator=self.gen)
            gum = -torch.log(-torch.log(u.clamp(min=1e-12)).clamp(min=1e-12))
            cell = torch.where(occ, torch.full_like(gum, -torch.inf), gum).argmax(-1)
            occ.scatter_(1, cell.unsqueeze(1), True)
            color = torch.randint(1, K + 1, (n,), device=dev, generator=self.gen)
            target.scatter_(1, cell.unsqueeze(1), color.unsqueeze(1))
        cursor = torch.full((n,), (G // 2) * G + G // 2, dtype=torch.long, device=dev)
        return {
            # the grid the OBS displays; the copy task shows the scoring
            # target itself, transformation subclasses show the input
            "shown": target.float(),
            "target": target.float(),
            "canvas": torch.zeros(n, G * G, device=dev),
            "cursor": cursor.float(),
            # first-time-correct latches: each cell pays its +1 ONCE per
            # episode (without this, paint-correct -> overwrite -> repaint
            # farms +0.95 per cycle; a PPO probe actually found it, return
            # 66 on a 10-cell board)
            "rewarded": torch.zeros(n, G * G, device=dev),
        }

    def _obs_fn(self, state: State) -> Tensor:
        n = state["cursor"].shape[0]
        dev = self.device
        target = state["shown"].long().view(n, G, G)
        canvas = state["canvas"].long().view(n, G, G)
        # colour one-hots without the empty class
        t_oh = torch.nn.functional.one_hot(target.clamp(0, K), K + 1)[..., 1:].float()
        c_oh = torch.nn.functional.one_hot(canvas.clamp(0, K), K + 1)[..., 1:].float()
        cur = torch.zeros(n, G, G, 1, device=dev)
        cpos = state["cursor"].long()
        bidx = torch.arange(n, device=dev)
        cur[bidx, cpos // G, cpos % G, 0] = self._one_f
        correct = (
            ((state["target"] == state["canvas"]) & (state["target"] > 0))
            .float()
            .sum(-1, keepdim=True)
        ) / float(N_SPRITE)
        prog = correct.view(n, 1, 1, 1).expand(n, G, G, 1)
        return torch.cat([t_oh, c_oh, cur, prog], dim=-1)

    def _step_fn(self, state: State, action: Tensor) -> Tuple[State, Tensor, Tensor]:
        n = state["cursor"].shape[0]
        dev = self.device
        a = action.long().clamp(0, 3 + K)
        cpos = state["cursor"].long()
        r, c = cpos // G, cpos % G
        move = a < 4
        d = torch.where(move, a, torch.zeros_like(a))
        nr = (r + torch.where(move, self._dr[d], torch.zeros_like(r))).clamp(0, G - 1)
        nc = (c + torch.where(move, self._dc[d], torch.zeros_like(c))).clamp(0, G - 1)
        npos = nr * G + nc

        paint = a >= 4
        color = (a - 3).clamp(1, K).float()  # 1..K
        tgt_here = state["target"].gather(1, npos.unsqueeze(1)).squeeze(1)
        old_here = state["canvas"].gather(1, npos.unsqueeze(1)).squeeze(1)
        new_here = torch.where(paint, color, old_here)
        canvas = state["canvas"].scatter(1, npos.unsqueeze(1), new_here.unsqueeze(1))

        rewarded_here = state["rewarded"].gather(1, npos.unsqueeze(1)).squeeze(1)
        newly_correct = (
            paint & (new_here == tgt_here) & (tgt_here > 0) & (rewarded_here < 0.5)
        )
        wrong = paint & (new_here != tgt_here)
        reward = newly_correct.float() - 0.05 * wrong.float()
        rewarded = state["rewarded"].scatter(
            1, npos.unsqueeze(1),
            torch.maximum(rewarded_here, newly_correct.float()).unsqueeze(1),
        )
        # completion: every coloured target cell painted correctly (boolean
        # implication target>0 -> canvas==target)
        terminated = ((state["target"] > 0) <= (canvas == state["target"])).all(dim=-1)
        return (
            {"shown": state["shown"], "target": state["target"], "canvas": canvas,
             "cursor": npos.float(), "rewarded": rewarded},
            reward,
            terminated,
        )


class GridMirror(GridCopy):
    """ARC concept-class transformation task (reference jaxarc concept
    grouping): the obs shows an INPUT sprite; the scored target is its
    HORIZONTAL MIRROR. Unlike GridCopy (answer visible), the agent must
    internalise the transformation rule — paint the mirror of what it
    sees."""

    def _reset_fn(self, n):
        state = super()._reset_fn(n)
        mirrored = state["shown"].view(n, G, G).flip(-1).reshape(n, G * G)
        state["target"] = mirrored
        return state
