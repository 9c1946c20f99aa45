"""Prioritised replay with a device-resident sum-tree.

Functional parity with flashbax ``make_prioritised_trajectory_buffer`` as
used by Rainbow / R2D2 (/root/reference/stoix/systems/q_learning/
ff_rainbow.py:433-444 incl. ``set_priorities``, rec_r2d2 sequences).

The sum-tree is a flat level-order array of size 2*cap. Every operation is
capture-legal (no torch.unique, no host syncs, device max-priority cursor),
so the whole Rainbow update — sample, loss, priority writeback — can be
captured into ONE hip graph (ops/graph.try_enable_update_graph). On GPU the
update and the stratified sampling descent run as hand-written HIP kernels
(ops/csrc/per.hip: scatter + idempotent per-level ancestor repair, one
launch for batch-sized updates); the torch fallback does the identical
per-level repair with strided tensor ops (CPU tests + numerics parity).
"""
from __future__ import annotations

from typing import Dict

import torch

from stoix_amd.buffers.trajectory import TrajectoryBuffer

Tensor = torch.Tensor


class SumTree:
    def __init__(self, capacity: int, device: torch.device | str = "cpu"):
        self.capacity = 1
        self.depth = 0
        while self.capacity < capacity:
            self.capacity *= 2
            self.depth += 1
        self.n_items = capacity
        self.device = torch.device(device)
        self.tree = torch.zeros(2 * self.capacity, device=self.device)
        self._hip = None
        if self.device.type == "cuda":
            from stoix_amd import ops

            self._hip = ops.ext(required=True)

    @torch.no_grad()
    def set(self, idx: Tensor, priority: Tensor) -> None:
        """Set priorities at item indices (batched, capture-legal).

        Duplicate indices: an arbitrary scatter winner, then an idempotent
        per-level ancestor recompute (parent = sum of children read fresh)
        — never double-counts, unlike atomic delta propagation."""
        idx = idx.long()
        priority = priority.float()
        if self._hip is not None:
            self._hip.sumtree_update(self.tree, idx.contiguous(),
                                     priority.contiguous(), self.capacity,
                                     self.depth)
            return
        leaf = idx + self.capacity
        self.tree[leaf] = priority
        node = leaf
        for _ in range(self.depth):
            node = node >> 1
            # gather children first, then scatter: duplicate nodes write
            # identical values, and reads (level l-1) never alias writes
            # (level l)
            self.tree[node] = self.tree[2 * node] + self.tree[2 * node + 1]

    @property
    def total(self) -> Tensor:
        return self.tree[1]

    @torch.no_grad()
    def sample(self, batch_size: int, generator=None) -> Tensor:
        """Stratified proportional sampling: batched tree descent."""
        u = torch.rand(batch_size, device=self.device, generator=generator)
        if self._hip is not None:
            out = torch.empty(batch_size, dtype=torch.long, device=self.device)
            self._hip.sumtree_sample(self.tree, u, out, self.capacity,
                                     self.depth, self.n_items)
            return out
        seg = self.total / batch_size
        mass = (torch.arange(batch_size, device=self.device, dtype=torch.float32) + u) * seg
        node = torch.ones(batch_size, dtype=torch.long, device=self.device)
        for _ in range(self.depth):
            left = 2 * node
            left_sum = self.tree[left]
            go_right = mass >= left_sum
            mass = torch.where(go_right, mass - left_sum, mass)
            node = torch.where(go_right, left + 1, left)
        return (node - self.capacity).clamp(0, self.n_items - 1)

    def get(self, idx: Tensor) -> Tensor:
        return self.tree[idx.long() + self.capacity]


class PrioritisedBuffer(TrajectoryBuffer):
    """Prioritised sequence buffer: proportional sampling with alpha-powered
    priorities and IS weights (1/(N p))^beta / max. Fully device-resident
    cursors (graph-capturable end to end)."""

    def __init__(
        self,
        add_batch_size: int,
        max_length_time_axis: int,
        sample_sequence_length: int,
        device: torch.device | str = "cpu",
        seed: int = 0,
        period: int = 1,
        priority_exponent: float = 0.5,
    ):
        super().__init__(add_batch_size, max_length_time_axis, sample_sequence_length, device, seed, period)
        self.alpha = priority_exponent
        # one priority per (row, t0) start slot
        self.n_slots = self.rows * self.t_max
        self.tree = SumTree(self.n_slots, device)
        self._max_priority = torch.ones((), device=self.device)
        self._rowss = torch.arange(self.rows, device=self.device)

    def _slot(self, rows: Tensor, t0: Tensor) -> Tensor:
        return rows * self.t_max + t0

    @torch.no_grad()
    def add(self, batch: Dict[str, Tensor]) -> None:
        t_block = next(iter(batch.values())).shape[1]
        t_start = self._t_ptr.clone()  # device cursors, pre-advance
        t_filled_pre = self._t_filled.clone()
        super().add(batch)
        S = self.seq_len - 1
        # ONE tree update per add covering [t_start - S, t_start + t_block):
        #   * the S columns BEHIND the old head are re-validated — their
        #     windows now continue into the fresh data (skipping this, as
        #     round 1 did, permanently starved every window straddling a
        #     block boundary of sampling until the ring wrapped);
        #     guarded by t_filled so unwritten slots stay at priority 0;
        #   * the new columns get max priority so they are sampled at
        #     least once;
        #   * the S columns straight behind the NEW head get priority 0
        #     (their windows would cross the write head).
        # All shapes are static and all values device-computed, so the call
        # is hip-graph capturable.
        if S + t_block > self.t_max:
            raise ValueError(
                f"add block ({t_block}) + seq overlap ({S}) exceed the ring "
                f"length ({self.t_max}); priorities would alias"
            )
        rel = torch.arange(-S, t_block, device=self.device)  # [S + t_block]
        offs = (rel + t_start) % self.t_max
        maxp = self._max_priority ** self.alpha
        col_prio = torch.where(
            rel >= t_block - S,  # the S columns behind the NEW head
            torch.zeros((), device=self.device),
            maxp,
        )
        # behind-columns only valid where data exists (rel >= -t_filled_pre)
        col_prio = torch.where(rel < -t_filled_pre, torch.zeros((), device=self.device), col_prio)
        slots = (self._rowss.unsqueeze(1) * self.t_max + offs.unsqueeze(0)).reshape(-1)
        prio = col_prio.unsqueeze(0).expand(self.rows, -1).reshape(-1).contiguous()
        self.tree.set(slots, prio)

    @torch.no_grad()
    def sample(self, batch_size: int, importance_sampling_exponent=0.4) -> Dict[str, Tensor]:
        gen = None if self.graph_safe_rng else self.gen
        slots = self.tree.sample(batch_size, gen)
        rows = slots // self.t_max
        t0 = slots % self.t_max
        offs = torch.arange(self.seq_len, device=self.device)
        tidx = (t0.unsqueeze(1) + offs.unsqueeze(0)) % self.t_max
        out = {k: v[rows.unsqueeze(1), tidx] for k, v in self.storage.items()}
        pr = self.tree.get(slots)
        probs = pr / self.tree.total.clamp(min=1e-12)
        n = (self.tree.tree[self.tree.capacity :] > 0).sum().clamp(min=1)
        beta = importance_sampling_exponent
        if not torch.is_tensor(beta):
            beta = torch.as_tensor(float(beta), device=self.device)
        weights = (1.0 / (probs * n).clamp(min=1e-12)) ** beta
        weights = weights / weights.max().clamp(min=1e-12)
        out["_rows"] = rows
        out["_t0"] = t0
        out["_slots"] = slots
        out["_weights"] = weights
        return out

    @torch.no_grad()
    def set_priorities(self, slots: Tensor, priorities: Tensor) -> None:
        priorities = priorities.abs().clamp(min=1e-6)
        # in-place update of the STABLE max-priority buffer: add() reads this
        # tensor's address inside the captured graph, so rebinding the
        # attribute to a fresh tensor would leave replays reading stale data
        self._max_priority.copy_(torch.maximum(self._max_priority, priorities.max()))
        self.tree.set(slots, priorities**self.alpha)

    @property
    def can_sample(self) -> bool:
        return self.t_filled >= self.seq_len and float(self.tree.total) > 0
