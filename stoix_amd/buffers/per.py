"""Prioritised replay with a vectorised device-resident sum-tree.

Functional parity with flashbax ``make_prioritised_trajectory_buffer`` as
used by Rainbow / R2D2 (/root/reference/stoix/systems/q_learning/
ff_rainbow.py:433-444 incl. ``set_priorities``, rec_r2d2 sequences).

The sum-tree is a flat level-order array of size 2*cap; batched updates are
scatter-adds per level and batched sampling is a log2(cap)-step descent done
with gathers — both fully vectorised device ops (the same structure the HIP
kernel version uses; each level touch is one coalesced gather).
"""
from __future__ import annotations

from typing import Dict, Tuple

import torch

from stoix_amd.buffers.trajectory import TrajectoryBuffer

Tensor = torch.Tensor


class SumTree:
    def __init__(self, capacity: int, device: torch.device | str = "cpu"):
        self.capacity = 1
        while self.capacity < capacity:
            self.capacity *= 2
        self.n_items = capacity
        self.device = torch.device(device)
        self.tree = torch.zeros(2 * self.capacity, device=self.device)

    @torch.no_grad()
    def set(self, idx: Tensor, priority: Tensor) -> None:
        """Set priorities at item indices (batched; duplicate idx keep the
        last write via index_put)."""
        leaf = idx.long() + self.capacity
        self.tree[leaf] = priority
        # rebuild ancestors of touched leaves level by level
        nodes = torch.unique(leaf // 2)
        while nodes.numel() > 0 and nodes[0] >= 1:
            self.tree[nodes] = self.tree[2 * nodes] + self.tree[2 * nodes + 1]
            nodes = torch.unique(nodes // 2)
            if nodes.numel() == 1 and nodes[0] == 0:
                break

    @property
    def total(self) -> Tensor:
        return self.tree[1]

    @torch.no_grad()
    def sample(self, batch_size: int, generator=None) -> Tensor:
        """Stratified proportional sampling: batched tree descent."""
        seg = self.total / batch_size
        u = torch.rand(batch_size, device=self.device, generator=generator)
        mass = (torch.arange(batch_size, device=self.device, dtype=torch.float32) + u) * seg
        node = torch.ones(batch_size, dtype=torch.long, device=self.device)
        depth = int(torch.log2(torch.tensor(float(self.capacity))).item())
        for _ in range(depth):
            left = 2 * node
            left_sum = self.tree[left]
            go_right = mass >= left_sum
            mass = torch.where(go_right, mass - left_sum, mass)
            node = torch.where(go_right, left + 1, left)
        item = (node - self.capacity).clamp(0, self.n_items - 1)
        return item

    def get(self, idx: Tensor) -> Tensor:
        return self.tree[idx.long() + self.capacity]


class PrioritisedBuffer(TrajectoryBuffer):
    """Prioritised sequence buffer: proportional sampling with alpha-powered
    priorities and IS weights (1/(N p))^beta / max."""

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
        self._max_priority = 1.0

    def _slot(self, rows: Tensor, t0: Tensor) -> Tensor:
        return rows * self.t_max + t0

    @torch.no_grad()
    def add(self, batch: Dict[str, Tensor]) -> None:
        t_block = next(iter(batch.values())).shape[1]
        t_start = self.t_ptr
        super().add(batch)
        # new items get max priority so they are sampled at least once
        offs = (torch.arange(t_block, device=self.device) + t_start) % self.t_max
        rows = torch.arange(self.rows, device=self.device)
        slots = (rows.unsqueeze(1) * self.t_max + offs.unsqueeze(0)).reshape(-1)
        # only starts with a full valid window ahead are sampleable; priority
        # zero marks unsampleable slots. A start is valid when its whole
        # window lies in filled data; approximate by marking slots older than
        # seq_len behind the pointer valid (exact masking at sample()).
        self.tree.set(slots, torch.full((slots.numel(),), self._max_priority**self.alpha, device=self.device))
        # invalidate the seq_len-1 slots straight behind the new pointer
        # (their windows would cross the write head)
        inv = (torch.arange(self.seq_len - 1, device=self.device) + self.t_ptr - (self.seq_len - 1)) % self.t_max
        inv_slots = (rows.unsqueeze(1) * self.t_max + inv.unsqueeze(0)).reshape(-1)
        self.tree.set(inv_slots, torch.zeros(inv_slots.numel(), device=self.device))

    @torch.no_grad()
    def sample(self, batch_size: int, importance_sampling_exponent: float = 0.4) -> Dict[str, Tensor]:
        slots = self.tree.sample(batch_size, self.gen)
        rows = slots // self.t_max
        t0 = slots % self.t_max
        offs = torch.arange(self.seq_len, device=self.device)
        tidx = (t0.unsqueeze(1) + offs.unsqueeze(0)) % self.t_max
        out = {k: v[rows.unsqueeze(1), tidx] for k, v in self.storage.items()}
        pr = self.tree.get(slots)
        probs = pr / self.tree.total.clamp(min=1e-12)
        n = (self.tree.tree[self.tree.capacity :] > 0).sum().clamp(min=1)
        weights = (1.0 / (probs * n).clamp(min=1e-12)) ** importance_sampling_exponent
        weights = weights / weights.max().clamp(min=1e-12)
        out["_rows"] = rows
        out["_t0"] = t0
        out["_slots"] = slots
        out["_weights"] = weights
        return out

    @torch.no_grad()
    def set_priorities(self, slots: Tensor, priorities: Tensor) -> None:
        priorities = priorities.abs().clamp(min=1e-6)
        self._max_priority = max(self._max_priority, float(priorities.max()))
        self.tree.set(slots, priorities**self.alpha)

    @property
    def can_sample(self) -> bool:
        return self.t_filled >= self.seq_len and float(self.tree.total) > 0
