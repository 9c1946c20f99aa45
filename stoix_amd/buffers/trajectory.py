"""HBM-resident trajectory (sequence) replay buffer.

Functional parity with flashbax ``make_trajectory_buffer`` as used by the
reference (/root/reference/stoix/systems/ddpg/ff_d4pg.py:477,
awr/ff_awr.py:431, mpo/ff_mpo.py:539-545): rows are environment streams,
time is circular per row; sampling draws contiguous windows of
``sample_sequence_length`` steps.

The time cursor and fill count live in DEVICE tensors (same pattern as
ItemBuffer): a captured add() advances the cursor on-device every hip-graph
replay, and sample() computes window starts from the live cursor with no
Python branching — so sequence-replay systems (AWR / D4PG / MPO / MuZero)
can be captured whole by ops/graph.try_enable_update_graph.
"""
from __future__ import annotations

from typing import Dict

import torch

Tensor = torch.Tensor


class TrajectoryBuffer:
    def __init__(
        self,
        add_batch_size: int,
        max_length_time_axis: int,
        sample_sequence_length: int,
        device: torch.device | str = "cpu",
        seed: int = 0,
        period: int = 1,
    ):
        self.rows = int(add_batch_size)
        self.t_max = int(max_length_time_axis)
        self.seq_len = int(sample_sequence_length)
        self.period = int(period)
        self.device = torch.device(device)
        self.gen = torch.Generator(device=self.device)
        self.gen.manual_seed(seed)
        self.storage: Dict[str, Tensor] = {}
        self._t_ptr = torch.zeros(1, dtype=torch.int64, device=self.device)
        self._t_filled = torch.zeros(1, dtype=torch.int64, device=self.device)
        # graph mode switches sampling to the default (graph-aware) RNG
        self.graph_safe_rng = False

    @property
    def t_ptr(self) -> int:
        return int(self._t_ptr.item())

    @property
    def t_filled(self) -> int:
        return int(self._t_filled.item())

    def _alloc(self, example: Dict[str, Tensor]) -> None:
        for k, v in example.items():
            # v: [rows, T_block, ...]
            self.storage[k] = torch.zeros(
                (self.rows, self.t_max, *v.shape[2:]), dtype=v.dtype, device=self.device
            )

    @torch.no_grad()
    def add(self, batch: Dict[str, Tensor]) -> None:
        """Add a [rows, T_block, ...] slab of per-env time steps."""
        if not self.storage:
            self._alloc(batch)
        t_block = next(iter(batch.values())).shape[1]
        idx = (torch.arange(t_block, device=self.device) + self._t_ptr) % self.t_max
        for k, v in batch.items():
            self.storage[k][:, idx] = v.to(self.device)
        self._t_ptr.add_(t_block).remainder_(self.t_max)
        self._t_filled.add_(t_block).clamp_(max=self.t_max)

    @property
    def can_sample(self) -> bool:
        return bool((self._t_filled >= self.seq_len).item())

    def _window_starts(self, batch_size: int) -> tuple:
        """Sample (row, t0) pairs with valid contiguous windows.

        When the buffer has wrapped, windows crossing the write pointer mix
        old/new data; we sample starts in the contiguous valid region behind
        the pointer (standard flashbax behaviour of masking invalid items).
        Branch-free: the oldest valid step is at (t_ptr - t_filled) mod t_max
        — which is 0 before wrapping (t_ptr == t_filled) and t_ptr after.
        """
        gen = None if self.graph_safe_rng else self.gen
        n_starts = torch.div(self._t_filled - self.seq_len, self.period, rounding_mode="floor") + 1
        u_row = torch.rand(batch_size, device=self.device, generator=gen)
        u_s = torch.rand(batch_size, device=self.device, generator=gen)
        rows = (u_row * self.rows).long().clamp_(max=self.rows - 1)
        s = torch.minimum((u_s * n_starts.to(torch.float32)).long(), n_starts - 1) * self.period
        t0 = (self._t_ptr - self._t_filled + s) % self.t_max
        return rows, t0

    @torch.no_grad()
    def sample(self, batch_size: int) -> Dict[str, Tensor]:
        """Sample [batch, seq_len, ...] windows (+ '_rows'/'_t0' indices for
        priority writeback by prioritised subclasses)."""
        rows, t0 = self._window_starts(batch_size)
        offs = torch.arange(self.seq_len, device=self.device)
        tidx = (t0.unsqueeze(1) + offs.unsqueeze(0)) % self.t_max  # [batch, seq]
        out = {}
        for k, v in self.storage.items():
            out[k] = v[rows.unsqueeze(1), tidx]
        out["_rows"] = rows
        out["_t0"] = t0
        return out
