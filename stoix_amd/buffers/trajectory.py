"""HBM-resident trajectory (sequence) replay buffer.

Functional parity with flashbax ``make_trajectory_buffer`` as used by the
reference (/root/reference/stoix/systems/ddpg/ff_d4pg.py:477,
awr/ff_awr.py:431, mpo/ff_mpo.py:539-545): rows are environment streams,
time is circular per row; sampling draws contiguous windows of
``sample_sequence_length`` steps.
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
        self.t_ptr = 0
        self.t_filled = 0

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
        idx = (torch.arange(t_block, device=self.device) + self.t_ptr) % self.t_max
        for k, v in batch.items():
            self.storage[k][:, idx] = v.to(self.device)
        self.t_ptr = (self.t_ptr + t_block) % self.t_max
        self.t_filled = min(self.t_filled + t_block, self.t_max)

    @property
    def can_sample(self) -> bool:
        return self.t_filled >= self.seq_len

    def _window_starts(self, batch_size: int) -> tuple:
        """Sample (row, t0) pairs with valid contiguous windows.

        When the buffer has wrapped, windows crossing the write pointer mix
        old/new data; we sample starts in the contiguous valid region behind
        the pointer (standard flashbax behaviour of masking invalid items).
        """
        n_starts = (self.t_filled - self.seq_len) // self.period + 1
        rows = torch.randint(0, self.rows, (batch_size,), device=self.device, generator=self.gen)
        s = torch.randint(0, n_starts, (batch_size,), device=self.device, generator=self.gen) * self.period
        if self.t_filled == self.t_max:
            # oldest data starts at t_ptr
            t0 = (self.t_ptr + s) % self.t_max
        else:
            t0 = s
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
