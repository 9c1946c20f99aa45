"""HBM-resident uniform item replay buffer.

Functional parity with flashbax ``make_item_buffer`` as used by the
reference's off-policy systems (/root/reference/stoix/systems/q_learning/
ff_dqn.py:339-345, ddpg/ff_td3.py:465). Storage is a dict of preallocated
torch tensors on the training device — with 288 GB of HBM3E per MI355X the
whole 1M-transition buffer stays resident; adds and samples are pure device
ops (index copies / gathers), graph-capturable.
"""
from __future__ import annotations

from typing import Dict

import torch

Tensor = torch.Tensor


class ItemBuffer:
    """The write cursor and size live in DEVICE tensors so that add() and
    sample() are hip-graph-capturable: a captured add advances the cursor
    on-device every replay (a Python-int cursor would bake the capture-time
    slots into the graph and silently overwrite them forever)."""

    def __init__(self, capacity: int, device: torch.device | str = "cpu", seed: int = 0):
        self.capacity = int(capacity)
        self.device = torch.device(device)
        self.gen = torch.Generator(device=self.device)
        self.gen.manual_seed(seed)
        self.storage: Dict[str, Tensor] = {}
        self._ptr = torch.zeros(1, dtype=torch.int64, device=self.device)
        self._size = torch.zeros(1, dtype=torch.int64, device=self.device)
        # graph mode switches sampling to the default (graph-aware) RNG
        self.graph_safe_rng = False

    def _alloc(self, example: Dict[str, Tensor]) -> None:
        for k, v in example.items():
            self.storage[k] = torch.zeros(
                (self.capacity, *v.shape[1:]), dtype=v.dtype, device=self.device
            )

    @property
    def ptr(self) -> int:
        return int(self._ptr.item())

    @property
    def size(self) -> int:
        return int(self._size.item())

    @torch.no_grad()
    def add(self, batch: Dict[str, Tensor]) -> None:
        """Add a batch of items (leading dim = batch); pure device ops."""
        if not self.storage:
            self._alloc(batch)
        b = next(iter(batch.values())).shape[0]
        idx = (torch.arange(b, device=self.device) + self._ptr) % self.capacity
        for k, v in batch.items():
            self.storage[k][idx] = v.to(self.device)
        self._ptr.add_(b).remainder_(self.capacity)
        self._size.add_(b).clamp_(max=self.capacity)

    @torch.no_grad()
    def sample(self, batch_size: int) -> Dict[str, Tensor]:
        gen = None if self.graph_safe_rng else self.gen
        u = torch.rand(batch_size, device=self.device, generator=gen)
        # defensive bound against size-1 (not capacity-1): an index in
        # [size, capacity) would read unwritten zero slots pre-fill
        idx = torch.minimum(
            (u * self._size.to(torch.float32)).long(), self._size - 1
        )
        return {k: v[idx] for k, v in self.storage.items()}

    @property
    def can_sample(self) -> bool:
        return bool((self._size > 0).item())
