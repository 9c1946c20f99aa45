"""Observation / action spaces.

Equivalent surface to the reference's external `stoa` spaces as consumed at
/root/reference/stoix/systems/ppo/anakin/ff_ppo.py:432,466 and
ff_sac.py:354-359: ``num_values``, ``shape``, ``minimum``, ``maximum`` and
``generate_value()`` (used for network init).
"""
from __future__ import annotations

from typing import Optional, Sequence, Tuple

import torch


class Space:
    shape: Tuple[int, ...]
    dtype: torch.dtype

    def generate_value(self) -> torch.Tensor:
        raise NotImplementedError

    def sample(self, batch: int, device: torch.device, generator: Optional[torch.Generator] = None) -> torch.Tensor:
        raise NotImplementedError


class DiscreteSpace(Space):
    """A single categorical action / observation with ``num_values`` values."""

    def __init__(self, num_values: int, dtype: torch.dtype = torch.long):
        self.num_values = int(num_values)
        self.shape = ()
        self.dtype = dtype

    def generate_value(self) -> torch.Tensor:
        return torch.zeros((), dtype=self.dtype)

    def sample(self, batch, device, generator=None):
        return torch.randint(0, self.num_values, (batch,), device=device, generator=generator)

    def __repr__(self):
        return f"DiscreteSpace({self.num_values})"


class MultiDiscreteSpace(Space):
    def __init__(self, num_values: Sequence[int], dtype: torch.dtype = torch.long):
        self.num_values_list = [int(n) for n in num_values]
        self.shape = (len(self.num_values_list),)
        self.dtype = dtype

    def generate_value(self) -> torch.Tensor:
        return torch.zeros(self.shape, dtype=self.dtype)

    def sample(self, batch, device, generator=None):
        cols = [torch.randint(0, n, (batch,), device=device, generator=generator) for n in self.num_values_list]
        return torch.stack(cols, dim=-1)

    def __repr__(self):
        return f"MultiDiscreteSpace({self.num_values_list})"


class BoxSpace(Space):
    """Bounded continuous space."""

    def __init__(
        self,
        shape: Sequence[int],
        minimum: float | torch.Tensor = -1.0,
        maximum: float | torch.Tensor = 1.0,
        dtype: torch.dtype = torch.float32,
    ):
        self.shape = tuple(int(s) for s in shape)
        self.dtype = dtype
        self.minimum = torch.as_tensor(minimum, dtype=dtype).expand(self.shape).clone() if self.shape else torch.as_tensor(minimum, dtype=dtype)
        self.maximum = torch.as_tensor(maximum, dtype=dtype).expand(self.shape).clone() if self.shape else torch.as_tensor(maximum, dtype=dtype)

    def generate_value(self) -> torch.Tensor:
        return torch.zeros(self.shape, dtype=self.dtype)

    def sample(self, batch, device, generator=None):
        lo = self.minimum.to(device)
        hi = self.maximum.to(device)
        u = torch.rand((batch, *self.shape), device=device, generator=generator)
        lo_f = torch.where(torch.isfinite(lo), lo, torch.full_like(lo, -1.0))
        hi_f = torch.where(torch.isfinite(hi), hi, torch.full_like(hi, 1.0))
        return lo_f + u * (hi_f - lo_f)

    def __repr__(self):
        return f"BoxSpace(shape={self.shape})"


def action_dim(space: Space) -> int:
    """Flat action dimension for continuous, num_values for discrete."""
    if isinstance(space, DiscreteSpace):
        return space.num_values
    if isinstance(space, MultiDiscreteSpace):
        return len(space.num_values_list)
    total = 1
    for s in space.shape:
        total *= s
    return total
