"""Running observation statistics (Welford) with cross-rank reduction.

Parity with /root/reference/stoix/utils/running_statistics.py (init :100-135,
update :205-330 with psum over pmap axes at :62-70, normalize/denormalize/clip
:333-...). The cross-device ``jax.lax.psum`` becomes a
``torch.distributed.all_reduce`` on the batch partials (one fused call for
count/sum/sum-sq), overlapped nowhere — it is tiny and once per rollout.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist

Tensor = torch.Tensor


@dataclass
class RunningStatisticsState:
    count: Tensor  # scalar
    mean: Tensor  # [*obs_shape]
    summed_variance: Tensor  # [*obs_shape] (M2 in Welford terms)
    std: Tensor  # [*obs_shape]


def init_state(shape, device="cpu", dtype=torch.float32) -> RunningStatisticsState:
    return RunningStatisticsState(
        count=torch.zeros((), device=device, dtype=dtype),
        mean=torch.zeros(shape, device=device, dtype=dtype),
        summed_variance=torch.zeros(shape, device=device, dtype=dtype),
        std=torch.ones(shape, device=device, dtype=dtype),
    )


@torch.no_grad()
def update(
    state: RunningStatisticsState,
    batch: Tensor,
    std_min_value: float = 1e-6,
    std_max_value: float = 1e6,
    all_reduce: bool = False,
) -> RunningStatisticsState:
    """Welford batched update. ``batch`` is [..., *obs_shape] with every
    leading dim treated as batch. With ``all_reduce`` and an initialised
    process group, partials are summed across ranks (reference
    running_statistics.py:62-70, 297-310)."""
    obs_ndim = state.mean.dim()
    flat = batch.reshape(-1, *state.mean.shape) if obs_ndim else batch.reshape(-1)
    # torch.full (device fill kernel), NOT torch.tensor(scalar, device=...):
    # the latter is a pageable H2D copy — illegal inside hip-graph capture
    # (the fused obs-norm path runs this inside the rollout graph)
    n = torch.full((), float(flat.shape[0]), device=batch.device, dtype=state.mean.dtype)
    s = flat.sum(dim=0)
    if all_reduce and dist.is_initialized() and dist.get_world_size() > 1:
        packed = torch.cat([n.reshape(1), s.reshape(-1)])
        dist.all_reduce(packed)
        n = packed[0]
        s = packed[1:].reshape(state.mean.shape)
    new_count = state.count + n
    batch_mean_global = s / n
    diff_to_old = batch_mean_global - state.mean
    new_mean = state.mean + diff_to_old * (n / new_count)
    # M2 update: sum over batch of (x - new_mean)*(x - old_mean)
    m2_local = ((flat - new_mean) * (flat - state.mean)).sum(dim=0)
    if all_reduce and dist.is_initialized() and dist.get_world_size() > 1:
        m2 = m2_local.reshape(-1).clone()
        dist.all_reduce(m2)
        m2_local = m2.reshape(state.mean.shape)
    new_m2 = state.summed_variance + m2_local
    std = torch.sqrt(torch.clamp(new_m2 / torch.clamp(new_count, min=1.0), min=0.0))
    std = std.clamp(std_min_value, std_max_value)
    return RunningStatisticsState(count=new_count, mean=new_mean, summed_variance=new_m2, std=std)


@torch.no_grad()
def update_(
    state: RunningStatisticsState,
    batch: Tensor,
    std_min_value: float = 1e-6,
    std_max_value: float = 1e6,
    all_reduce: bool = False,
) -> RunningStatisticsState:
    """In-place Welford update: writes into the state's STABLE tensors so
    kernels (and hip graphs) holding their addresses see fresh statistics
    without rebinding (fused PPO obs-norm path)."""
    new = update(state, batch, std_min_value, std_max_value, all_reduce)
    state.count.copy_(new.count)
    state.mean.copy_(new.mean)
    state.summed_variance.copy_(new.summed_variance)
    state.std.copy_(new.std)
    return state


def normalize(batch: Tensor, state: RunningStatisticsState, max_abs_value: Optional[float] = None) -> Tensor:
    out = (batch - state.mean) / state.std
    if max_abs_value is not None:
        out = out.clamp(-max_abs_value, max_abs_value)
    return out


def denormalize(batch: Tensor, state: RunningStatisticsState) -> Tensor:
    return batch * state.std + state.mean
