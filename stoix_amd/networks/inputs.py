"""Input-embedding layers (parity with
/root/reference/stoix/networks/inputs.py:7-45)."""
from __future__ import annotations

import torch
import torch.nn as nn

Tensor = torch.Tensor


class ArrayInput(nn.Module):
    """Identity on a flat observation tensor."""

    def forward(self, obs: Tensor) -> Tensor:
        return obs


class EmbeddingActionInput(nn.Module):
    """Concatenate observation and continuous action: Q(s, a) input
    (DDPG/TD3/SAC critics)."""

    def forward(self, obs: Tensor, action: Tensor) -> Tensor:
        return torch.cat([obs, action], dim=-1)


class EmbeddingActionOnehotInput(nn.Module):
    """Concatenate observation and one-hot discrete action."""

    def __init__(self, num_actions: int):
        super().__init__()
        self.num_actions = num_actions

    def forward(self, obs: Tensor, action: Tensor) -> Tensor:
        onehot = torch.nn.functional.one_hot(action.long(), self.num_actions).to(obs.dtype)
        return torch.cat([obs, onehot], dim=-1)


class FeatureInput(nn.Module):
    """Select one named feature from a dict observation (reference
    inputs.py FeatureInput: structured-observation envs expose a dict; the
    network consumes a single flat field)."""

    def __init__(self, feature: str = "obs"):
        super().__init__()
        self.feature = feature

    def forward(self, obs) -> Tensor:
        x = obs[self.feature] if isinstance(obs, dict) else obs
        return x.flatten(1) if x.dim() > 2 else x
