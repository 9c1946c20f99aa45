"""Dueling Q-networks (parity with /root/reference/stoix/networks/dueling.py:
DuelingQNetwork :15-47, DistributionalDuelingQNetwork :50-87,
NoisyDistributionalDuelingQNetwork :90-124)."""
from __future__ import annotations

from typing import Sequence

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.networks.heads import C51Output
from stoix_amd.networks.distributions import Categorical, EpsilonGreedy
from stoix_amd.networks.torso import MLPTorso, NoisyMLPTorso

Tensor = torch.Tensor


class DuelingQNetwork(nn.Module):
    """Q(s,a) = V(s) + A(s,a) - mean_a A(s,a)."""

    def __init__(self, input_dim: int, num_actions: int, layer_sizes: Sequence[int] = (256,), activation: str = "relu", epsilon: float = 0.1):
        super().__init__()
        self.value = MLPTorso(input_dim, (*layer_sizes, 1), activation=activation)
        self.advantage = MLPTorso(input_dim, (*layer_sizes, num_actions), activation=activation)
        # strip final activation: use raw linear outputs
        self.value.net = self.value.net[:-1]
        self.advantage.net = self.advantage.net[:-1]
        self.epsilon = epsilon

    def q_values(self, x: Tensor) -> Tensor:
        if x.dim() > 2:
            x = x.reshape(x.shape[0], -1)
        v = self.value(x)
        a = self.advantage(x)
        return v + a - a.mean(dim=-1, keepdim=True)

    def forward(self, x: Tensor) -> EpsilonGreedy:
        return EpsilonGreedy(self.q_values(x), self.epsilon)


class DistributionalDuelingQNetwork(nn.Module):
    """C51 + dueling over atom logits (reference dueling.py:50-87)."""

    def __init__(
        self,
        input_dim: int,
        num_actions: int,
        num_atoms: int = 51,
        vmin: float = -200.0,
        vmax: float = 200.0,
        layer_sizes: Sequence[int] = (256,),
        activation: str = "relu",
        noisy: bool = False,
        sigma_zero: float = 0.5,
    ):
        super().__init__()
        Torso = NoisyMLPTorso if noisy else MLPTorso
        kwargs = {"sigma_zero": sigma_zero} if noisy else {}
        self.value = Torso(input_dim, (*layer_sizes, num_atoms), activation=activation, **kwargs)
        self.advantage = Torso(input_dim, (*layer_sizes, num_actions * num_atoms), activation=activation, **kwargs)
        self.value.net = self.value.net[:-1]
        self.advantage.net = self.advantage.net[:-1]
        self.num_actions = num_actions
        self.num_atoms = num_atoms
        self.register_buffer("atoms", torch.linspace(vmin, vmax, num_atoms))

    def forward(self, x: Tensor) -> C51Output:
        if x.dim() > 2:
            x = x.reshape(x.shape[0], -1)  # flatten grid/pixel observations
        v = self.value(x).view(*x.shape[:-1], 1, self.num_atoms)
        a = self.advantage(x).view(*x.shape[:-1], self.num_actions, self.num_atoms)
        logits = v + a - a.mean(dim=-2, keepdim=True)
        probs = F.softmax(logits, dim=-1)
        q_values = (probs * self.atoms).sum(-1)
        return C51Output(Categorical(logits=q_values), logits, self.atoms, q_values)


def NoisyDistributionalDuelingQNetwork(*args, **kwargs) -> DistributionalDuelingQNetwork:
    """Rainbow network (reference dueling.py:90-124)."""
    kwargs["noisy"] = True
    return DistributionalDuelingQNetwork(*args, **kwargs)
