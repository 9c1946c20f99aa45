"""Network heads returning distributions or values.

Parity with /root/reference/stoix/networks/heads.py: CategoricalHead :30-41,
NormalAffineTanhDistributionHead :44-65, BetaDistributionHead :68-98,
MultivariateNormalDiagHead :101-114, DeterministicHead :117-126,
ScalarCriticHead :129-134, CategoricalCriticHead/DiscreteValuedTfpHead
:137-199, DiscreteQNetworkHead :202-217, PolicyValueHead :220-232,
DistributionalDiscreteQNetwork (C51) :235-256, DistributionalContinuousQNetwork
(D4PG) :259-274, QuantileDiscreteQNetwork :277-291, LinearHead :294-311,
MultiDiscreteHead :314-339.
"""
from __future__ import annotations

from typing import NamedTuple, Sequence, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.networks.distributions import (
    AffineTanhTransformedDistribution,
    Categorical,
    ClippedBeta,
    DiscreteValuedDistribution,
    EpsilonGreedy,
    MultiDiscreteDistribution,
    MultivariateNormalDiag,
)
from stoix_amd.networks.torso import orthogonal_init

Tensor = torch.Tensor


class CategoricalHead(nn.Module):
    def __init__(self, input_dim: int, num_actions: int):
        super().__init__()
        self.linear = orthogonal_init(nn.Linear(input_dim, num_actions), scale=0.01)

    def forward(self, x: Tensor) -> Categorical:
        return Categorical(self.linear(x))


class NormalAffineTanhDistributionHead(nn.Module):
    """tanh-normal mapped to [minimum, maximum] (continuous PPO/SAC)."""

    def __init__(
        self,
        input_dim: int,
        action_dim: int,
        minimum: float,
        maximum: float,
        min_scale: float = 1e-3,
    ):
        super().__init__()
        self.loc = orthogonal_init(nn.Linear(input_dim, action_dim), scale=0.01)
        self.scale = orthogonal_init(nn.Linear(input_dim, action_dim), scale=0.01)
        self.minimum = minimum
        self.maximum = maximum
        self.min_scale = min_scale

    def forward(self, x: Tensor) -> AffineTanhTransformedDistribution:
        loc = self.loc(x)
        scale = F.softplus(self.scale(x)) + self.min_scale
        return AffineTanhTransformedDistribution(loc, scale, self.minimum, self.maximum)


class BetaDistributionHead(nn.Module):
    def __init__(self, input_dim: int, action_dim: int, minimum: float, maximum: float):
        super().__init__()
        self.alpha = orthogonal_init(nn.Linear(input_dim, action_dim), scale=0.01)
        self.beta = orthogonal_init(nn.Linear(input_dim, action_dim), scale=0.01)
        self.minimum = minimum
        self.maximum = maximum

    def forward(self, x: Tensor) -> ClippedBeta:
        a = F.softplus(self.alpha(x)) + 1.0
        b = F.softplus(self.beta(x)) + 1.0
        return ClippedBeta(a, b, self.minimum, self.maximum)


class MultivariateNormalDiagHead(nn.Module):
    def __init__(self, input_dim: int, action_dim: int, init_scale: float = 0.3, min_scale: float = 1e-6):
        super().__init__()
        self.loc = orthogonal_init(nn.Linear(input_dim, action_dim), scale=0.01)
        self.scale = orthogonal_init(nn.Linear(input_dim, action_dim), scale=0.01)
        self.init_scale = init_scale
        self.min_scale = min_scale

    def forward(self, x: Tensor) -> MultivariateNormalDiag:
        loc = self.loc(x)
        scale = self.init_scale * F.softplus(self.scale(x)) / F.softplus(torch.zeros((), device=x.device))
        return MultivariateNormalDiag(loc, scale + self.min_scale)


class DeterministicHead(nn.Module):
    """DDPG/TD3 deterministic action (tanh-bounded to the action range)."""

    def __init__(self, input_dim: int, action_dim: int, minimum: float = -1.0, maximum: float = 1.0):
        super().__init__()
        self.linear = orthogonal_init(nn.Linear(input_dim, action_dim), scale=0.01)
        self.minimum = minimum
        self.maximum = maximum

    def forward(self, x: Tensor) -> Tensor:
        y = torch.tanh(self.linear(x))
        return y * (self.maximum - self.minimum) / 2.0 + (self.maximum + self.minimum) / 2.0


class ScalarCriticHead(nn.Module):
    def __init__(self, input_dim: int):
        super().__init__()
        self.linear = orthogonal_init(nn.Linear(input_dim, 1), scale=1.0)

    def forward(self, x: Tensor) -> Tensor:
        return self.linear(x).squeeze(-1)


class CategoricalCriticHead(nn.Module):
    """Critic over a discrete value support (returns DiscreteValuedDistribution)."""

    def __init__(self, input_dim: int, vmin: float = -300.0, vmax: float = 300.0, num_atoms: int = 601):
        super().__init__()
        self.linear = orthogonal_init(nn.Linear(input_dim, num_atoms), scale=0.01)
        self.register_buffer("atoms", torch.linspace(vmin, vmax, num_atoms))

    def forward(self, x: Tensor) -> DiscreteValuedDistribution:
        return DiscreteValuedDistribution(self.linear(x), self.atoms)


class DiscreteQNetworkHead(nn.Module):
    """Q-values + epsilon-greedy distribution (reference heads.py:202-217)."""

    def __init__(self, input_dim: int, num_actions: int, epsilon: float = 0.1):
        super().__init__()
        self.linear = orthogonal_init(nn.Linear(input_dim, num_actions), scale=1.0)
        self.epsilon = epsilon

    def forward(self, x: Tensor) -> EpsilonGreedy:
        return EpsilonGreedy(self.linear(x), self.epsilon)

    def q_values(self, x: Tensor) -> Tensor:
        return self.linear(x)


class PolicyValueHead(nn.Module):
    """Joint policy + value head for shared-torso nets (IMPALA/AZ)."""

    def __init__(self, input_dim: int, action_head: nn.Module):
        super().__init__()
        self.action_head = action_head
        self.value = orthogonal_init(nn.Linear(input_dim, 1), scale=1.0)

    def forward(self, x: Tensor):
        return self.action_head(x), self.value(x).squeeze(-1)


class C51Output(NamedTuple):
    q_dist: Categorical  # epsilon-greedy-compatible action dist goes via q_values
    q_logits: Tensor  # [B, A, num_atoms]
    atoms: Tensor  # [num_atoms]
    q_values: Tensor  # [B, A]


class DistributionalDiscreteQNetworkHead(nn.Module):
    """C51 head (reference heads.py:235-256)."""

    def __init__(self, input_dim: int, num_actions: int, num_atoms: int = 51, vmin: float = -200.0, vmax: float = 200.0, epsilon: float = 0.1):
        super().__init__()
        self.num_actions = num_actions
        self.num_atoms = num_atoms
        self.linear = orthogonal_init(nn.Linear(input_dim, num_actions * num_atoms), scale=0.1)
        self.register_buffer("atoms", torch.linspace(vmin, vmax, num_atoms))
        self.epsilon = epsilon

    def forward(self, x: Tensor) -> C51Output:
        logits = self.linear(x).view(*x.shape[:-1], self.num_actions, self.num_atoms)
        probs = F.softmax(logits, dim=-1)
        q_values = (probs * self.atoms).sum(-1)
        return C51Output(Categorical(logits=q_values), logits, self.atoms, q_values)

    def act_dist(self, x: Tensor) -> EpsilonGreedy:
        return EpsilonGreedy(self.forward(x).q_values, self.epsilon)


class D4PGOutput(NamedTuple):
    value: Tensor
    logits: Tensor
    atoms: Tensor


class DistributionalContinuousQNetworkHead(nn.Module):
    """D4PG critic head returning (value, logits, atoms) (heads.py:259-274)."""

    def __init__(self, input_dim: int, num_atoms: int = 51, vmin: float = -150.0, vmax: float = 150.0):
        super().__init__()
        self.linear = orthogonal_init(nn.Linear(input_dim, num_atoms), scale=0.1)
        self.register_buffer("atoms", torch.linspace(vmin, vmax, num_atoms))

    def forward(self, x: Tensor) -> D4PGOutput:
        logits = self.linear(x)
        probs = F.softmax(logits, dim=-1)
        value = (probs * self.atoms).sum(-1)
        return D4PGOutput(value, logits, self.atoms)


class QuantileOutput(NamedTuple):
    q_dist: Tensor  # [B, num_quantiles, A]
    q_values: Tensor  # [B, A]
    taus: Tensor  # [num_quantiles]


class QuantileDiscreteQNetworkHead(nn.Module):
    """QR-DQN head (heads.py:277-291)."""

    def __init__(self, input_dim: int, num_actions: int, num_quantiles: int = 200, epsilon: float = 0.1):
        super().__init__()
        self.num_actions = num_actions
        self.num_quantiles = num_quantiles
        self.linear = orthogonal_init(nn.Linear(input_dim, num_actions * num_quantiles), scale=0.1)
        taus = (torch.arange(num_quantiles, dtype=torch.float32) + 0.5) / num_quantiles
        self.register_buffer("taus", taus)
        self.epsilon = epsilon

    def forward(self, x: Tensor) -> QuantileOutput:
        dist = self.linear(x).view(*x.shape[:-1], self.num_quantiles, self.num_actions)
        q_values = dist.mean(dim=-2)
        return QuantileOutput(dist, q_values, self.taus)


class LinearHead(nn.Module):
    def __init__(self, input_dim: int, output_dim: int, scale: float = 1.0):
        super().__init__()
        self.linear = orthogonal_init(nn.Linear(input_dim, output_dim), scale=scale)

    def forward(self, x: Tensor) -> Tensor:
        return self.linear(x)


class MultiDiscreteHead(nn.Module):
    def __init__(self, input_dim: int, num_values: Sequence[int]):
        super().__init__()
        self.num_values = list(num_values)
        self.linear = orthogonal_init(nn.Linear(input_dim, sum(self.num_values)), scale=0.01)

    def forward(self, x: Tensor) -> MultiDiscreteDistribution:
        return MultiDiscreteDistribution(self.linear(x), self.num_values)
