"""Action postprocessors (parity with
/root/reference/stoix/networks/postprocessors.py:13-81): rescale / clip /
tanh-to-spec and DDPG-style exploration noise applied on top of a network's
output action or distribution."""
from __future__ import annotations

from typing import Optional

import torch

Tensor = torch.Tensor


def clip_to_spec(action: Tensor, minimum: float, maximum: float) -> Tensor:
    return action.clamp(minimum, maximum)


def rescale_to_spec(action: Tensor, minimum: float, maximum: float) -> Tensor:
    """Map action in [-1, 1] to [minimum, maximum]."""
    return (action + 1.0) * 0.5 * (maximum - minimum) + minimum


def tanh_to_spec(action: Tensor, minimum: float, maximum: float) -> Tensor:
    return rescale_to_spec(torch.tanh(action), minimum, maximum)


class ExplorationNoisePostProcessor:
    """Additive Gaussian exploration noise + clip (DDPG/TD3 acting;
    reference ff_td3.py:49-51)."""

    def __init__(self, sigma: float, minimum: float, maximum: float):
        self.sigma = sigma
        self.minimum = minimum
        self.maximum = maximum

    def __call__(self, action: Tensor, generator: Optional[torch.Generator] = None) -> Tensor:
        noise = torch.randn(action.shape, device=action.device, generator=generator) * self.sigma
        return (action + noise).clamp(self.minimum, self.maximum)


class PostProcessedDistribution:
    """Wrap a distribution so sampled/greedy actions pass through a
    postprocessor fn (reference postprocessors.py:13-30); log-prob/entropy
    delegate to the base distribution."""

    def __init__(self, base, postprocessor):
        self.base = base
        self.post = postprocessor

    def sample(self, generator: Optional[torch.Generator] = None) -> Tensor:
        return self.post(self.base.sample(generator))

    def mode(self) -> Tensor:
        return self.post(self.base.mode())

    def log_prob(self, value: Tensor) -> Tensor:
        return self.base.log_prob(value)

    def entropy(self, *a, **kw) -> Tensor:
        return self.base.entropy(*a, **kw)
