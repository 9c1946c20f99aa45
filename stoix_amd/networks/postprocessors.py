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
