"""Network torsos (parity with /root/reference/stoix/networks/torso.py:
MLPTorso :12-33, NoisyMLPTorso :36-57, CNNTorso :60-108)."""
from __future__ import annotations

from typing import List, Sequence

import torch
import torch.nn as nn

from stoix_amd.networks.layers import NoisyLinear
from stoix_amd.networks.utils import get_activation


def orthogonal_init(layer: nn.Linear, scale: float = 2.0**0.5) -> nn.Linear:
    nn.init.orthogonal_(layer.weight, gain=scale)
    nn.init.zeros_(layer.bias)
    return layer


class MLPTorso(nn.Module):
    """MLP with orthogonal init and optional LayerNorm per hidden layer."""

    def __init__(
        self,
        input_dim: int,
        layer_sizes: Sequence[int] = (256, 256),
        activation: str = "silu",
        use_layer_norm: bool = False,
        init_scale: float = 2.0**0.5,
    ):
        super().__init__()
        self.output_dim = layer_sizes[-1] if layer_sizes else input_dim
        act = get_activation(activation)
        mods: List[nn.Module] = []
        in_d = input_dim
        for h in layer_sizes:
            mods.append(orthogonal_init(nn.Linear(in_d, h), init_scale))
            if use_layer_norm:
                mods.append(nn.LayerNorm(h))
            mods.append(act())
            in_d = h
        self._in_dim = input_dim
        self.net = nn.Sequential(*mods)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() > 2 and x.shape[-1] != self._in_dim:
            x = x.reshape(x.shape[0], -1)  # flatten grid/pixel observations
        return self.net(x)


class NoisyMLPTorso(nn.Module):
    """MLP of NoisyLinear layers (factorised Gaussian noise; Rainbow)."""

    def __init__(
        self,
        input_dim: int,
        layer_sizes: Sequence[int] = (256, 256),
        activation: str = "relu",
        use_layer_norm: bool = False,
        sigma_zero: float = 0.5,
    ):
        super().__init__()
        self.output_dim = layer_sizes[-1] if layer_sizes else input_dim
        act = get_activation(activation)
        mods: List[nn.Module] = []
        in_d = input_dim
        for h in layer_sizes:
            mods.append(NoisyLinear(in_d, h, sigma_zero=sigma_zero))
            if use_layer_norm:
                mods.append(nn.LayerNorm(h))
            mods.append(act())
            in_d = h
        self._in_dim = input_dim
        self.net = nn.Sequential(*mods)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() > 2 and x.shape[-1] != self._in_dim:
            x = x.reshape(x.shape[0], -1)
        return self.net(x)


class CNNTorso(nn.Module):
    """Conv stack + flatten + MLP (reference torso.py:60-108). Input is
    [B, C, H, W] (or [B, H, W, C] with channel_first=False)."""

    def __init__(
        self,
        input_shape: Sequence[int],  # (C, H, W)
        channel_sizes: Sequence[int] = (32, 64, 64),
        kernel_sizes: Sequence[int] = (8, 4, 3),
        strides: Sequence[int] = (4, 2, 1),
        mlp_sizes: Sequence[int] = (512,),
        activation: str = "relu",
        channel_first: bool = True,
    ):
        super().__init__()
        act = get_activation(activation)
        self.channel_first = channel_first
        c, h, w = input_shape
        convs: List[nn.Module] = []
        for out_c, k, s in zip(channel_sizes, kernel_sizes, strides):
            convs.append(nn.Conv2d(c, out_c, k, stride=s))
            convs.append(act())
            c = out_c
            h = (h - k) // s + 1
            w = (w - k) // s + 1
        self.convs = nn.Sequential(*convs)
        flat = c * h * w
        self.mlp = MLPTorso(flat, mlp_sizes, activation=activation)
        self.output_dim = self.mlp.output_dim

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not self.channel_first:
            x = x.permute(0, 3, 1, 2)
        z = self.convs(x / 255.0 if x.dtype == torch.uint8 else x)
        return self.mlp(z.flatten(1))
