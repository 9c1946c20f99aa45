"""IMPALA-style visual ResNet torso and dense residual torso (parity with
/root/reference/stoix/networks/resnet.py: VisualResNetTorso :108-162,
ResNetTorso :165-188, downsampling :48-105)."""
from __future__ import annotations

from typing import List, Sequence

import torch
import torch.nn as nn

from stoix_amd.networks.torso import MLPTorso, orthogonal_init
from stoix_amd.networks.utils import get_activation

Tensor = torch.Tensor


class ResidualConvBlock(nn.Module):
    def __init__(self, channels: int, activation: str = "relu"):
        super().__init__()
        self.act = get_activation(activation)()
        self.conv1 = nn.Conv2d(channels, channels, 3, padding=1)
        self.conv2 = nn.Conv2d(channels, channels, 3, padding=1)

    def forward(self, x: Tensor) -> Tensor:
        y = self.conv1(self.act(x))
        y = self.conv2(self.act(y))
        return x + y


class VisualResNetTorso(nn.Module):
    """IMPALA ResNet: per stage conv + 3x3/2 max-pool + 2 residual blocks."""

    def __init__(
        self,
        input_shape: Sequence[int],  # (C, H, W)
        channels_per_group: Sequence[int] = (16, 32, 32),
        blocks_per_group: Sequence[int] = (2, 2, 2),
        mlp_sizes: Sequence[int] = (256,),
        activation: str = "relu",
        channel_first: bool = True,
    ):
        super().__init__()
        self.channel_first = channel_first
        c, h, w = input_shape
        stages: List[nn.Module] = []
        for out_c, n_blocks in zip(channels_per_group, blocks_per_group):
            stages.append(nn.Conv2d(c, out_c, 3, padding=1))
            stages.append(nn.MaxPool2d(3, stride=2, padding=1))
            for _ in range(n_blocks):
                stages.append(ResidualConvBlock(out_c, activation))
            c = out_c
            h = (h + 1) // 2
            w = (w + 1) // 2
        self.stages = nn.Sequential(*stages)
        self.act = get_activation(activation)()
        self.mlp = MLPTorso(c * h * w, mlp_sizes, activation=activation)
        self.output_dim = self.mlp.output_dim

    def forward(self, x: Tensor) -> Tensor:
        if not self.channel_first:
            x = x.permute(0, 3, 1, 2)
        if x.dtype == torch.uint8:
            x = x.to(torch.float32) / 255.0
        z = self.stages(x)
        return self.mlp(self.act(z).flatten(1))


class ResidualDenseBlock(nn.Module):
    def __init__(self, dim: int, activation: str = "relu", use_layer_norm: bool = True):
        super().__init__()
        self.act = get_activation(activation)()
        self.ln = nn.LayerNorm(dim) if use_layer_norm else nn.Identity()
        self.l1 = orthogonal_init(nn.Linear(dim, dim))
        self.l2 = orthogonal_init(nn.Linear(dim, dim))

    def forward(self, x: Tensor) -> Tensor:
        y = self.l1(self.act(self.ln(x)))
        y = self.l2(self.act(y))
        return x + y


class ResNetTorso(nn.Module):
    """Dense residual torso (reference resnet.py:165-188)."""

    def __init__(self, input_dim: int, hidden_dim: int = 256, num_blocks: int = 2, activation: str = "relu", use_layer_norm: bool = True):
        super().__init__()
        self.proj = orthogonal_init(nn.Linear(input_dim, hidden_dim))
        self.blocks = nn.Sequential(
            *[ResidualDenseBlock(hidden_dim, activation, use_layer_norm) for _ in range(num_blocks)]
        )
        self.output_dim = hidden_dim

    def forward(self, x: Tensor) -> Tensor:
        return self.blocks(self.proj(x))
