"""Special layers: NoisyLinear, RNN cells, StackedRNN.

Parity with /root/reference/stoix/networks/layers.py (StackedRNN :16-68,
NoisyLinear factorised-Gaussian :71-169) and the rnn-cell registry
(networks/utils.py:7-37: lstm, gru, mgu, simple).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.networks.utils import register_rnn_cell

Tensor = torch.Tensor


class NoisyLinear(nn.Module):
    """Factorised Gaussian noisy linear layer (Fortunato et al. 2018).

    y = (mu_w + sigma_w * eps_w) x + (mu_b + sigma_b * eps_b), with
    eps_w = f(eps_out) f(eps_in)^T, f(x) = sign(x) sqrt(|x|). Noise is
    resampled by ``resample_noise()`` — the caller controls when (the
    reference threads an explicit 'noise' rng into each apply,
    ff_rainbow.py:176-186).
    """

    def __init__(self, in_features: int, out_features: int, sigma_zero: float = 0.5):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        bound = 1.0 / math.sqrt(in_features)
        self.weight_mu = nn.Parameter(torch.empty(out_features, in_features).uniform_(-bound, bound))
        self.bias_mu = nn.Parameter(torch.empty(out_features).uniform_(-bound, bound))
        sigma_init = sigma_zero / math.sqrt(in_features)
        self.weight_sigma = nn.Parameter(torch.full((out_features, in_features), sigma_init))
        self.bias_sigma = nn.Parameter(torch.full((out_features,), sigma_init))
        self.register_buffer("eps_in", torch.zeros(in_features))
        self.register_buffer("eps_out", torch.zeros(out_features))
        self.use_noise = True

    @staticmethod
    def _f(x: Tensor) -> Tensor:
        return x.sign() * x.abs().sqrt()

    def resample_noise(self, generator: Optional[torch.Generator] = None) -> None:
        self.eps_in = self._f(torch.randn(self.in_features, device=self.weight_mu.device, generator=generator))
        self.eps_out = self._f(torch.randn(self.out_features, device=self.weight_mu.device, generator=generator))

    def forward(self, x: Tensor) -> Tensor:
        if not self.use_noise:
            return F.linear(x, self.weight_mu, self.bias_mu)
        if self._mat is not None:
            return F.linear(x, self._mat[0], self._mat[1])
        w = self.weight_mu + self.weight_sigma * torch.outer(self.eps_out, self.eps_in)
        b = self.bias_mu + self.bias_sigma * self.eps_out
        return F.linear(x, w, b)

    _mat = None

    def materialize(self) -> None:
        """Build the noisy weight/bias ONCE for the current eps draw; every
        forward until the next materialize/resample reuses them (saves the
        outer-product + 2 muls + 2 adds per apply; with 2-3 applies per
        epoch that's most of the noisy-layer elementwise traffic).
        Gradients still flow to mu/sigma through the cached tensors."""
        self._mat = (
            self.weight_mu + self.weight_sigma * torch.outer(self.eps_out, self.eps_in),
            self.bias_mu + self.bias_sigma * self.eps_out,
        )


def resample_all_noise(module: nn.Module, generator: Optional[torch.Generator] = None) -> None:
    for m in module.modules():
        if isinstance(m, NoisyLinear):
            m.resample_noise(generator)


def set_noise_enabled(module: nn.Module, enabled: bool) -> None:
    for m in module.modules():
        if isinstance(m, NoisyLinear):
            m.use_noise = enabled


class NoiseBank:
    """Fused noise resampling for every NoisyLinear in a module.

    The per-layer resample path costs 4 kernels per eps tensor (randn,
    sign, abs-sqrt, mul) x 2 tensors x layers x applies — hundreds of
    ~4 us launches per Rainbow update (measured: sign/sqrt alone were ~2k
    calls per 13 updates, profiles/r02_rainbow). The bank points every
    layer's eps_in/eps_out at views of ONE flat buffer and resamples with
    3 kernels total (in-place normal_, then f(x)=sign(x)sqrt(|x|) fused as
    two in-place ops). In-place writes keep the views valid, so this is
    also hip-graph capture-friendly (stable addresses)."""

    def __init__(self, module: nn.Module):
        self.layers = [m for m in module.modules() if isinstance(m, NoisyLinear)]
        sizes = []
        for m in self.layers:
            sizes.append(m.in_features)
            sizes.append(m.out_features)
        total = sum(sizes)
        dev = self.layers[0].weight_mu.device if self.layers else torch.device("cpu")
        self.flat = torch.zeros(total, device=dev)
        off = 0
        for m in self.layers:
            m.eps_in = self.flat[off : off + m.in_features]
            off += m.in_features
            m.eps_out = self.flat[off : off + m.out_features]
            off += m.out_features

    def resample(self, generator: Optional[torch.Generator] = None) -> None:
        if self.flat.numel() == 0:
            return
        self.flat.normal_(generator=generator)
        s = self.flat.sign()
        self.flat.abs_().sqrt_().mul_(s)
        for m in self.layers:
            m.materialize()


# ------------------------------------------------------------------ RNN cells


@register_rnn_cell("lstm")
@register_rnn_cell("optimised_lstm")
class LSTMCell(nn.Module):
    def __init__(self, input_dim: int, hidden_dim: int):
        super().__init__()
        self.cell = nn.LSTMCell(input_dim, hidden_dim)
        self.hidden_dim = hidden_dim

    def initial_state(self, batch: int, device) -> Tuple[Tensor, Tensor]:
        z = torch.zeros(batch, self.hidden_dim, device=device)
        return (z, z.clone())

    def forward(self, x: Tensor, state: Tuple[Tensor, Tensor]) -> Tuple[Tensor, Tuple[Tensor, Tensor]]:
        h, c = self.cell(x, state)
        return h, (h, c)


@register_rnn_cell("gru")
class GRUCell(nn.Module):
    def __init__(self, input_dim: int, hidden_dim: int):
        super().__init__()
        self.cell = nn.GRUCell(input_dim, hidden_dim)
        self.hidden_dim = hidden_dim

    def initial_state(self, batch: int, device) -> Tensor:
        return torch.zeros(batch, self.hidden_dim, device=device)

    def forward(self, x: Tensor, state: Tensor) -> Tuple[Tensor, Tensor]:
        h = self.cell(x, state)
        return h, h


@register_rnn_cell("mgu")
class MGUCell(nn.Module):
    """Minimal gated unit (Zhou et al. 2016)."""

    def __init__(self, input_dim: int, hidden_dim: int):
        super().__init__()
        self.hidden_dim = hidden_dim
        self.wf = nn.Linear(input_dim + hidden_dim, hidden_dim)
        self.wh = nn.Linear(input_dim + hidden_dim, hidden_dim)

    def initial_state(self, batch: int, device) -> Tensor:
        return torch.zeros(batch, self.hidden_dim, device=device)

    def forward(self, x: Tensor, state: Tensor) -> Tuple[Tensor, Tensor]:
        f = torch.sigmoid(self.wf(torch.cat([x, state], -1)))
        h_tilde = torch.tanh(self.wh(torch.cat([x, f * state], -1)))
        h = (1 - f) * state + f * h_tilde
        return h, h


@register_rnn_cell("simple")
class SimpleCell(nn.Module):
    def __init__(self, input_dim: int, hidden_dim: int):
        super().__init__()
        self.hidden_dim = hidden_dim
        self.w = nn.Linear(input_dim + hidden_dim, hidden_dim)

    def initial_state(self, batch: int, device) -> Tensor:
        return torch.zeros(batch, self.hidden_dim, device=device)

    def forward(self, x: Tensor, state: Tensor) -> Tuple[Tensor, Tensor]:
        h = torch.tanh(self.w(torch.cat([x, state], -1)))
        return h, h


class StackedRNN(nn.Module):
    """Stack of RNN cells applied per step (reference layers.py:16-68)."""

    def __init__(self, input_dim: int, hidden_dim: int, num_layers: int = 1, cell_type: str = "lstm"):
        super().__init__()
        from stoix_amd.networks.utils import get_rnn_cell

        cls = get_rnn_cell(cell_type)
        self.cells = nn.ModuleList()
        d = input_dim
        for _ in range(num_layers):
            self.cells.append(cls(d, hidden_dim))
            d = hidden_dim
        self.hidden_dim = hidden_dim

    def initial_state(self, batch: int, device) -> list:
        return [c.initial_state(batch, device) for c in self.cells]

    def forward(self, x: Tensor, states: list) -> Tuple[Tensor, list]:
        new_states = []
        h = x
        for cell, st in zip(self.cells, states):
            h, ns = cell(h, st)
            new_states.append(ns)
        return h, new_states
