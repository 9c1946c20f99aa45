"""Actor/critic containers and recurrent networks.

Parity with /root/reference/stoix/networks/base.py: FeedForwardActor/Critic
:18-59, CompositeNetwork :62-86, MultiNetwork :88-121, ScannedRNN :124-159,
RecurrentActor/Critic :162-222, chained_torsos :225-252.
"""
from __future__ import annotations

from typing import Any, List, Optional, Sequence, Tuple

import torch
import torch.nn as nn

from stoix_amd.networks.layers import StackedRNN

Tensor = torch.Tensor


class FeedForwardActor(nn.Module):
    """input-embed -> torso -> action head (returns a Distribution)."""

    def __init__(self, torso: nn.Module, action_head: nn.Module, input_layer: Optional[nn.Module] = None):
        super().__init__()
        self.input_layer = input_layer
        self.torso = torso
        self.action_head = action_head

    def forward(self, obs: Any):
        x = self.input_layer(obs) if self.input_layer is not None else obs
        return self.action_head(self.torso(x))


class FeedForwardCritic(nn.Module):
    """input-embed -> torso -> critic head (scalar / distributional)."""

    def __init__(self, torso: nn.Module, critic_head: nn.Module, input_layer: Optional[nn.Module] = None):
        super().__init__()
        self.input_layer = input_layer
        self.torso = torso
        self.critic_head = critic_head

    def forward(self, *inputs: Any):
        if self.input_layer is not None:
            x = self.input_layer(*inputs)
        else:
            x = inputs[0] if len(inputs) == 1 else torch.cat([i.flatten(1) for i in inputs], dim=-1)
        return self.critic_head(self.torso(x))


class CompositeNetwork(nn.Module):
    """Arbitrary input-layer/torso/head composition (reference base.py:62-86)."""

    def __init__(self, input_layer: nn.Module, torso: nn.Module, head: nn.Module):
        super().__init__()
        self.input_layer = input_layer
        self.torso = torso
        self.head = head

    def forward(self, *inputs: Any):
        return self.head(self.torso(self.input_layer(*inputs)))


class MultiNetwork(nn.Module):
    """N copies of a network evaluated on the same input, outputs stacked on
    a leading 'ensemble' dim (twin critics; reference base.py:88-121)."""

    def __init__(self, networks: Sequence[nn.Module]):
        super().__init__()
        self.networks = nn.ModuleList(networks)

    def forward(self, *inputs: Any) -> Tensor:
        outs = [net(*inputs) for net in self.networks]
        return torch.stack(outs, dim=0)


class ScannedRNN(nn.Module):
    """RNN scanned over time with per-step hidden reset on done
    (reference base.py:124-159). Input is time-major [T, B, D]; ``resets``
    [T, B] bool re-initialises the hidden state *before* consuming step t."""

    def __init__(self, input_dim: int, hidden_dim: int, cell_type: str = "gru", num_layers: int = 1):
        super().__init__()
        self.rnn = StackedRNN(input_dim, hidden_dim, num_layers=num_layers, cell_type=cell_type)
        self.hidden_dim = hidden_dim

    def initial_state(self, batch: int, device) -> list:
        return self.rnn.initial_state(batch, device)

    @staticmethod
    def _mask_state(state: Any, keep: Tensor) -> Any:
        # keep: [B] float (1 keep, 0 reset-to-zero)
        if isinstance(state, tuple):
            return tuple(ScannedRNN._mask_state(s, keep) for s in state)
        if isinstance(state, list):
            return [ScannedRNN._mask_state(s, keep) for s in state]
        return state * keep.unsqueeze(-1)

    def _single_cell(self):
        """The (cell, kind) pair when the stack is one GRU/LSTM cell —
        the fast-path eligible shapes."""
        from stoix_amd.networks.layers import GRUCell, LSTMCell

        if len(self.rnn.cells) != 1:
            return None, None
        c = self.rnn.cells[0]
        if isinstance(c, GRUCell):
            return c.cell, "gru"
        if isinstance(c, LSTMCell):
            return c.cell, "lstm"
        return None, None

    def _hip_scan(self, x: Tensor, resets: Tensor, state: list, cell, kind):
        """K13: fused HIP scan (ops/csrc/rnn.hip) — no-grad path (acting,
        evaluation, R2D2 burn-in)."""
        from stoix_amd import ops

        ext = ops.ext(required=True)
        T, B = x.shape[:2]
        H = self.hidden_dim
        xp = torch.nn.functional.linear(
            x.reshape(T * B, -1), cell.weight_ih, cell.bias_ih
        ).reshape(T, B, -1)
        whh16 = cell.weight_hh.to(torch.bfloat16)
        rs = resets.to(torch.uint8).contiguous()
        Hout = torch.empty(T, B, H, device=x.device)
        hT = torch.empty(B, H, device=x.device)
        empty = torch.zeros(0, device=x.device)
        if kind == "lstm":
            h0, c0 = state[0]
            cT = torch.empty(B, H, device=x.device)
            ext.rnn_scan(xp, whh16, cell.bias_hh, rs, h0.contiguous(),
                         c0.contiguous(), Hout, hT, cT, 1)
            return Hout, [(hT, cT)]
        h0 = state[0]
        ext.rnn_scan(xp, whh16, cell.bias_hh, rs, h0.contiguous(), empty,
                     Hout, hT, empty, 0)
        return Hout, [hT]

    def _hoisted_scan(self, x: Tensor, resets: Tensor, state: list, cell, kind):
        """Autograd training path with the input projection for ALL T
        hoisted into one GEMM (the cuDNN trick); only the [B,H]x[H,G*H]
        recurrent GEMM + gate math stay per-step. Matches
        nn.GRUCell/nn.LSTMCell math exactly."""
        T, B = x.shape[:2]
        H = self.hidden_dim
        xp = torch.nn.functional.linear(
            x.reshape(T * B, -1), cell.weight_ih, cell.bias_ih
        ).reshape(T, B, -1)
        outs = []
        if kind == "gru":
            h = state[0]
            for t in range(T):
                keep = (~resets[t].bool()).to(x.dtype).unsqueeze(-1)
                h = h * keep
                hg = torch.nn.functional.linear(h, cell.weight_hh, cell.bias_hh)
                xr, xz, xn = xp[t].chunk(3, -1)
                hr, hz, hn = hg.chunk(3, -1)
                r = torch.sigmoid(xr + hr)
                z = torch.sigmoid(xz + hz)
                n = torch.tanh(xn + r * hn)
                h = (1 - z) * n + z * h
                outs.append(h)
            return torch.stack(outs), [h]
        h, c = state[0]
        for t in range(T):
            keep = (~resets[t].bool()).to(x.dtype).unsqueeze(-1)
            h = h * keep
            c = c * keep
            hg = torch.nn.functional.linear(h, cell.weight_hh, cell.bias_hh)
            i, f, g, o = (xp[t] + hg).chunk(4, -1)
            c = torch.sigmoid(f) * c + torch.sigmoid(i) * torch.tanh(g)
            h = torch.sigmoid(o) * torch.tanh(c)
            outs.append(h)
        return torch.stack(outs), [(h, c)]

    def forward(self, x: Tensor, resets: Tensor, state: list) -> Tuple[Tensor, list]:
        cell, kind = self._single_cell()
        if cell is not None:
            if (
                not torch.is_grad_enabled()
                and x.is_cuda
                and self.hidden_dim in (128, 256)
                and x.dtype == torch.float32
            ):
                from stoix_amd import ops

                if ops.have_ext():
                    return self._hip_scan(x, resets, state, cell, kind)
            return self._hoisted_scan(x, resets, state, cell, kind)
        T = x.shape[0]
        outs = []
        for t in range(T):
            keep = (~resets[t].bool()).to(x.dtype)
            state = self._mask_state(state, keep)
            h, state = self.rnn(x[t], state)
            outs.append(h)
        return torch.stack(outs), state


class RecurrentActor(nn.Module):
    """pre-torso -> ScannedRNN -> post-torso -> head (reference base.py:162-192)."""

    def __init__(self, pre_torso: nn.Module, rnn: ScannedRNN, post_torso: nn.Module, action_head: nn.Module):
        super().__init__()
        self.pre_torso = pre_torso
        self.rnn = rnn
        self.post_torso = post_torso
        self.action_head = action_head

    def initial_state(self, batch: int, device) -> list:
        return self.rnn.initial_state(batch, device)

    def forward(self, obs: Tensor, resets: Tensor, state: list):
        # obs [T, B, D]; flatten time for the MLPs, keep for the RNN
        T, B = obs.shape[:2]
        z = self.pre_torso(obs.reshape(T * B, -1)).reshape(T, B, -1)
        h, state = self.rnn(z, resets, state)
        y = self.post_torso(h.reshape(T * B, -1)).reshape(T, B, -1)
        dist = self.action_head(y)
        return dist, state


class RecurrentCritic(nn.Module):
    def __init__(self, pre_torso: nn.Module, rnn: ScannedRNN, post_torso: nn.Module, critic_head: nn.Module):
        super().__init__()
        self.pre_torso = pre_torso
        self.rnn = rnn
        self.post_torso = post_torso
        self.critic_head = critic_head

    def initial_state(self, batch: int, device) -> list:
        return self.rnn.initial_state(batch, device)

    def forward(self, obs: Tensor, resets: Tensor, state: list):
        T, B = obs.shape[:2]
        z = self.pre_torso(obs.reshape(T * B, -1)).reshape(T, B, -1)
        h, state = self.rnn(z, resets, state)
        v = self.critic_head(self.post_torso(h.reshape(T * B, -1)))
        if isinstance(v, torch.Tensor):
            v = v.reshape(T, B)
        return v, state


class ChainedTorsos(nn.Module):
    """Sequential composition of torsos (reference base.py:225-252)."""

    def __init__(self, torsos: Sequence[nn.Module]):
        super().__init__()
        self.torsos = nn.ModuleList(torsos)
        self.output_dim = getattr(torsos[-1], "output_dim", None)

    def forward(self, x: Tensor) -> Tensor:
        for t in self.torsos:
            x = t(x)
        return x


class SharedPolicyValueNetwork(nn.Module):
    """One torso feeding a joint PolicyValueHead (IMPALA shared-torso /
    AZ-style nets; reference base.py + heads.py:220-232). forward returns
    ``(Distribution, value)``."""

    def __init__(self, torso: nn.Module, action_head: nn.Module):
        super().__init__()
        from stoix_amd.networks.heads import PolicyValueHead

        self.torso = torso
        self.head = PolicyValueHead(torso.output_dim, action_head)

    def forward(self, obs: Tensor):
        return self.head(self.torso(obs))


def chained_torsos(*torsos: nn.Module) -> nn.Module:
    """Compose torsos sequentially (reference base.py:225-252)."""

    class _Chained(nn.Module):
        def __init__(self):
            super().__init__()
            self.torsos = nn.ModuleList(torsos)
            self.output_dim = getattr(torsos[-1], "output_dim", None)

        def forward(self, x: Tensor) -> Tensor:
            for t in self.torsos:
                x = t(x)
            return x

    return _Chained()
