"""Specialised networks: the DisCo-103 agent net.

Parity with /root/reference/stoix/networks/specialised/disco103.py:
``LSTMActionConditionedTorso`` (root embedding -> one LSTM transition per
action in parallel, :13-110) and ``DiscoAgentNetwork`` (shared torso + five
heads: logits / q / y / z / aux_pi, :113-152). Torch re-design: the
all-actions LSTM transition is ONE batched LSTMCell call over [B*A] rows.
"""
from __future__ import annotations

from typing import NamedTuple, Sequence

import torch
import torch.nn as nn

Tensor = torch.Tensor


class DiscoAgentOutput(NamedTuple):
    """Reference systems/disco_rl/disco_rl_types.py:11-19."""

    logits: Tensor  # [B, A] policy
    q: Tensor  # [B, A, num_bins] categorical action values
    y: Tensor  # [B, pred_size] auxiliary prediction
    z: Tensor  # [B, A, pred_size] action-conditional auxiliary prediction
    aux_pi: Tensor  # [B, A] auxiliary policy


class LSTMActionConditionedTorso(nn.Module):
    """Root embedding from the shared-torso output, then one LSTM
    transition for every action in parallel (Muesli/MuZero-style model
    step; reference disco103.py:13-110)."""

    def __init__(self, input_dim: int, num_actions: int, lstm_size: int,
                 root_mlp_sizes: Sequence[int] = ()):
        super().__init__()
        self.num_actions = num_actions
        self.lstm_size = lstm_size
        layers: list[nn.Module] = []
        d = input_dim
        for s in root_mlp_sizes:
            layers += [nn.Linear(d, s), nn.ReLU()]
            d = s
        self.root_mlp = nn.Sequential(*layers)
        self.root_cell = nn.Linear(d, lstm_size)
        self.cell = nn.LSTMCell(num_actions, lstm_size)
        self.register_buffer("_eye", torch.eye(num_actions), persistent=False)

    def forward(self, x: Tensor) -> Tensor:
        B = x.shape[0]
        A = self.num_actions
        c = self.root_cell(self.root_mlp(x))  # [B, H] cell state
        h = torch.tanh(c)
        one_hot = self._eye.repeat(B, 1)  # [B*A, A]
        h_rep = h.repeat_interleave(A, dim=0)
        c_rep = c.repeat_interleave(A, dim=0)
        h_out, _ = self.cell(one_hot, (h_rep, c_rep))
        return h_out.view(B, A, self.lstm_size)


class DiscoAgentNetwork(nn.Module):
    """Shared torso + five heads (reference disco103.py:113-152)."""

    def __init__(self, obs_dim: int, num_actions: int,
                 torso_sizes: Sequence[int] = (256, 256),
                 lstm_size: int = 256, num_bins: int = 601,
                 prediction_size: int = 600):
        super().__init__()
        layers: list[nn.Module] = []
        d = obs_dim
        for s in torso_sizes:
            layers += [nn.Linear(d, s), nn.ReLU()]
            d = s
        self.shared_torso = nn.Sequential(*layers)
        self.action_conditional_torso = LSTMActionConditionedTorso(d, num_actions, lstm_size)
        self.logits_head = nn.Linear(d, num_actions)
        self.y_head = nn.Linear(d, prediction_size)
        self.q_head = nn.Linear(lstm_size, num_bins)
        self.z_head = nn.Linear(lstm_size, prediction_size)
        self.aux_pi_head = nn.Linear(lstm_size, 1)
        self.num_bins = num_bins
        self.prediction_size = prediction_size
        self.num_actions = num_actions

    def forward(self, obs: Tensor) -> DiscoAgentOutput:
        x = obs.reshape(obs.shape[0], -1)
        t = self.shared_torso(x)
        logits = self.logits_head(t)
        y = self.y_head(t)
        ac = self.action_conditional_torso(t)  # [B, A, H]
        q = self.q_head(ac)  # [B, A, bins]
        z = self.z_head(ac)  # [B, A, pred]
        aux_pi = self.aux_pi_head(ac).squeeze(-1)  # [B, A]
        return DiscoAgentOutput(logits=logits, q=q, y=y, z=z, aux_pi=aux_pi)
