"""World model for MuZero (parity: /root/reference/stoix/networks/
model_based.py:15-129 — ``RewardBasedWorldModel`` with initial_inference /
recurrent_inference, stacked-RNN dynamics, hidden-state min-max
normalisation, reward head)."""
from __future__ import annotations

from typing import NamedTuple, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from stoix_amd.networks.layers import StackedRNN
from stoix_amd.networks.torso import MLPTorso, orthogonal_init

Tensor = torch.Tensor


def two_hot(x: Tensor, atoms: Tensor) -> Tensor:
    """Project scalars onto a categorical support with two-hot encoding."""
    n = atoms.shape[0]
    vmin, vmax = atoms[0], atoms[-1]
    dz = (vmax - vmin) / (n - 1)
    x = x.clamp(vmin, vmax)
    b = (x - vmin) / dz
    lo = b.floor().clamp(0, n - 1)
    hi = b.ceil().clamp(0, n - 1)
    w_hi = b - lo
    w_lo = 1.0 - w_hi
    same = (lo == hi).to(x.dtype)
    w_lo = w_lo + same * w_hi
    w_hi = w_hi * (1.0 - same)
    out = torch.zeros((*x.shape, n), dtype=x.dtype, device=x.device)
    out.scatter_add_(-1, lo.long().unsqueeze(-1), w_lo.unsqueeze(-1))
    out.scatter_add_(-1, hi.long().unsqueeze(-1), w_hi.unsqueeze(-1))
    return out


class CategoricalValueHead(nn.Module):
    """Two-hot categorical value/reward head (MuZero-style)."""

    def __init__(self, input_dim: int, vmin: float = -50.0, vmax: float = 50.0, num_atoms: int = 101):
        super().__init__()
        self.linear = orthogonal_init(nn.Linear(input_dim, num_atoms), scale=0.01)
        self.register_buffer("atoms", torch.linspace(vmin, vmax, num_atoms))

    def forward(self, x: Tensor) -> Tuple[Tensor, Tensor]:
        logits = self.linear(x)
        value = (F.softmax(logits, dim=-1) * self.atoms).sum(-1)
        return value, logits

    def ce_loss(self, logits: Tensor, target_scalar: Tensor) -> Tensor:
        tgt = two_hot(target_scalar.detach(), self.atoms)
        return -(tgt * F.log_softmax(logits, dim=-1)).sum(-1)


class ModelOutput(NamedTuple):
    hidden: Tensor
    rnn_state: list
    reward: Tensor
    reward_logits: Tensor
    policy_logits: Tensor
    value: Tensor
    value_logits: Tensor


class RewardBasedWorldModel(nn.Module):
    """representation + RNN dynamics + reward/policy/value prediction."""

    def __init__(
        self,
        obs_dim: int,
        num_actions: int,
        hidden_dim: int = 128,
        repr_layers=(128,),
        head_layers=(64,),
        rnn_layers: int = 1,
        vmin: float = -50.0,
        vmax: float = 50.0,
        num_atoms: int = 51,
    ):
        super().__init__()
        self.num_actions = num_actions
        self.hidden_dim = hidden_dim
        self.repr_net = MLPTorso(obs_dim, (*repr_layers, hidden_dim))
        self.dynamics = StackedRNN(num_actions, hidden_dim, num_layers=rnn_layers, cell_type="gru")
        self.reward_head = CategoricalValueHead(hidden_dim, vmin, vmax, num_atoms)
        self.policy_torso = MLPTorso(hidden_dim, head_layers)
        self.policy_head = orthogonal_init(nn.Linear(self.policy_torso.output_dim, num_actions), scale=0.01)
        self.value_torso = MLPTorso(hidden_dim, head_layers)
        self.value_head = CategoricalValueHead(self.value_torso.output_dim, vmin, vmax, num_atoms)

    @staticmethod
    def _normalize_hidden(h: Tensor) -> Tensor:
        """min-max normalise each hidden state (reference model_based.py)."""
        mn = h.min(dim=-1, keepdim=True).values
        mx = h.max(dim=-1, keepdim=True).values
        return (h - mn) / (mx - mn).clamp(min=1e-5)

    def initial_inference(self, obs: Tensor) -> ModelOutput:
        h = self._normalize_hidden(self.repr_net(obs))
        rnn_state = [s * 0 + h if isinstance(s, Tensor) else s for s in self.dynamics.initial_state(obs.shape[0], obs.device)]
        # seed the GRU state with the representation
        rnn_state = [h.clone() for _ in rnn_state]
        pol = self.policy_head(self.policy_torso(h))
        val, val_logits = self.value_head(self.value_torso(h))
        zero_r = torch.zeros_like(val)
        zero_rl = torch.zeros_like(val_logits)
        return ModelOutput(h, rnn_state, zero_r, zero_rl, pol, val, val_logits)

    def recurrent_inference(self, rnn_state: list, action: Tensor) -> ModelOutput:
        a_onehot = F.one_hot(action.long(), self.num_actions).to(torch.float32)
        h, new_state = self.dynamics(a_onehot, rnn_state)
        h = self._normalize_hidden(h)
        new_state = [self._normalize_hidden(s) if isinstance(s, Tensor) else s for s in new_state]
        reward, reward_logits = self.reward_head(h)
        pol = self.policy_head(self.policy_torso(h))
        val, val_logits = self.value_head(self.value_torso(h))
        return ModelOutput(h, new_state, reward, reward_logits, pol, val, val_logits)


class ContinuousModelOutput(NamedTuple):
    hidden: Tensor
    rnn_state: list
    reward: Tensor
    reward_logits: Tensor
    policy_loc: Tensor
    policy_scale: Tensor
    value: Tensor
    value_logits: Tensor


class ContinuousRewardBasedWorldModel(nn.Module):
    """Continuous-action world model for Sampled MuZero (reference
    ff_sampled_mz.py uses the same RewardBasedWorldModel with an
    action-sampled search; here the dynamics consume the raw action vector
    and the policy head is a tanh-normal parameterisation)."""

    def __init__(
        self,
        obs_dim: int,
        action_dim: int,
        hidden_dim: int = 128,
        repr_layers=(128,),
        head_layers=(64,),
        rnn_layers: int = 1,
        vmin: float = -50.0,
        vmax: float = 50.0,
        num_atoms: int = 51,
        min_scale: float = 1e-3,
    ):
        super().__init__()
        self.action_dim = action_dim
        self.hidden_dim = hidden_dim
        self.min_scale = min_scale
        self.repr_net = MLPTorso(obs_dim, (*repr_layers, hidden_dim))
        self.dynamics = StackedRNN(action_dim, hidden_dim, num_layers=rnn_layers, cell_type="gru")
        self.reward_head = CategoricalValueHead(hidden_dim, vmin, vmax, num_atoms)
        self.policy_torso = MLPTorso(hidden_dim, head_layers)
        self.loc_head = orthogonal_init(nn.Linear(self.policy_torso.output_dim, action_dim), scale=0.01)
        self.scale_head = orthogonal_init(nn.Linear(self.policy_torso.output_dim, action_dim), scale=0.01)
        self.value_torso = MLPTorso(hidden_dim, head_layers)
        self.value_head = CategoricalValueHead(self.value_torso.output_dim, vmin, vmax, num_atoms)

    def _policy(self, h: Tensor) -> Tuple[Tensor, Tensor]:
        p = self.policy_torso(h)
        loc = self.loc_head(p)
        scale = F.softplus(self.scale_head(p)) + self.min_scale
        return loc, scale

    def initial_inference(self, obs: Tensor) -> ContinuousModelOutput:
        h = RewardBasedWorldModel._normalize_hidden(self.repr_net(obs))
        rnn_state = [h.clone() for _ in self.dynamics.initial_state(obs.shape[0], obs.device)]
        loc, scale = self._policy(h)
        val, val_logits = self.value_head(self.value_torso(h))
        zero_r = torch.zeros_like(val)
        zero_rl = torch.zeros_like(val_logits)
        return ContinuousModelOutput(h, rnn_state, zero_r, zero_rl, loc, scale, val, val_logits)

    def recurrent_inference(self, rnn_state: list, action: Tensor) -> ContinuousModelOutput:
        h, new_state = self.dynamics(action.to(torch.float32), rnn_state)
        h = RewardBasedWorldModel._normalize_hidden(h)
        new_state = [RewardBasedWorldModel._normalize_hidden(s) if isinstance(s, Tensor) else s for s in new_state]
        reward, reward_logits = self.reward_head(h)
        loc, scale = self._policy(h)
        val, val_logits = self.value_head(self.value_torso(h))
        return ContinuousModelOutput(h, new_state, reward, reward_logits, loc, scale, val, val_logits)
