"""Build networks from config.

The reference instantiates Flax modules via hydra ``_target_`` paths
(/root/reference/stoix/configs/network/mlp.yaml ->
hydra.utils.instantiate at ff_ppo.py:439-447). Here network configs name
component types symbolically and this factory resolves them, inferring input
dimensions from the environment spaces (torch modules need explicit in_dims,
unlike Flax's lazy shape inference).
"""
from __future__ import annotations

from typing import Optional

import torch.nn as nn

from stoix_amd.envs.spaces import BoxSpace, DiscreteSpace, MultiDiscreteSpace, Space
from stoix_amd.networks import heads as H
from stoix_amd.networks import torso as T
from stoix_amd.networks.base import FeedForwardActor, FeedForwardCritic, RecurrentActor, RecurrentCritic, ScannedRNN
from stoix_amd.networks.dueling import DistributionalDuelingQNetwork, DuelingQNetwork
from stoix_amd.networks.inputs import EmbeddingActionInput, EmbeddingActionOnehotInput
from stoix_amd.networks.resnet import ResNetTorso, VisualResNetTorso

TORSOS = {
    "mlp": T.MLPTorso,
    "MLPTorso": T.MLPTorso,
    "noisy_mlp": T.NoisyMLPTorso,
    "NoisyMLPTorso": T.NoisyMLPTorso,
    "cnn": T.CNNTorso,
    "CNNTorso": T.CNNTorso,
    "resnet": ResNetTorso,
    "ResNetTorso": ResNetTorso,
    "visual_resnet": VisualResNetTorso,
    "VisualResNetTorso": VisualResNetTorso,
}


def _obs_dim(space: Space) -> int:
    total = 1
    for s in space.shape:
        total *= s
    return total


def build_torso(cfg: dict, input_dim: int) -> nn.Module:
    cfg = dict(cfg)
    kind = cfg.pop("_target_", cfg.pop("type", "mlp")).split(".")[-1]
    cls = TORSOS[kind]
    if cls in (T.CNNTorso, VisualResNetTorso):
        return cls(input_shape=cfg.pop("input_shape"), **cfg)
    return cls(input_dim=input_dim, **cfg)


def build_action_head(cfg: dict, input_dim: int, action_space: Space) -> nn.Module:
    cfg = dict(cfg)
    kind = cfg.pop("_target_", cfg.pop("type", None)).split(".")[-1]
    if kind in ("CategoricalHead", "categorical"):
        assert isinstance(action_space, DiscreteSpace)
        return H.CategoricalHead(input_dim, action_space.num_values)
    if kind in ("NormalAffineTanhDistributionHead", "tanh_normal"):
        assert isinstance(action_space, BoxSpace)
        mn = float(action_space.minimum.min())
        mx = float(action_space.maximum.max())
        return H.NormalAffineTanhDistributionHead(input_dim, action_space.shape[0], mn, mx, **cfg)
    if kind in ("BetaDistributionHead", "beta"):
        mn = float(action_space.minimum.min())
        mx = float(action_space.maximum.max())
        return H.BetaDistributionHead(input_dim, action_space.shape[0], mn, mx)
    if kind in ("MultivariateNormalDiagHead", "mvn_diag"):
        return H.MultivariateNormalDiagHead(input_dim, action_space.shape[0], **cfg)
    if kind in ("DeterministicHead", "deterministic"):
        mn = float(action_space.minimum.min())
        mx = float(action_space.maximum.max())
        return H.DeterministicHead(input_dim, action_space.shape[0], mn, mx)
    if kind in ("DiscreteQNetworkHead", "discrete_q"):
        assert isinstance(action_space, DiscreteSpace)
        return H.DiscreteQNetworkHead(input_dim, action_space.num_values, **cfg)
    if kind in ("DistributionalDiscreteQNetworkHead", "c51"):
        return H.DistributionalDiscreteQNetworkHead(input_dim, action_space.num_values, **cfg)
    if kind in ("QuantileDiscreteQNetworkHead", "qr"):
        return H.QuantileDiscreteQNetworkHead(input_dim, action_space.num_values, **cfg)
    if kind in ("MultiDiscreteHead", "multi_discrete"):
        assert isinstance(action_space, MultiDiscreteSpace)
        return H.MultiDiscreteHead(input_dim, action_space.num_values_list)
    raise ValueError(f"unknown action head '{kind}'")


def build_critic_head(cfg: dict, input_dim: int) -> nn.Module:
    cfg = dict(cfg)
    kind = cfg.pop("_target_", cfg.pop("type", "ScalarCriticHead")).split(".")[-1]
    if kind in ("ScalarCriticHead", "scalar"):
        return H.ScalarCriticHead(input_dim)
    if kind in ("CategoricalCriticHead", "categorical_critic"):
        return H.CategoricalCriticHead(input_dim, **cfg)
    if kind in ("DistributionalContinuousQNetworkHead", "d4pg"):
        return H.DistributionalContinuousQNetworkHead(input_dim, **cfg)
    if kind in ("LinearHead", "linear"):
        return H.LinearHead(input_dim, **cfg)
    raise ValueError(f"unknown critic head '{kind}'")


def build_actor(net_cfg: dict, obs_space: Space, action_space: Space) -> nn.Module:
    torso = build_torso(net_cfg["pre_torso"], _obs_dim(obs_space))
    head = build_action_head(net_cfg["action_head"], torso.output_dim, action_space)
    return FeedForwardActor(torso, head)


def build_critic(
    net_cfg: dict,
    obs_space: Space,
    action_space: Optional[Space] = None,
    obs_action_input: bool = False,
    onehot_action_input: bool = False,
) -> nn.Module:
    in_dim = _obs_dim(obs_space)
    input_layer: Optional[nn.Module] = None
    if obs_action_input:
        assert action_space is not None
        if onehot_action_input:
            assert isinstance(action_space, DiscreteSpace)
            input_layer = EmbeddingActionOnehotInput(action_space.num_values)
            in_dim += action_space.num_values
        else:
            input_layer = EmbeddingActionInput()
            in_dim += action_space.shape[0]
    torso = build_torso(net_cfg["pre_torso"], in_dim)
    head = build_critic_head(net_cfg.get("critic_head", {}), torso.output_dim)
    return FeedForwardCritic(torso, head, input_layer=input_layer)


def build_q_network(net_cfg: dict, obs_space: Space, action_space: DiscreteSpace, epsilon: float) -> nn.Module:
    """Build a discrete Q-network: plain, dueling, C51, QR or noisy-dueling."""
    cfg = dict(net_cfg.get("action_head", {}))
    kind = cfg.pop("_target_", cfg.pop("type", "DiscreteQNetworkHead")).split(".")[-1]
    in_dim = _obs_dim(obs_space)
    if kind in ("DuelingQNetwork", "dueling"):
        return DuelingQNetwork(in_dim, action_space.num_values, epsilon=epsilon, **cfg)
    if kind in ("DistributionalDuelingQNetwork", "NoisyDistributionalDuelingQNetwork", "rainbow"):
        noisy = kind != "DistributionalDuelingQNetwork" or cfg.pop("noisy", False)
        return DistributionalDuelingQNetwork(in_dim, action_space.num_values, noisy=noisy, **cfg)
    torso = build_torso(net_cfg["pre_torso"], in_dim)
    head = build_action_head({**cfg, "_target_": kind, "epsilon": epsilon}, torso.output_dim, action_space)
    return FeedForwardActor(torso, head)


def build_recurrent_actor(net_cfg: dict, obs_space: Space, action_space: Space) -> RecurrentActor:
    pre = build_torso(net_cfg["pre_torso"], _obs_dim(obs_space))
    rnn_cfg = dict(net_cfg.get("rnn", {}))
    rnn = ScannedRNN(pre.output_dim, rnn_cfg.get("hidden_dim", 256), rnn_cfg.get("cell_type", "gru"))
    post = build_torso(net_cfg["post_torso"], rnn.hidden_dim)
    head = build_action_head(net_cfg["action_head"], post.output_dim, action_space)
    return RecurrentActor(pre, rnn, post, head)


def build_recurrent_critic(net_cfg: dict, obs_space: Space) -> RecurrentCritic:
    pre = build_torso(net_cfg["pre_torso"], _obs_dim(obs_space))
    rnn_cfg = dict(net_cfg.get("rnn", {}))
    rnn = ScannedRNN(pre.output_dim, rnn_cfg.get("hidden_dim", 256), rnn_cfg.get("cell_type", "gru"))
    post = build_torso(net_cfg["post_torso"], rnn.hidden_dim)
    head = build_critic_head(net_cfg.get("critic_head", {}), post.output_dim)
    return RecurrentCritic(pre, rnn, post, head)


def build_shared_policy_value(net_cfg: dict, obs_space: Space, action_space: Space) -> nn.Module:
    """Single-network policy+value with a shared torso (reference
    ff_impala_shared_torso.py uses one net with PolicyValueHead)."""
    from stoix_amd.networks.base import SharedPolicyValueNetwork

    torso = build_torso(net_cfg["pre_torso"], _obs_dim(obs_space))
    action_head = build_action_head(net_cfg["action_head"], torso.output_dim, action_space)
    return SharedPolicyValueNetwork(torso, action_head)
