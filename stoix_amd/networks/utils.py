"""Activation / RNN-cell registries (parity with
/root/reference/stoix/networks/utils.py:7-37)."""
from __future__ import annotations

import torch.nn as nn

_ACTIVATIONS = {
    "relu": nn.ReLU,
    "silu": nn.SiLU,
    "swish": nn.SiLU,
    "tanh": nn.Tanh,
    "gelu": nn.GELU,
    "elu": nn.ELU,
    "leaky_relu": nn.LeakyReLU,
    "identity": nn.Identity,
}


def get_activation(name: str):
    if name not in _ACTIVATIONS:
        raise ValueError(f"unknown activation '{name}' (have {list(_ACTIVATIONS)})")
    return _ACTIVATIONS[name]


_RNN_CELLS = {}


def register_rnn_cell(name: str):
    def deco(cls):
        _RNN_CELLS[name] = cls
        return cls

    return deco


def get_rnn_cell(name: str):
    # populated by stoix_amd.networks.layers at import time
    from stoix_amd.networks import layers  # noqa: F401

    if name not in _RNN_CELLS:
        raise ValueError(f"unknown rnn cell '{name}' (have {list(_RNN_CELLS)})")
    return _RNN_CELLS[name]
