"""MLP with optional orthogonal init and spectral normalization.

Mirrors the reference MLP (gcbf/nn/mlp.py:9-47) including its state-dict
layout: parameters live in an ``nn.Sequential`` named ``net`` with Linear
modules at the same indices (activations between), and ``limit_lip=True``
wraps each Linear in old-style spectral norm (keys ``weight_orig``,
``weight_u``, ``weight_v``) — so reference checkpoints load directly.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn
from torch.nn.utils import spectral_norm


def init_param(module: nn.Module, gain: float = 1.0) -> nn.Module:
    """Orthogonal weight init, zero bias (reference gcbf/nn/utils.py:4-7)."""
    nn.init.orthogonal_(module.weight.data, gain=gain)
    nn.init.constant_(module.bias.data, 0)
    return module


class MLP(nn.Module):

    def __init__(self, in_channels: int, out_channels: int,
                 hidden_layers: Tuple[int, ...],
                 hidden_activation: Optional[nn.Module] = None,
                 output_activation: Optional[nn.Module] = None,
                 init: bool = True, gain: float = 1.0,
                 limit_lip: bool = False):
        super().__init__()
        if hidden_activation is None:
            hidden_activation = nn.ReLU()

        def make_linear(n_in: int, n_out: int) -> nn.Module:
            lin = nn.Linear(n_in, n_out)
            if init:
                init_param(lin, gain=gain)
            if limit_lip:
                lin = spectral_norm(lin)
            return lin

        layers = []
        units = in_channels
        for next_units in hidden_layers:
            layers.append(make_linear(units, next_units))
            layers.append(hidden_activation)
            units = next_units
        layers.append(make_linear(units, out_channels))
        if output_activation is not None:
            layers.append(output_activation)
        self.net = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)
