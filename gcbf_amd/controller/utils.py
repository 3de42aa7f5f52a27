"""Stochastic-policy helpers (SAC-style log-probabilities).

API-parity port of the reference's unused controller/utils.py
(gcbf/controller/utils.py:8-102); kept for downstream RL extensions.
"""
import math

import torch
from torch import Tensor


def atanh(x: Tensor) -> Tensor:
    return 0.5 * (torch.log(1 + x + 1e-6) - torch.log(1 - x + 1e-6))


def calculate_log_pi(log_stds: Tensor, noises: Tensor,
                     actions: Tensor) -> Tensor:
    """log π(a|s) for a tanh-squashed Gaussian policy."""
    gaussian_log_probs = (
        -0.5 * noises.pow(2) - log_stds
    ).sum(dim=-1, keepdim=True) - 0.5 * math.log(2 * math.pi) * log_stds.size(-1)
    return gaussian_log_probs - torch.log(
        1 - actions.pow(2) + 1e-6).sum(dim=-1, keepdim=True)


def reparameterize(means: Tensor, log_stds: Tensor):
    """Sample a squashed-Gaussian action with the reparameterization trick."""
    noises = torch.randn_like(means)
    us = means + noises * log_stds.exp()
    actions = torch.tanh(us)
    return actions, calculate_log_pi(log_stds, noises, actions)


def evaluate_log_pi(means: Tensor, log_stds: Tensor,
                    actions: Tensor) -> Tensor:
    """log π(a|s) of given actions under the squashed Gaussian."""
    noises = (atanh(actions) - means) / (log_stds.exp() + 1e-8)
    return calculate_log_pi(log_stds, noises, actions)
