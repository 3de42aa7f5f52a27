"""MLP with optional orthogonal init and spectral normalization.

Mirrors the reference MLP (gcbf/nn/mlp.py:9-47) including its state-dict
layout: parameters live in an ``nn.Sequential`` named ``net`` with Linear
modules at the same indices (activations between), and ``limit_lip=True``
uses spectral normalization with old-style keys (``weight_orig``,
``weight_u``, ``weight_v``) — so reference checkpoints load directly.

Spectral norm is implemented natively (:class:`SNLinear`) rather than via
``torch.nn.utils.spectral_norm`` because the stock hook's power iteration
uses ``normalize(..., out=buffer)`` which breaks under bf16 autocast; here
the power iteration runs in fp32 with autocast disabled, while the actual
GEMM autocasts to bf16 on GPU.  Semantics match torch's: one power
iteration per forward in training mode, σ = uᵀWv with u,v as constants,
effective weight W/σ.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


def init_param(module: nn.Module, gain: float = 1.0) -> nn.Module:
    """Orthogonal weight init, zero bias (reference gcbf/nn/utils.py:4-7)."""
    nn.init.orthogonal_(module.weight.data, gain=gain)
    nn.init.constant_(module.bias.data, 0)
    return module


_SN_REUSE_DEPTH = 0
_SN_CACHED_LAYERS: list = []


class sn_weight_reuse:
    """Context manager: within the block, each SNLinear computes its
    normalized weight W/σ ONCE and reuses it for every forward (gradients
    still flow through all uses — the tensor is shared in the autograd
    graph).  One training iteration runs several forwards of the same CBF
    (doubled h/h_next batch + the re-linked residue pass); the reference
    re-runs the power iteration and the full-weight division per forward
    (torch.nn.utils.spectral_norm semantics), which re-normalizes a
    2048x2048 weight 12x per iteration for σ values that differ only by
    one extra power-iteration step.  Documented deviation: the power
    iteration advances once per ITERATION here instead of once per
    forward; σ converges to the same fixed point.
    """

    def __enter__(self):
        global _SN_REUSE_DEPTH
        _SN_REUSE_DEPTH += 1
        return self

    def __exit__(self, *exc):
        global _SN_REUSE_DEPTH
        _SN_REUSE_DEPTH -= 1
        if _SN_REUSE_DEPTH == 0:
            for layer in _SN_CACHED_LAYERS:
                layer._w_cache = None
            _SN_CACHED_LAYERS.clear()
        return False


class SNLinear(nn.Module):
    """Linear layer with spectral normalization (Lipschitz ≤ 1).

    State-dict keys match old-style ``torch.nn.utils.spectral_norm`` on an
    ``nn.Linear``: ``weight_orig``, ``bias``, ``weight_u``, ``weight_v``.
    """

    def __init__(self, in_features: int, out_features: int,
                 n_power_iterations: int = 1, eps: float = 1e-12):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.n_power_iterations = n_power_iterations
        self.eps = eps
        self.weight_orig = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features))
        nn.init.kaiming_uniform_(self.weight_orig, a=5 ** 0.5)
        u = F.normalize(torch.randn(out_features), dim=0, eps=eps)
        v = F.normalize(torch.randn(in_features), dim=0, eps=eps)
        self.register_buffer("weight_u", u)
        self.register_buffer("weight_v", v)
        self._w_cache = None  # per-iteration W/σ reuse (sn_weight_reuse)

    @property
    def weight(self) -> torch.Tensor:
        """Effective (normalized) weight — read-only convenience."""
        with torch.no_grad():
            sigma = torch.dot(self.weight_u,
                              torch.mv(self.weight_orig, self.weight_v))
        return self.weight_orig / sigma

    def effective_weight(self) -> torch.Tensor:
        """W/σ with gradient through W (u, v constants), running one power
        iteration first in training mode — torch's spectral_norm semantics.

        Runs with autocast OFF: the matvecs are tiny in fp32 but autocast
        would route them to a pathologically slow bf16 GEMV (~20x the whole
        layer's cost, measured on MI355X).
        """
        if _SN_REUSE_DEPTH and self._w_cache is not None:
            return self._w_cache
        W = self.weight_orig
        with torch.autocast(W.device.type if W.device.type != "cpu"
                            else "cpu", enabled=False):
            if self.training:
                with torch.no_grad():
                    Wf = W.detach().float()
                    u, v = self.weight_u, self.weight_v
                    for _ in range(self.n_power_iterations):
                        v = F.normalize(Wf.t().mv(u), dim=0, eps=self.eps)
                        u = F.normalize(Wf.mv(v), dim=0, eps=self.eps)
                    self.weight_u.copy_(u)
                    self.weight_v.copy_(v)
            # clone() so the next forward's in-place buffer update doesn't
            # invalidate this graph's saved tensors.
            u = self.weight_u.clone()
            v = self.weight_v.clone()
            sigma = torch.dot(u, torch.mv(W.float(), v))
            out = W / sigma
            if _SN_REUSE_DEPTH:
                self._w_cache = out
                _SN_CACHED_LAYERS.append(self)
            return out

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.effective_weight(), self.bias)

    def extra_repr(self) -> str:
        return f"in_features={self.in_features}, " \
               f"out_features={self.out_features}, spectral_norm=True"


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
            if limit_lip:
                lin = SNLinear(n_in, n_out)
                if init:
                    nn.init.orthogonal_(lin.weight_orig.data, gain=gain)
                    nn.init.constant_(lin.bias.data, 0)
            else:
                lin = nn.Linear(n_in, n_out)
                if init:
                    init_param(lin, gain=gain)
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
        # execution plan for the fused MFMA path: (linear, fused act code)
        from .fused import ACT_NONE, ACT_RELU, ACT_TANH
        plan = []
        mods = list(self.net)
        i = 0
        while i < len(mods):
            m = mods[i]
            if isinstance(m, (nn.Linear, SNLinear)):
                act = ACT_NONE
                if i + 1 < len(mods) and isinstance(mods[i + 1], nn.ReLU):
                    act = ACT_RELU
                    i += 1
                elif i + 1 < len(mods) and isinstance(mods[i + 1], nn.Tanh):
                    act = ACT_TANH
                    i += 1
                plan.append((m, act))
            i += 1
        self._plan = plan

    # GEMM row-bucketing: graph batches make M (edge/node counts) unique on
    # almost every call, and hipBLASLt pays a per-novel-shape algorithm
    # search (~1.4 ms bf16 / ~4.5 ms fp32 measured on MI355X vs ~0.1 ms
    # cached).  Padding M up to a bucket multiple makes shapes recur; the
    # zero rows are sliced off after the chain (values/grads unchanged).
    BUCKET = 256
    # set True (by utils.amp.enable_bf16) to route big layers through the
    # hand-written MFMA kernel with fused bias+activation epilogues
    fused_mfma = False

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.fused_mfma and x.is_cuda:
            from . import fused
            return fused.run_plan(self._plan, x, bucket=self.BUCKET)
        M = x.shape[0]
        if x.is_cuda and M > 0 and M % self.BUCKET != 0:
            pad = self.BUCKET - M % self.BUCKET
            x = torch.cat([x, x.new_zeros(pad, x.shape[1])])
            return self.net(x)[:M]
        return self.net(x)
