"""GNN layers as explicit gather → φ → gate → segment-softmax → Σ → γ pipelines.

MI355X-native equivalents of the reference's PyG ``MessagePassing`` layers
(gcbf/nn/gnn.py:14-135).  Instead of PyG's collect/message/aggregate/update
machinery, each layer is a flat pipeline over dst-sorted edge lists whose
segment ops dispatch to HIP kernels on GPU (gcbf_amd/ops).  Submodule names
(``phi``, ``gamma``, ``aggr_module.gate_nn``) match the reference so
state-dicts are interchangeable.

Structural optimization vs. the reference: the per-node update MLP γ is
row-wise, so when the caller only consumes agent rows (obstacle rows are
masked away right after the layer, gcbf/algo/gcbf.py:52-53), γ runs on the
agent rows only — identical outputs on the rows that are used.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
from torch import Tensor

from .. import ops
from ..graph import GraphBatch
from .mlp import MLP


class AttentionalAggregation(nn.Module):
    """gate = gate_nn(msg); att = scatter-softmax(gate, dst); out = Σ att·msg.

    Equivalent of PyG ``AttentionalAggregation`` (reference gcbf/nn/gnn.py:17).
    """

    def __init__(self, gate_nn: nn.Module):
        super().__init__()
        self.gate_nn = gate_nn

    def forward(self, msg: Tensor, dst: Tensor, num_nodes: int) -> Tensor:
        gate = self.gate_nn(msg)
        return ops.segment_attn_aggregate(msg, gate, dst, num_nodes)


def _gather_edge_inputs(x: Tensor, edge_attr: Tensor,
                        edge_index: Tensor) -> Tensor:
    """cat[x_i, x_j, edge_attr] per edge, with i=dst (receiver), j=src.

    PyG convention used by the reference (gcbf/nn/gnn.py:30-32): ``x_i`` is the
    target node's features, ``x_j`` the source's.
    """
    src, dst = edge_index[0], edge_index[1]
    return torch.cat([x.index_select(0, dst), x.index_select(0, src),
                      edge_attr], dim=1)


class CBFGNNLayer(nn.Module):
    """Attention message-passing layer of the CBF GNN.

    Reference: gcbf/nn/gnn.py:14-53.  φ and γ are spectral-normed
    (Lipschitz-limited); the gate MLP is not.
    """

    def __init__(self, node_dim: int, edge_dim: int, output_dim: int,
                 phi_dim: int):
        super().__init__()
        self.phi = MLP(in_channels=2 * node_dim + edge_dim,
                       out_channels=phi_dim, hidden_layers=(2048, 2048),
                       limit_lip=True)
        self.gamma = MLP(in_channels=phi_dim + node_dim,
                         out_channels=output_dim, hidden_layers=(2048, 2048),
                         limit_lip=True)
        self.aggr_module = AttentionalAggregation(
            gate_nn=MLP(in_channels=phi_dim, out_channels=1,
                        hidden_layers=(128, 128), limit_lip=False))

    def forward(self, x: Tensor, edge_attr: Tensor, edge_index: Tensor,
                node_mask: Optional[Tensor] = None,
                seg_dst: Optional[Tensor] = None) -> Tensor:
        num_nodes = x.shape[0]
        msg = self.phi(_gather_edge_inputs(x, edge_attr, edge_index))
        dst = edge_index[1] if seg_dst is None else seg_dst
        aggr = self.aggr_module(msg, dst, num_nodes)
        gamma_in = torch.cat([aggr, x], dim=1)
        if node_mask is not None:
            # int = "first n rows" static slice (capture-safe); tensor =
            # boolean mask (batched graphs)
            gamma_in = (gamma_in[:node_mask] if isinstance(node_mask, int)
                        else gamma_in[node_mask])
        return self.gamma(gamma_in)

    def attention(self, data: GraphBatch) -> Tensor:
        """Per-edge softmax attention weights (for plotting; reference
        gcbf/nn/gnn.py:44-53)."""
        msg = self.phi(_gather_edge_inputs(data.x, data.edge_attr,
                                           data.edge_index))
        gate = self.aggr_module.gate_nn(msg)
        return ops.segment_softmax(gate, data.edge_index[1], data.num_nodes)


class ControllerGNNLayer(nn.Module):
    """Same structure as CBFGNNLayer without spectral norm (reference
    gcbf/nn/gnn.py:56-79)."""

    def __init__(self, node_dim: int, edge_dim: int, output_dim: int,
                 phi_dim: int):
        super().__init__()
        self.phi = MLP(in_channels=2 * node_dim + edge_dim,
                       out_channels=phi_dim, hidden_layers=(2048, 2048))
        self.gamma = MLP(in_channels=phi_dim + node_dim,
                         out_channels=output_dim, hidden_layers=(2048, 2048))
        self.aggr_module = AttentionalAggregation(
            gate_nn=MLP(in_channels=phi_dim, out_channels=1,
                        hidden_layers=(128, 128)))

    def forward(self, x: Tensor, edge_attr: Tensor, edge_index: Tensor,
                node_mask: Optional[Tensor] = None,
                seg_dst: Optional[Tensor] = None) -> Tensor:
        num_nodes = x.shape[0]
        msg = self.phi(_gather_edge_inputs(x, edge_attr, edge_index))
        dst = edge_index[1] if seg_dst is None else seg_dst
        aggr = self.aggr_module(msg, dst, num_nodes)
        gamma_in = torch.cat([aggr, x], dim=1)
        if node_mask is not None:
            gamma_in = (gamma_in[:node_mask] if isinstance(node_mask, int)
                        else gamma_in[node_mask])
        return self.gamma(gamma_in)


class CBFNetLayer(nn.Module):
    """MACBF per-edge CBF: φ output returned per edge, no aggregation
    (reference gcbf/nn/gnn.py:82-111)."""

    def __init__(self, node_dim: int, edge_dim: int, output_dim: int):
        super().__init__()
        self.phi = MLP(in_channels=2 * node_dim + edge_dim,
                       out_channels=output_dim, hidden_layers=(64, 128, 64),
                       limit_lip=False)

    def forward(self, x: Tensor, edge_attr: Tensor,
                edge_index: Tensor) -> Tensor:
        return self.phi(_gather_edge_inputs(x, edge_attr, edge_index))


class MACBFControllerLayer(nn.Module):
    """Max-aggregation message passing (reference gcbf/nn/gnn.py:114-135)."""

    def __init__(self, node_dim: int, edge_dim: int, output_dim: int,
                 phi_dim: int):
        super().__init__()
        self.phi = MLP(in_channels=2 * node_dim + edge_dim,
                       out_channels=phi_dim, hidden_layers=(64,))
        self.gamma = MLP(in_channels=phi_dim, out_channels=output_dim,
                         hidden_layers=(64, 128, 64))

    def forward(self, x: Tensor, edge_attr: Tensor, edge_index: Tensor,
                node_mask: Optional[Tensor] = None) -> Tensor:
        num_nodes = x.shape[0]
        msg = self.phi(_gather_edge_inputs(x, edge_attr, edge_index))
        aggr = ops.segment_max(msg, edge_index[1], num_nodes)
        if node_mask is not None:
            aggr = aggr[node_mask]
        return self.gamma(aggr)
