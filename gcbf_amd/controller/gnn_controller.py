"""GNN policy controller (reference gcbf/controller/gnn_controller.py:13-48).

State-dict layout matches the reference (``feat_transformer.module_0.*``,
``feat_2_action.*``) so its checkpoints load directly.
"""
from __future__ import annotations

import torch
import torch.nn as nn
from torch import Tensor

from ..graph import GraphBatch
from ..nn import MLP, ControllerGNNLayer
from .base import MultiAgentController


class _Seq(nn.Module):
    """Name-compat container mirroring PyG ``Sequential``'s ``module_0``."""

    def __init__(self, layer: nn.Module):
        super().__init__()
        self.module_0 = layer


class GNNController(MultiAgentController):

    def __init__(self, num_agents: int, node_dim: int, edge_dim: int,
                 phi_dim: int, action_dim: int):
        super().__init__(num_agents=num_agents, node_dim=node_dim,
                         edge_dim=edge_dim, action_dim=action_dim)
        self.feat_transformer = _Seq(ControllerGNNLayer(
            node_dim=node_dim, edge_dim=edge_dim, output_dim=1024,
            phi_dim=phi_dim))
        self.feat_2_action = MLP(in_channels=1024 + action_dim,
                                 out_channels=action_dim,
                                 hidden_layers=(512, 128, 32))

    def forward(self, data: GraphBatch) -> Tensor:
        nm = data.agent_mask
        if data.agent_index is not None:
            nm = data.agent_index      # static LONG indices (capture-safe)
        elif data.agents_first_n is not None:
            nm = data.agents_first_n
        x = self.feat_transformer.module_0(
            data.x, data.edge_attr, data.edge_index,
            node_mask=nm, seg_dst=data.seg_dst)
        return self.feat_2_action(torch.cat([x, data.u_ref], dim=1))
