"""Controller ABC (reference gcbf/controller/base.py:8-48)."""
from __future__ import annotations

from abc import ABC, abstractmethod

import torch.nn as nn
from torch import Tensor

from ..graph import GraphBatch


class MultiAgentController(nn.Module, ABC):

    def __init__(self, num_agents: int, node_dim: int, edge_dim: int,
                 action_dim: int):
        super().__init__()
        self._num_agents = num_agents
        self._node_dim = node_dim
        self._edge_dim = edge_dim
        self._action_dim = action_dim

    @property
    def num_agents(self) -> int:
        return self._num_agents

    @property
    def node_dim(self) -> int:
        return self._node_dim

    @property
    def edge_dim(self) -> int:
        return self._edge_dim

    @property
    def action_dim(self) -> int:
        return self._action_dim

    @abstractmethod
    def forward(self, data: GraphBatch) -> Tensor:
        """Control actions (bs·n_agents, action_dim) for the batched graph."""
