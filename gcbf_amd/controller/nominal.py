"""Zero-residual nominal controller (reference gcbf/controller/nominal.py)."""
from __future__ import annotations

import torch
from torch import Tensor

from ..graph import GraphBatch
from .base import MultiAgentController


class NominalController(MultiAgentController):
    """Returns zeros: the environment adds ``u_ref`` inside ``step``, so a
    zero residual is pure LQR/PID nominal control."""

    def forward(self, data: GraphBatch) -> Tensor:
        if data.agent_mask is not None:
            num_agents = int(data.agent_mask.sum().item())
        else:
            num_agents = data.num_nodes
        return torch.zeros(num_agents, self.action_dim).type_as(data.states)
