"""Nominal algorithm: pure reference-controller rollout
(reference gcbf/algo/nominal.py:14-59)."""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from ..controller import NominalController
from ..env import MultiAgentEnv
from ..graph import GraphBatch
from .base import Algorithm


class Nominal(Algorithm):

    def __init__(self, env: MultiAgentEnv, num_agents: int, node_dim: int,
                 edge_dim: int, action_dim: int, device: torch.device):
        super().__init__(env=env, num_agents=num_agents, node_dim=node_dim,
                         edge_dim=edge_dim, action_dim=action_dim,
                         device=device)
        self.actor = NominalController(
            num_agents=num_agents, node_dim=node_dim, edge_dim=edge_dim,
            action_dim=action_dim).to(device)

    def step(self, data: GraphBatch, prob: float) -> Tensor:
        raise NotImplementedError

    def is_update(self, step: int) -> bool:
        raise NotImplementedError

    def update(self, step: int, writer=None):
        raise NotImplementedError

    def save(self, save_dir: str):
        raise NotImplementedError

    def load(self, load_dir: str):
        raise NotImplementedError

    def act(self, data: GraphBatch) -> Tensor:
        with torch.no_grad():
            return self.actor(data)

    def apply(self, data: GraphBatch, rand: Optional[float] = 30) -> Tensor:
        return self.act(data)
