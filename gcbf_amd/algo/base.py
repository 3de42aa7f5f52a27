"""Algorithm ABC (reference gcbf/algo/base.py:13-189)."""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Optional

import numpy as np
import torch
from torch import Tensor

from ..env import MultiAgentEnv
from ..graph import GraphBatch


class Algorithm(ABC):

    def __init__(self, env: MultiAgentEnv, num_agents: int, node_dim: int,
                 edge_dim: int, action_dim: int, device: torch.device):
        super().__init__()
        self._env = env
        self._num_agents = num_agents
        self._node_dim = node_dim
        self._edge_dim = edge_dim
        self._action_dim = action_dim
        self._device = device
        self.params = {}

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

    @property
    def device(self) -> torch.device:
        return self._device

    @abstractmethod
    def act(self, data: GraphBatch) -> Tensor:
        """No-grad policy action."""

    @abstractmethod
    def step(self, data: GraphBatch, prob: float) -> Tensor:
        """Training-time action (with exploration and buffer bookkeeping)."""

    def post_step(self, data: GraphBatch, action: Tensor, reward, done: bool,
                  next_data: GraphBatch):
        pass

    def sample(self, data: GraphBatch, prob: float = 0.01) -> Tensor:
        """Policy action with occasional exploration noise
        (reference gcbf/algo/base.py:95-116)."""
        actions = self.act(data)
        action_lim = self._env.action_lim
        if np.random.uniform() < prob:
            noise = torch.randn_like(actions) * 0.3 * (
                action_lim[1] - action_lim[0])
            actions = actions + noise
        return actions

    @abstractmethod
    def is_update(self, step: int) -> bool: ...

    @abstractmethod
    def update(self, step: int, writer=None) -> dict: ...

    @abstractmethod
    def save(self, save_dir: str): ...

    @abstractmethod
    def load(self, load_dir: str): ...

    def apply(self, data: GraphBatch, rand: Optional[float] = 30) -> Tensor:
        """Test-time action (possibly refined against the CBF condition)."""
