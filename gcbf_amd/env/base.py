"""Multi-agent environment base class.

Behavioral equivalent of the reference ``MultiAgentEnv`` ABC
(gcbf/env/base.py:11-398) over :class:`gcbf_amd.graph.GraphBatch` instead of
PyG ``Data``.  The concrete Euler integrator ``forward`` matches
gcbf/env/base.py:381-398.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from ..graph import GraphBatch


class MultiAgentEnv(ABC):

    def __init__(self, num_agents: int, device: torch.device, dt: float = 0.03,
                 params: Optional[dict] = None,
                 max_neighbors: Optional[int] = None):
        super().__init__()
        self._num_agents = num_agents
        self._device = device
        self._dt = dt
        if params is None:
            params = self.default_params
        self._params = params
        self._max_neighbors = max_neighbors
        self._data: Optional[GraphBatch] = None
        self._t = 0
        self._mode = "train"

    # mode switches (reference gcbf/env/base.py:33-40)
    def train(self):
        self._mode = "train"

    def test(self):
        self._mode = "test"

    def demo(self, idx: int):
        self._mode = f"demo_{idx}"

    @property
    def num_agents(self) -> int:
        return self._num_agents

    @property
    def dt(self) -> float:
        return self._dt

    @property
    def device(self) -> torch.device:
        return self._device

    @property
    def data(self) -> GraphBatch:
        return self._data

    @property
    def state(self) -> Tensor:
        return self._data.states

    @property
    def params(self) -> dict:
        return self._params

    @property
    @abstractmethod
    def state_dim(self) -> int: ...

    @property
    @abstractmethod
    def node_dim(self) -> int: ...

    @property
    @abstractmethod
    def edge_dim(self) -> int: ...

    @property
    @abstractmethod
    def action_dim(self) -> int: ...

    @property
    @abstractmethod
    def max_episode_steps(self) -> int: ...

    @property
    @abstractmethod
    def default_params(self) -> dict: ...

    @abstractmethod
    def dynamics(self, data: GraphBatch, u: Tensor) -> Tensor:
        """Time derivative of the state given control input u."""

    @abstractmethod
    def reset(self) -> GraphBatch: ...

    @abstractmethod
    def step(self, action: Tensor) -> Tuple[GraphBatch, np.ndarray, bool, dict]:
        """One env step: (next_data, reward, done, info).

        Info-dict contract (differs from the reference): ``info["collision"]``
        and ``info["reach"]`` are per-agent BOOLEAN MASK tensors on device,
        not lists of agent indices (the reference DubinsCar returned
        indices, gcbf/env/dubins_car.py:612-615).  Boolean masks keep the
        step host-sync-free; index consumers use ``mask.nonzero()``.
        """

    @abstractmethod
    def forward_graph(self, data: GraphBatch, action: Tensor) -> GraphBatch:
        """Advance the batched graph one step, keeping edge topology."""

    @abstractmethod
    def render(self, traj=None, return_ax: bool = False,
               plot_edge: bool = True, ax=None): ...

    @abstractmethod
    def edge_attr(self, state: Tensor, edge_index: Tensor) -> Tensor: ...

    @abstractmethod
    def add_communication_links(self, data: GraphBatch) -> GraphBatch:
        """Rebuild edge_index/edge_attr from positions (single graph)."""

    def add_communication_links_batched(self, data: GraphBatch) -> GraphBatch:
        """Rebuild communication links for a whole uniform batch at once.

        New capability vs. the reference, which loops over graphs in Python
        (gcbf/algo/gcbf.py:196-200); used by the ḣ re-link step in training.
        Default implementation falls back to add_communication_links semantics
        per graph but vectorized in the concrete envs.
        """
        raise NotImplementedError

    @property
    @abstractmethod
    def state_lim(self) -> Tuple[Tensor, Tensor]: ...

    @property
    @abstractmethod
    def action_lim(self) -> Tuple[Tensor, Tensor]: ...

    @abstractmethod
    def u_ref(self, data: GraphBatch) -> Tensor: ...

    @abstractmethod
    def safe_mask(self, data: GraphBatch, return_edge: bool = False) -> Tensor: ...

    @abstractmethod
    def unsafe_mask(self, data: GraphBatch, return_edge: bool = False) -> Tensor: ...

    @abstractmethod
    def collision_mask(self, data: GraphBatch) -> Tensor: ...

    def forward(self, data: GraphBatch, u: Tensor) -> Tensor:
        """Euler step x + ẋ·dt (reference gcbf/env/base.py:381-398)."""
        xdot = self.dynamics(data, u)
        return data.states + xdot * self.dt
