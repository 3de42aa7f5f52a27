"""Algorithm factory (reference gcbf/algo/__init__.py:12-36)."""
import torch

from typing import Optional

from ..env import MultiAgentEnv
from .base import Algorithm
from .buffer import Buffer
from .gcbf import GCBF, CBFGNN
from .macbf import MACBF, CBFNet
from .nominal import Nominal


def make_algo(algo: str, env: MultiAgentEnv, num_agents: int, node_dim: int,
              edge_dim: int, action_dim: int, device: torch.device,
              batch_size: int = 128,
              hyperparams: Optional[dict] = None) -> Algorithm:
    if algo == "nominal":
        return Nominal(env, num_agents, node_dim, edge_dim, action_dim,
                       device)
    if algo == "gcbf":
        return GCBF(env, num_agents, node_dim, edge_dim, action_dim, device,
                    batch_size, hyperparams)
    elif algo == "macbf":
        return MACBF(env, num_agents, node_dim, edge_dim, action_dim, device,
                     batch_size, hyperparams)
    else:
        raise NotImplementedError("Unknown Algorithm!")


__all__ = ["Algorithm", "Buffer", "GCBF", "CBFGNN", "MACBF", "CBFNet",
           "Nominal", "make_algo"]
