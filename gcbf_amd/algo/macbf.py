"""MACBF baseline algorithm (reference gcbf/algo/macbf.py:20-239).

Pairwise edge CBF (``CBFNet``) + max-aggregation controller; losses share the
GCBF structure but operate on per-edge h with edge masks, and the ḣ loss uses
only the fixed-topology path (no re-link residue).
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch
import torch.nn as nn
from torch import Tensor
from torch.optim import Adam

from ..controller import MACBFController
from ..env import MultiAgentEnv
from ..graph import GraphBatch
from ..nn import CBFNetLayer
from .buffer import Buffer
from .gcbf import GCBF, _Seq


class CBFNet(nn.Module):
    """Per-edge CBF values (reference gcbf/algo/macbf.py:20-48)."""

    def __init__(self, num_agents: int, node_dim: int, edge_dim: int):
        super().__init__()
        self._num_agents = num_agents
        self._top_k = 12
        self.net = _Seq(CBFNetLayer(node_dim=node_dim, edge_dim=edge_dim,
                                    output_dim=1))

    def forward(self, data: GraphBatch) -> Tensor:
        return self.net.module_0(data.x, data.edge_attr, data.edge_index)


class MACBF(GCBF):

    def __init__(self, env: MultiAgentEnv, num_agents: int, node_dim: int,
                 edge_dim: int, action_dim: int, device: torch.device,
                 batch_size: int = 500, params: Optional[dict] = None):
        super().__init__(env=env, num_agents=num_agents, node_dim=node_dim,
                         edge_dim=edge_dim, action_dim=action_dim,
                         device=device, batch_size=batch_size, params=params)
        # replace the GCBF networks with the MACBF ones
        self.cbf = CBFNet(num_agents=num_agents, node_dim=node_dim,
                          edge_dim=edge_dim).to(device)
        self.actor = MACBFController(num_agents=num_agents, node_dim=node_dim,
                                     edge_dim=edge_dim, phi_dim=128,
                                     action_dim=action_dim).to(device)
        self.optim_cbf = Adam(self.cbf.parameters(), lr=3e-4)
        self.optim_actor = Adam(self.actor.parameters(), lr=1e-3)
        self.buffer = Buffer()
        self.memory = Buffer()

    @torch.no_grad()
    def step(self, data: GraphBatch, prob: float) -> Tensor:
        # exploration probability floored at 0.5 (reference macbf.py:109)
        action = self.actor(data)
        prob = max(prob, 0.5)
        if np.random.rand() < prob:
            action = torch.zeros_like(action)
        is_safe = torch.logical_not(torch.any(self._env.unsafe_mask(data)))
        self.buffer.append(data, is_safe)
        return action

    def update(self, step: int, writer=None) -> dict:
        seg_len = 3
        inner_iter = self.params["inner_iter"]
        eps = self.params["eps"]
        alpha = self.params["alpha"]
        logs = []

        for i_inner in range(inner_iter):
            if self.memory.size == 0:
                graph_list = self.buffer.sample(self.batch_size // 5, seg_len)
            else:
                curr = self.buffer.sample(self.batch_size // 10, seg_len, True)
                prev = self.memory.sample(
                    self.batch_size // 5 - self.batch_size // 10, seg_len, True)
                graph_list = curr + prev

            graphs = GraphBatch.from_list(graph_list)
            graphs.edge_attr.requires_grad_(True)
            h = self.cbf(graphs)
            actions = self.actor(graphs)

            # per-edge masks (reference macbf.py:144, 156) as weighted
            # sums — identical values without the nonzero host sync
            hv = h[:, 0]
            unsafe_mask = self._env.unsafe_mask(graphs, return_edge=True)
            wu = unsafe_mask.to(hv.dtype)
            cu = wu.sum()
            cu1 = cu.clamp(min=1)
            any_u = (cu > 0).to(hv.dtype)
            loss_unsafe = any_u * (torch.relu(hv + eps) * wu).sum() / cu1
            acc_unsafe = (any_u * ((hv < 0).to(hv.dtype) * wu).sum() / cu1
                          + (1 - any_u))

            safe_mask = self._env.safe_mask(graphs, return_edge=True)
            ws = safe_mask.to(hv.dtype)
            cs = ws.sum()
            cs1 = cs.clamp(min=1)
            any_s = (cs > 0).to(hv.dtype)
            loss_safe = any_s * (torch.relu(-hv + eps) * ws).sum() / cs1
            acc_safe = (any_s * ((hv >= 0).to(hv.dtype) * ws).sum() / cs1
                        + (1 - any_s))

            # ḣ on the fixed topology only (reference macbf.py:168-173)
            graphs_next = self._env.forward_graph(graphs, actions)
            h_next = self.cbf(graphs_next)
            h_dot = (h_next - h) / self._env.dt
            loss_h_dot = torch.mean(torch.relu(-h_dot - alpha * h + eps))
            acc_h_dot = torch.mean(
                torch.greater_equal(h_dot + alpha * h, 0).type_as(h_dot))

            loss_action = torch.mean(torch.square(actions).sum(dim=1))

            loss = (self.params["loss_unsafe_coef"] * loss_unsafe +
                    self.params["loss_safe_coef"] * loss_safe +
                    self.params["loss_h_dot_coef"] * loss_h_dot +
                    self.params["loss_action_coef"] * loss_action)

            self.optim_cbf.zero_grad(set_to_none=True)
            self.optim_actor.zero_grad(set_to_none=True)
            loss.backward()
            if self.grad_sync is not None:
                self.grad_sync()
            torch.nn.utils.clip_grad_norm_(self.cbf.parameters(), 1e-3)
            torch.nn.utils.clip_grad_norm_(self.actor.parameters(), 1e-3)
            self.optim_cbf.step()
            self.optim_actor.step()

            logs.append(torch.stack([
                loss_unsafe.detach(), loss_safe.detach(), loss_h_dot.detach(),
                loss_action.detach(), acc_unsafe.detach(), acc_safe.detach(),
                acc_h_dot.detach()]))

        # mean-reduce across DP ranks so rank-0 curves reflect the global
        # batch (same semantics as GCBF._update_tail)
        log_stack = torch.stack(logs).detach()
        import torch.distributed as dist
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(log_stack, op=dist.ReduceOp.SUM)
            log_stack /= dist.get_world_size()
        log_vals = log_stack.cpu()
        if writer is not None:
            names = ("loss/unsafe", "loss/safe", "loss/derivative",
                     "loss/action", "acc/unsafe", "acc/safe", "acc/derivative")
            for i_inner in range(inner_iter):
                t = step * inner_iter + i_inner
                for k, name in enumerate(names):
                    writer.add_scalar(name, float(log_vals[i_inner, k]), t)

        # refresh bf16 weight mirrors (captured rollout graphs read them
        # by address; Python is skipped during replay)
        from ..nn.fused import sync_bf16_mirrors
        sync_bf16_mirrors(self.actor)
        sync_bf16_mirrors(self.cbf)

        self.memory.merge(self.buffer)
        self.buffer.clear()
        return {
            "acc/safe": float(log_vals[-1, 5]),
            "acc/unsafe": float(log_vals[-1, 4]),
            "acc/derivative": float(log_vals[-1, 6]),
        }

    def apply(self, data: GraphBatch, rand: Optional[float] = 0) -> Tensor:
        """Reference MACBF.apply (macbf.py:213-239) runs Adam over a detached
        action tensor whose grad never populates, so the loop observably
        returns the raw actor action (possibly after an early loss==0 break).
        Reproduce the observable behavior directly."""
        with torch.no_grad():
            return self.actor(data)
