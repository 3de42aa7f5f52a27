"""SimpleCar: 2D double-integrator multi-agent environment.

Behavioral equivalent of the reference SimpleCar (gcbf/env/simple_car.py):
state [x, y, vx, vy], action [ax, ay], LQR reference controller, dense
radius graph over agents, pairwise safety masks.  All O(N²) mask work is
batched over the whole graph batch in single vectorized ops (the reference
loops over graphs in Python, gcbf/env/simple_car.py:313-327).
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from .. import ops
from ..graph import GraphBatch
from .base import MultiAgentEnv
from .utils import lqr, plot_graph, fig_to_rgb_array, rejection_sample_positions


class SimpleCar(MultiAgentEnv):

    def __init__(self, num_agents: int, device: torch.device, dt: float = 0.03,
                 params: Optional[dict] = None,
                 max_neighbors: Optional[int] = None):
        super().__init__(num_agents, device, dt, params, max_neighbors)
        self._K: Optional[Tensor] = None
        self._goal: Optional[Tensor] = None
        self._xy_min = None
        self._xy_max = None

    # ------------------------------------------------------------------ dims
    @property
    def state_dim(self) -> int:
        return 4

    @property
    def node_dim(self) -> int:
        return 4

    @property
    def edge_dim(self) -> int:
        return 4

    @property
    def action_dim(self) -> int:
        return 2

    @property
    def max_episode_steps(self) -> int:
        return 500 if self._mode == "train" else 2500

    @property
    def default_params(self) -> dict:
        # reference gcbf/env/simple_car.py:67-76
        return {
            "m": 1.0,
            "comm_radius": 1.0,
            "car_radius": 0.05,
            "dist2goal": 0.04,
            "speed_limit": 0.8,
            "max_distance": 4.0,
            "area_size": 4.0,
        }

    # -------------------------------------------------------------- dynamics
    def dynamics(self, data: GraphBatch, u) -> Tensor:
        if not torch.is_tensor(u):
            # symbolic/numpy overload for external CBF-QP use (reference
            # gcbf/env/simple_car.py:80-87 with a cvxpy Expression): works
            # with any object supporting "@" (cvxpy Expression, numpy array)
            x = data.states.cpu().detach().numpy()
            A = np.zeros((self.state_dim, self.state_dim))
            A[0, 2] = 1.0
            A[1, 3] = 1.0
            B = np.array([[1, 0], [0, 1], [0, 0], [0, 0]])
            return x @ A.T + u @ B.T
        x = data.states
        return torch.cat([x[:, 2:], u], dim=1)

    # ----------------------------------------------------------------- reset
    def reset(self) -> GraphBatch:
        self._t = 0
        side = self._params["area_size"]
        r = self._params["car_radius"]
        if self._mode in ("train", "test", "demo_2"):
            pos = rejection_sample_positions(self.num_agents, 2, side, 4 * r)
            if self._mode == "demo_2":
                goals = self._sample_goals_near(pos, 4 * r)
            else:
                goals = rejection_sample_positions(self.num_agents, 2, side, 4 * r)
        else:
            raise ValueError("Reset environment: unknown type of mode!")
        pos = pos.to(self.device)
        goals = goals.to(self.device)

        states = torch.cat([pos, torch.zeros_like(pos)], dim=1)
        self._goal = goals

        data = GraphBatch(x=torch.zeros_like(states), pos=states[:, :2],
                          states=states)
        data = self.add_communication_links(data)
        self._data = data

        self._set_plot_limits(torch.cat([states[:, :2], goals], dim=0))
        return data

    def _sample_goals_near(self, pos: Tensor, min_sep: float) -> Tensor:
        """demo_2 goal sampling: within max_distance of the agent's start
        (reference gcbf/env/simple_car.py:111-121)."""
        side = self._params["area_size"]
        max_d = self._params["max_distance"]
        goals = torch.zeros(self.num_agents, pos.shape[1])
        i = 0
        while i < self.num_agents:
            cand = (torch.rand(pos.shape[1]) * 2 - 1) * max_d + pos[i].cpu()
            if (cand > side).any() or (cand < 0).any():
                continue
            if torch.norm(goals - cand, dim=1).min() <= min_sep:
                continue
            goals[i] = cand
            i += 1
        return goals

    def _set_plot_limits(self, points: Tensor):
        pts = points.detach().cpu().numpy()
        r = self._params["car_radius"]
        xy_min = np.min(pts, axis=0) - r * 5
        xy_max = np.max(pts, axis=0) + r * 5
        max_interval = (xy_max - xy_min).max()
        self._xy_min = xy_min - 0.5 * (max_interval - (xy_max - xy_min))
        self._xy_max = xy_max + 0.5 * (max_interval - (xy_max - xy_min))

    # ------------------------------------------------------------------ step
    def _step_u_ref(self) -> Tensor:
        """u_ref for the env's own step: reuse the value the trainer already
        attached this step (same states, same goal — identical), else
        compute (reference recomputes it each time,
        gcbf/env/simple_car.py:151)."""
        if self._data.u_ref is not None:
            return self._data.u_ref
        return self.u_ref(self._data)

    def _get_K_tensor(self) -> Tensor:
        if self._K is None:
            self.u_ref(self._data)  # lazily builds the LQR gain
        return self._K.contiguous()

    def _finish_fused_step(self, out) -> Tuple[GraphBatch, Tensor, bool, dict]:
        """Assemble the step return from the fused-kernel outputs."""
        new_states, u_ref_next, reward, reach, collision = out
        pos_dim = 3 if self.state_dim == 6 else 2
        data = GraphBatch(
            x=self._data.x, pos=new_states[:, :pos_dim], states=new_states,
            agent_mask=self._data.agent_mask)
        if data.agent_mask is not None:
            data.agents_first_n = self.num_agents
            # keep the obstacle-state view fresh (the eager step path
            # refreshes it; only its shape is consumed today, but stale
            # values are a trap for future readers)
            self._obs = new_states[self.num_agents:]
        self._data = self.add_communication_links(data)
        self._data.u_ref = u_ref_next
        done = bool(self._t >= self.max_episode_steps or reach.all())
        safe = 1.0 - collision.sum() / self.num_agents
        return self._data, reward, done, {
            "safe": safe, "reach": reach, "collision": collision}

    def step(self, action: Tensor) -> Tuple[GraphBatch, Tensor, bool, dict]:
        self._t += 1
        # fused single-kernel path on GPU (ops/hip/env_step.hip)
        out = ops.env_step_fused(
            "car", self._data.states, self._goal, action,
            self._get_K_tensor(), self.dt, self._params["car_radius"],
            self._params["speed_limit"], self._params["dist2goal"], 10.0)
        if out is not None:
            return self._finish_fused_step(out)

        # reference gcbf/env/simple_car.py:146-176.  Rewards/info stay on
        # device (one host sync per step, for `done`); callers that need
        # numpy use .cpu().numpy().
        reward_action = -torch.norm(action, dim=1) * 0.0001
        action = action + self._step_u_ref()
        lower_lim, upper_lim = self.action_lim
        action = torch.clamp(action, lower_lim, upper_lim)
        prev_reach = torch.less(
            torch.norm(self.data.states[:, :2] - self._goal, dim=1),
            self._params["dist2goal"])
        with torch.no_grad():
            state = self.forward(self.data, action)

        data = GraphBatch(x=torch.zeros_like(state), pos=state[:, :2],
                          states=state)
        self._data = self.add_communication_links(data)

        time_up = self._t >= self.max_episode_steps
        reach = torch.less(
            torch.norm(self.data.states[:, :2] - self._goal, dim=1),
            self._params["dist2goal"])
        done = bool(time_up or reach.all())

        collision = self.collision_mask(data)
        reward_step = -0.01
        reward_collision = -collision.int() * 2
        reward_reach = (reach.int() - prev_reach.int()) * 4
        reward = reward_reach + reward_collision + reward_step + reward_action

        safe = 1.0 - collision.sum() / self.num_agents
        return self.data, reward.detach(), done, {
            "safe": safe, "reach": reach, "collision": collision}

    def forward_graph(self, data: GraphBatch, action: Tensor) -> GraphBatch:
        # reference gcbf/env/simple_car.py:178-194
        action = action + self.u_ref(data)
        lower_lim, upper_lim = self.action_lim
        action = torch.clamp(action, lower_lim, upper_lim)
        state = self.forward(data, action)
        return data.replace(
            x=torch.zeros_like(state),
            edge_attr=self.edge_attr(state, data.edge_index),
            pos=state[:, :2],
            states=state,
        )

    # ----------------------------------------------------------------- graph
    def edge_attr(self, state: Tensor, edge_index: Tensor) -> Tensor:
        # edge_attr[e] = state[src] - state[dst] (gcbf/env/simple_car.py:246)
        return state.index_select(0, edge_index[0]) - \
            state.index_select(0, edge_index[1])

    # edge_attr kind for the fused GPU builder (ops/hip/graph_build.hip)
    _attr_kind = ops.ATTR_DIFF

    def add_communication_links(self, data: GraphBatch) -> GraphBatch:
        n_rec = None if data.agent_mask is None else self.num_agents
        edge_index, edge_attr = ops.build_graph(
            data.pos, data.states, n_rec, self._params["comm_radius"],
            self._max_neighbors, data.num_graphs, self._attr_kind,
            self.edge_dim, self.edge_attr)
        data.update(edge_index=edge_index, edge_attr=edge_attr)
        return data

    # uniform node counts make the batched rebuild identical to the per-graph
    # one, so the same call handles both
    add_communication_links_batched = add_communication_links

    # ---------------------------------------------------------------- limits
    @property
    def state_lim(self) -> Tuple[Tensor, Tensor]:
        sl = self._params["speed_limit"]
        low = torch.tensor([self._xy_min[0], self._xy_min[1], -sl, -sl],
                           device=self.device)
        high = torch.tensor([self._xy_max[0], self._xy_max[1], sl, sl],
                            device=self.device)
        return low, high

    @property
    def action_lim(self) -> Tuple[Tensor, Tensor]:
        upper = torch.ones(2, device=self.device) * 10.0
        return -upper, upper

    # ----------------------------------------------------------------- u_ref
    def u_ref(self, data: GraphBatch) -> Tensor:
        # reference gcbf/env/simple_car.py:270-304
        goal = torch.cat([self._goal, torch.zeros_like(self._goal)], dim=1)
        states = data.states.reshape(-1, self.num_agents, self.state_dim)
        diff = states - goal

        if self._K is None:
            A = np.array([[0., 0., 1., 0.],
                          [0., 0., 0., 1.],
                          [0., 0., 0., 0.],
                          [0., 0., 0., 0.]]) * self.dt + np.eye(self.state_dim)
            B = np.array([[0., 0.],
                          [0., 0.],
                          [1., 0.],
                          [0., 1.]]) * self.dt
            K_np = lqr(A, B, np.eye(self.state_dim), np.eye(self.action_dim))
            self._K = torch.from_numpy(K_np).type_as(data.states)

        action = -torch.einsum("us,bns->bnu", self._K, diff)
        action = action.reshape(-1, self.action_dim)

        # speed-limit penalty (branch-free: relu gates the over-speed rows,
        # so no data-dependent host sync — reference simple_car.py:296-302)
        states = states.reshape(-1, self.state_dim)
        v = states[:, 2:]
        speed = v.norm(dim=1, keepdim=True)
        penalty = torch.relu(speed - self._params["speed_limit"]) * 50
        v_dir = v / speed.clamp(min=1e-12)
        return action - penalty * v_dir

    # ----------------------------------------------------------------- masks
    def _pairwise(self, data: GraphBatch, diag_offset: float
                  ) -> Tuple[Tensor, Tensor]:
        """(B, n_rec, N) pos-diff and distance (with diag offset) over the
        batch; rows are receiver agents (== all nodes for SimpleCar)."""
        B = data.num_graphs
        N = data.nodes_per_graph
        sv = data.states.view(B, N, -1)
        pd = sv[:, :, :2].unsqueeze(2) - sv[:, :, :2].unsqueeze(1)  # [b,i,j]
        dist = pd.norm(dim=-1)
        eye = torch.eye(N, device=data.device, dtype=dist.dtype)
        return pd, dist + eye * diag_offset

    _env_kind = ops.ENV_CAR

    def _fused_mask(self, data: GraphBatch, which: str):
        return ops.fused_masks(data.states, data.num_graphs,
                               self._mask_rows(data), self._params.get(
                                   "car_radius",
                                   self._params.get("drone_radius")),
                               self._env_kind, which)

    def _mask_rows(self, data: GraphBatch) -> int:
        # receiver rows per graph: all nodes for SimpleCar, agents otherwise
        if data.agent_mask is None:
            return data.nodes_per_graph
        return self.num_agents

    def safe_mask(self, data: GraphBatch, return_edge: bool = False) -> Tensor:
        r = self._params["car_radius"]
        if return_edge:
            return data.edge_attr[:, :2].norm(dim=-1) > 4 * r
        m = self._fused_mask(data, "safe")
        if m is not None:
            return m
        _, dist = self._pairwise(data, 4 * r + 1)
        return (dist > 4 * r).min(dim=2)[0].reshape(-1).bool()

    def unsafe_mask(self, data: GraphBatch, return_edge: bool = False) -> Tensor:
        r = self._params["car_radius"]
        if return_edge:
            return data.edge_attr[:, :2].norm(dim=-1) < 2 * r
        m = self._fused_mask(data, "unsafe")
        if m is not None:
            return m
        pd, dist = self._pairwise(data, 4 * r + 1)
        collision = (dist < 2 * r).max(dim=2)[0]

        # heading-into-neighbor cone inside the warn zone
        # (reference gcbf/env/simple_car.py:354-365)
        warn_zone = dist < 4 * r
        pos_vec = -(pd / (pd.norm(dim=-1, keepdim=True) + 1e-4))  # i -> j
        B, N = pd.shape[0], pd.shape[1]
        sv = data.states.view(B, N, -1)
        v = sv[:, :, 2:4].norm(dim=-1, keepdim=True) + 1e-5
        theta_vec = (sv[:, :, 2:4] / v).unsqueeze(2)  # [b, i, 1, 2]
        inner = (pos_vec * theta_vec).sum(dim=-1)
        thr = torch.cos(torch.asin(2 * r / (dist + 1e-7)))
        unsafe = torch.logical_and(inner > thr, warn_zone).max(dim=2)[0]
        return torch.logical_or(collision, unsafe).reshape(-1).bool()

    def collision_mask(self, data: GraphBatch) -> Tensor:
        r = self._params["car_radius"]
        m = self._fused_mask(data, "collision")
        if m is not None:
            return m
        _, dist = self._pairwise(data, 2 * r + 1)
        return (dist < 2 * r).max(dim=2)[0].reshape(-1).bool()

    # ---------------------------------------------------------------- render
    def render(self, traj=None, return_ax: bool = False, plot_edge: bool = True,
               ax=None):
        import matplotlib.pyplot as plt
        return_tuple = True
        if traj is None:
            traj = (self.data,)
            return_tuple = False
        r = self._params["car_radius"]
        gif = []
        for data in traj:
            fig, ax_ = plt.subplots(1, 1, figsize=(10, 10), dpi=80)
            if ax is not None:
                ax_ = ax
            plot_graph(ax_, data, radius=r, color="#FF8C00", with_label=True,
                       plot_edge=plot_edge, alpha=0.8)
            goal_data = GraphBatch(x=self._goal, pos=self._goal[:, :2],
                                   states=self._goal)
            plot_graph(ax_, goal_data, radius=r, color="#3CB371",
                       with_label=True, plot_edge=False, alpha=0.8)
            collision = self.collision_mask(data)
            idx = torch.where(collision)[0].cpu().numpy()
            ax_.text(0., 0.97, f"Collision: {idx}", transform=ax_.transAxes,
                     fontsize=14)
            x_int = self._xy_max[0] - self._xy_min[0]
            y_int = self._xy_max[1] - self._xy_min[1]
            ax_.set_xlim(self._xy_min[0], self._xy_min[0] + max(x_int, y_int))
            ax_.set_ylim(self._xy_min[1], self._xy_min[1] + max(x_int, y_int))
            plt.axis("off")
            if return_ax:
                return ax_
            gif.append(fig_to_rgb_array(fig))
            plt.close(fig)
        return tuple(gif) if return_tuple else gif[0]
