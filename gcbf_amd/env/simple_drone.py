"""SimpleDrone: 3D linear drone dynamics with static point obstacles.

Behavioral equivalent of the reference SimpleDrone (gcbf/env/simple_drone.py):
6D state [x,y,z,vx,vy,vz], 3D action, damped linear dynamics ẋ = Ax + Bu,
LQR reference controller, static obstacles as graph nodes.

Reference quirks reproduced:
* ``reset`` always spawns ``num_agents`` obstacles, ignoring ``num_obs``
  (gcbf/env/simple_drone.py:128-135).
* the unsafe velocity-cone uses [vx/v, vy/v, vz] — vz NOT normalized
  (gcbf/env/simple_drone.py:431-434).
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch
from torch import Tensor

from ..graph import GraphBatch
from .base import MultiAgentEnv
from .simple_car import SimpleCar
from .utils import lqr, plot_graph_3d, fig_to_rgb_array, \
    rejection_sample_positions


class SimpleDrone(MultiAgentEnv):

    def __init__(self, num_agents: int, device: torch.device, dt: float = 0.03,
                 params: Optional[dict] = None,
                 max_neighbors: Optional[int] = None):
        super().__init__(num_agents, device, dt, params, max_neighbors)
        self._K: Optional[Tensor] = None
        self._goal: Optional[Tensor] = None
        self._num_obs = self._params["num_obs"]
        self._obs: Optional[Tensor] = None
        self._xyz_min = np.array([0, 0, 0])
        self._xyz_max = np.ones(3) * self._params["area_size"]

    @property
    def state_dim(self) -> int:
        return 6

    @property
    def node_dim(self) -> int:
        return 4

    @property
    def edge_dim(self) -> int:
        return 6

    @property
    def action_dim(self) -> int:
        return 3

    @property
    def max_episode_steps(self) -> int:
        return 500 if self._mode == "train" else 2000

    @property
    def default_params(self) -> dict:
        # reference gcbf/env/simple_drone.py:71-82
        return {
            "area_size": 2.0,
            "speed_limit": 0.6,
            "drone_radius": 0.05,
            "comm_radius": 0.5,
            "dist2goal": 0.02,
            "obs_point_r": 0.05,
            "obs_len_max": 0.5,
            "max_distance": 4.0,
            "num_obs": 4,
        }

    @property
    def _A(self) -> Tensor:
        A = torch.zeros(6, 6, dtype=torch.float, device=self.device)
        A[0, 3] = 1.0
        A[1, 4] = 1.0
        A[2, 5] = 1.0
        A[3, 3] = -1.1
        A[4, 4] = -1.1
        A[5, 5] = -6.0
        return A

    @property
    def _B(self) -> Tensor:
        B = torch.zeros(6, 3, dtype=torch.float, device=self.device)
        B[3, 0] = 1.1
        B[4, 1] = 1.1
        B[5, 2] = 6.0
        return B

    # -------------------------------------------------------------- dynamics
    def dynamics(self, data: GraphBatch, u) -> Tensor:
        # reference gcbf/env/simple_drone.py:103-120
        if not torch.is_tensor(u):
            # symbolic/numpy overload (cvxpy Expression or numpy array)
            A = self._A.cpu().numpy()
            B = self._B.cpu().numpy()
            return data.states.cpu().detach().numpy() @ A.T + u @ B.T
        am = data.agent_mask
        s = data.states
        xdot = s @ self._A.t()
        mask_f = am.to(s.dtype).unsqueeze(1)
        xdot = xdot * mask_f  # obstacles static
        ctrl = torch.zeros_like(xdot)
        if data.agent_index is not None:    # capture-safe integer scatter
            ctrl = ctrl.index_copy(0, data.agent_index, u @ self._B.t())
        else:
            ctrl = ctrl.masked_scatter(
                am.unsqueeze(1).expand(-1, self.state_dim), u @ self._B.t())
        xdot = xdot + ctrl
        if s.shape[0] == self.num_agents + self._obs.shape[0]:
            agent_states = s[am]
            reach = torch.less(
                torch.norm(agent_states[:, :3] - self._goal[:, :3], dim=1),
                self._params["dist2goal"])
            keep = torch.logical_not(reach).to(s.dtype).unsqueeze(1)
            frozen = xdot[am] * keep
            xdot = xdot.masked_scatter(
                am.unsqueeze(1).expand(-1, self.state_dim), frozen)
        return xdot

    # ----------------------------------------------------------------- reset
    def reset(self) -> GraphBatch:
        self._t = 0
        side = self._params["area_size"]
        r = self._params["drone_radius"]
        obs_r = self._params["obs_point_r"]
        if self._mode not in ("train", "test"):
            raise NotImplementedError

        # reference quirk: always num_agents obstacles (simple_drone.py:128-135)
        obs_pos = torch.rand(self.num_agents, 3) * side
        self._obs = torch.zeros(self.num_agents, self.state_dim,
                                device=self.device)
        self._obs[:, :3] = obs_pos.to(self.device)

        pos = rejection_sample_positions(
            self.num_agents, 3, side, 4 * r,
            avoid=obs_pos, avoid_dist=2 * r + 2 * obs_r)
        goals3d = rejection_sample_positions(
            self.num_agents, 3, side, 4 * r,
            avoid=obs_pos, avoid_dist=2 * r + 2 * obs_r)
        pos = pos.to(self.device)
        goals3d = goals3d.to(self.device)

        states = torch.cat(
            [pos, torch.zeros(self.num_agents, 3, device=self.device)], dim=1)
        goals = torch.cat(
            [goals3d, torch.zeros(self.num_agents, 3, device=self.device)],
            dim=1)
        self._goal = goals

        n_obs = self._obs.shape[0]
        x = torch.cat([
            torch.zeros(self.num_agents, self.node_dim),
            torch.ones(n_obs, self.node_dim)], dim=0).type_as(states)
        agent_mask = torch.zeros(self.num_agents + n_obs, dtype=torch.bool,
                                 device=self.device)
        agent_mask[:self.num_agents] = True
        data = GraphBatch(
            x=x,
            pos=torch.cat([states[:, :3], self._obs[:, :3]], dim=0),
            states=torch.cat([states, self._obs], dim=0),
            agent_mask=agent_mask)
        data.agents_first_n = self.num_agents
        self._data = self.add_communication_links(data)
        return self._data

    # ------------------------------------------------------------------ step
    def _get_K_tensor(self) -> Tensor:
        if self._K is None:
            self.u_ref(self._data)
        return self._K.contiguous()

    _finish_fused_step = SimpleCar._finish_fused_step

    def step(self, action: Tensor) -> Tuple[GraphBatch, Tensor, bool, dict]:
        self._t += 1
        # fused single-kernel path on GPU (ops/hip/env_step.hip)
        out = self._ops.env_step_fused(
            "drone", self._data.states, self._goal, action,
            self._get_K_tensor(), self.dt, self._params["drone_radius"],
            self._params["speed_limit"], self._params["dist2goal"], 10.0)
        if out is not None:
            return self._finish_fused_step(out)

        # reference gcbf/env/simple_drone.py:191-234
        reward_action = -torch.norm(action, dim=1) * 0.001
        action = action + self._step_u_ref()
        lower_lim, upper_lim = self.action_lim
        action = torch.clamp(action, lower_lim, upper_lim)
        am = self._data.agent_mask
        prev_reach = torch.less(
            torch.norm(self.data.states[am, :3] - self._goal[:, :3], dim=1),
            self._params["dist2goal"])
        with torch.no_grad():
            state = self.forward(self._data, action)

        data = GraphBatch(x=self._data.x, pos=state[:, :3], states=state,
                          agent_mask=am)
        data.agents_first_n = self.num_agents
        self._data = self.add_communication_links(data)

        time_up = self._t >= self.max_episode_steps
        reach = torch.less(
            torch.norm(self.data.states[am, :3] - self._goal[:, :3], dim=1),
            self._params["dist2goal"])
        done = bool(time_up or reach.all())

        collision = self.collision_mask(data)
        reward_step = -0.01
        reward_collision = -collision.int()
        reward_reach = (reach.int() - prev_reach.int()) * 10
        reward = reward_reach + reward_collision + reward_step + reward_action

        safe = 1.0 - collision.sum() / self.num_agents
        return self.data, reward.detach(), done, {
            "safe": safe, "reach": reach, "collision": collision}

    def forward_graph(self, data: GraphBatch, action: Tensor) -> GraphBatch:
        action = action + self.u_ref(data)
        lower_lim, upper_lim = self.action_lim
        action = torch.clamp(action, lower_lim, upper_lim)
        state = self.forward(data, action)
        return data.replace(
            edge_attr=self.edge_attr(state, data.edge_index),
            pos=state[:, :3],
            states=state,
        )

    # ----------------------------------------------------------------- graph
    def edge_attr(self, state: Tensor, edge_index: Tensor) -> Tensor:
        return state.index_select(0, edge_index[0]) - \
            state.index_select(0, edge_index[1])

    from .. import ops as _ops
    _attr_kind = _ops.ATTR_DIFF
    _env_kind = _ops.ENV_DRONE
    add_communication_links = SimpleCar.add_communication_links
    add_communication_links_batched = SimpleCar.add_communication_links
    _fused_mask = SimpleCar._fused_mask
    _mask_rows = SimpleCar._mask_rows
    _step_u_ref = SimpleCar._step_u_ref

    @property
    def state_lim(self) -> Tuple[Tensor, Tensor]:
        low = torch.tensor([self._xyz_min[0], self._xyz_min[1],
                            self._xyz_min[2], -10, -10, -10],
                           device=self.device)
        high = torch.tensor([self._xyz_max[0], self._xyz_max[1],
                             self._xyz_max[2], 10, 10, 10],
                            device=self.device)
        return low, high

    @property
    def action_lim(self) -> Tuple[Tensor, Tensor]:
        upper = torch.ones(self.action_dim, device=self.device) * 10.0
        return -upper, upper

    # ----------------------------------------------------------------- u_ref
    def u_ref(self, data: GraphBatch) -> Tensor:
        # reference gcbf/env/simple_drone.py:349-377
        am = data.agent_mask
        if data.agent_index is not None:   # capture-safe integer indexing
            states = data.states.index_select(0, data.agent_index)
        elif am is not None:
            states = data.states[am]
        else:
            states = data.states
        states = states.reshape(-1, self.num_agents, self.state_dim)
        diff = states - self._goal

        if self._K is None:
            A = self._A.cpu().numpy() * self.dt + np.eye(self.state_dim)
            B = self._B.cpu().numpy() * self.dt
            K_np = lqr(A, B, np.eye(self.state_dim), np.eye(self.action_dim))
            self._K = torch.from_numpy(K_np).type_as(data.states)

        action = -torch.einsum("us,bns->bnu", self._K, diff)
        action = action.reshape(-1, self.action_dim)

        # branch-free over-speed penalty (reference simple_drone.py:367-375)
        states = states.reshape(-1, self.state_dim)
        v = states[:, 3:]
        speed = v.norm(dim=1, keepdim=True)
        penalty = torch.relu(speed - self._params["speed_limit"]) * 10
        v_dir = v / speed.clamp(min=1e-12)
        return action - penalty * v_dir

    # ----------------------------------------------------------------- masks
    def _pairwise_agent_rows(self, data: GraphBatch, diag_offset: float
                             ) -> Tuple[Tensor, Tensor]:
        B = data.num_graphs
        N = data.nodes_per_graph
        n = self.num_agents
        sv = data.states.view(B, N, -1)
        pd = sv[:, :n, :3].unsqueeze(2) - sv[:, :, :3].unsqueeze(1)
        dist = pd.norm(dim=-1)
        eye = torch.eye(N, device=data.device, dtype=dist.dtype)[:n]
        return pd, dist + eye * diag_offset

    def safe_mask(self, data: GraphBatch, return_edge: bool = False) -> Tensor:
        r = self._params["drone_radius"]
        if return_edge:
            return data.edge_attr[:, :3].norm(dim=-1) > 4 * r
        m = self._fused_mask(data, "safe")
        if m is not None:
            return m
        _, dist = self._pairwise_agent_rows(data, 4 * r + 1)
        return (dist > 4 * r).min(dim=2)[0].reshape(-1).bool()

    def unsafe_mask(self, data: GraphBatch, return_edge: bool = False) -> Tensor:
        r = self._params["drone_radius"]
        if return_edge:
            return data.edge_attr[:, :3].norm(dim=-1) < 2 * r
        m = self._fused_mask(data, "unsafe")
        if m is not None:
            return m
        pd, dist = self._pairwise_agent_rows(data, 2 * r + 1)
        collision = (dist < 2 * r).max(dim=2)[0]

        warn_zone = dist < 4 * r
        pos_vec = -(pd / (pd.norm(dim=-1, keepdim=True) + 1e-4))
        B, N = pd.shape[0], pd.shape[2]
        n = self.num_agents
        sv = data.states.view(B, N, -1)
        vel = sv[:, :n, 3:6]
        v = vel.norm(dim=-1, keepdim=True) + 1e-5
        # reference quirk: vz not normalized (simple_drone.py:431-434)
        theta_vec = torch.cat([vel[..., 0:1] / v, vel[..., 1:2] / v,
                               vel[..., 2:3]], dim=-1).unsqueeze(2)
        inner = (pos_vec * theta_vec).sum(dim=-1)
        thr = torch.cos(torch.asin(2 * r / (dist + 1e-7)))
        unsafe = torch.logical_and(inner > thr, warn_zone).max(dim=2)[0]
        return torch.logical_or(collision, unsafe).reshape(-1).bool()

    def collision_mask(self, data: GraphBatch) -> Tensor:
        r = self._params["drone_radius"]
        if self._mode not in ("train", "test", "demo_2"):
            raise NotImplementedError
        m = self._fused_mask(data, "collision")
        if m is not None:
            return m
        _, dist = self._pairwise_agent_rows(data, 2 * r + 1)
        return (dist < 2 * r).max(dim=2)[0].reshape(-1).bool()

    # ---------------------------------------------------------------- render
    def render(self, traj=None, return_ax: bool = False, plot_edge: bool = True,
               ax=None):
        import matplotlib.pyplot as plt
        return_tuple = True
        if traj is None:
            traj = (self.data,)
            return_tuple = False
        r = self._params["drone_radius"]
        gif = []
        for data in traj:
            fig = plt.figure(figsize=(10, 10), dpi=80)
            ax_ = fig.add_subplot(projection="3d")
            plot_graph_3d(ax_, data, radius=r, color="#FF8C00",
                          with_label=True, plot_edge=plot_edge, alpha=0.3)
            goal_data = GraphBatch(x=self._goal, pos=self._goal[:, :3],
                                   states=self._goal)
            plot_graph_3d(ax_, goal_data, radius=r, color="#3CB371",
                          with_label=True, plot_edge=False, alpha=0.3)
            unsafe = self.unsafe_mask(data)
            idx = torch.where(unsafe)[0].cpu().numpy()
            ax_.text2D(0., 0.97, f"Collision: {idx}", transform=ax_.transAxes,
                       fontsize=14)
            ax_.set_xlim(self._xyz_min[0], self._xyz_max[0])
            ax_.set_ylim(self._xyz_min[1], self._xyz_max[1])
            ax_.set_zlim(self._xyz_min[2], self._xyz_max[2])
            ax_.set_aspect("equal")
            if return_ax:
                return ax_
            gif.append(fig_to_rgb_array(fig))
            plt.close(fig)
        return tuple(gif) if return_tuple else gif[0]
