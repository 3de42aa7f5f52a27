"""DubinsCar: unicycle-dynamics cars with moving point obstacles.

Behavioral equivalent of the reference DubinsCar (gcbf/env/dubins_car.py):
state [x, y, θ, v], action [ω/10, a], obstacles as graph nodes (node feature
x=1), dense radius graph with agent-only receivers and optional k-nearest
cap, PID-style reference controller.

The pybullet demo modes (demo_0/1/3: LiDAR point clouds, URDF rendering)
require pybullet and are gated behind its availability; train/test/demo_2
paths are fully native.

Reference quirks reproduced on purpose (parity):
* ``dynamics`` clamps the *position* speed via clamp(v, max=limit) but the
  acceleration-zeroing for over-speed agents is a chained-indexing no-op in
  the reference (gcbf/env/dubins_car.py:122-124 writes into a copy), so no
  zeroing happens here either.
* mask diagonal offsets mirror the reference exactly (e.g. unsafe_mask adds
  4r+1 on the diagonal while thresholding at 2r/3r).
* ``u_ref`` measures progress against the env's *current* goal even for
  replayed graphs (gcbf/env/dubins_car.py:764-767).
"""
from __future__ import annotations

from typing import Optional, Tuple


import torch
from torch import Tensor

from .. import ops
from ..graph import GraphBatch
from .simple_car import SimpleCar
from .utils import plot_graph, fig_to_rgb_array, rejection_sample_positions


class DubinsCar(SimpleCar):

    def __init__(self, num_agents: int, device: torch.device, dt: float = 0.03,
                 params: Optional[dict] = None,
                 max_neighbors: Optional[int] = None):
        super().__init__(num_agents, device, dt, params, max_neighbors)
        self._num_obs = self._params["num_obs"]
        self._params["obs_len_max"] = self._params["area_size"] / 8.0
        self._obs: Optional[Tensor] = None  # obstacle states (n_obs, 4)

    @property
    def max_episode_steps(self) -> int:
        if self._mode == "train":
            return 500
        if self._mode in ("test", "demo_2", "demo_1"):
            return 2500
        return 2000  # demo_0 / demo_3

    @property
    def default_params(self) -> dict:
        # reference gcbf/env/dubins_car.py:87-100
        return {
            "max_distance": 4.0,
            "area_size": 4.0,
            "car_radius": 0.05,
            "dist2goal": 0.05,
            "comm_radius": 1.0,
            "obs_point_r": 0.05,
            "obs_len_max": 0.5,
            "speed_limit": 0.8,
            "obs_speed_limit": 0.2,
            "num_obs": 0,
        }

    @property
    def state_dim(self) -> int:
        return 4

    @property
    def edge_dim(self) -> int:
        return 5

    # -------------------------------------------------------------- dynamics
    def dynamics(self, data: GraphBatch, u) -> Tensor:
        # reference gcbf/env/dubins_car.py:110-132
        if not torch.is_tensor(u):
            raise NotImplementedError  # same as the reference (:112)
        agent_mask = data.agent_mask
        s = data.states
        sl = self._params["speed_limit"]
        v_capped = torch.clamp(s[:, 3], max=sl)
        xdot = torch.stack([
            v_capped * torch.cos(s[:, 2]),
            v_capped * torch.sin(s[:, 2]),
            torch.zeros_like(s[:, 2]),
            torch.zeros_like(s[:, 3]),
        ], dim=1)
        # agent controls: θ̇ = 10·u₀, v̇ = u₁ (autograd-friendly masked write)
        n_nodes = s.shape[0]
        ctrl = torch.zeros(n_nodes, 2, dtype=s.dtype, device=s.device)
        if agent_mask is None:
            ctrl = torch.stack([u[:, 0] * 10, u[:, 1]], dim=1)
        elif data.agent_index is not None:  # capture-safe integer scatter
            ctrl = ctrl.index_copy(
                0, data.agent_index,
                torch.stack([u[:, 0] * 10, u[:, 1]], dim=1))
        else:
            ctrl = ctrl.masked_scatter(
                agent_mask.unsqueeze(1).expand(-1, 2),
                torch.stack([u[:, 0] * 10, u[:, 1]], dim=1))
        xdot = torch.cat([xdot[:, :2], ctrl], dim=1)
        # NOTE: the reference's over-speed acceleration zeroing
        # (dubins_car.py:122-124) writes into an advanced-indexing copy and is
        # a no-op; reproduced by not zeroing.

        # freeze agents that reached their goal — single-graph only: the shape
        # check excludes batches exactly like the reference's
        # (gcbf/env/dubins_car.py:126-132)
        if self._obs is not None and \
                s.shape[0] == self.num_agents + self._obs.shape[0]:
            am = agent_mask if agent_mask is not None else \
                torch.ones(n_nodes, dtype=torch.bool, device=s.device)
            agent_states = s[am].reshape(-1, self.num_agents, self.state_dim)
            reach = torch.less(
                torch.norm(agent_states[..., :2] - self._goal[:, :2], dim=-1),
                self._params["dist2goal"]).reshape(-1)
            keep = torch.logical_not(reach).to(s.dtype).unsqueeze(1)
            frozen = xdot[am] * keep
            xdot = xdot.masked_scatter(
                am.unsqueeze(1).expand(-1, self.state_dim), frozen)
        return xdot

    # ----------------------------------------------------------------- reset
    def reset(self) -> GraphBatch:
        self._t = 0
        side = self._params["area_size"]
        r = self._params["car_radius"]
        obs_r = self._params["obs_point_r"]

        if self._mode not in ("train", "test", "demo_2"):
            raise RuntimeError(
                f"mode {self._mode}: pybullet demo modes need the optional "
                f"pybullet dependency (not available in this build)")

        # obstacles: uniform positions, random heading/speed
        # (reference gcbf/env/dubins_car.py:392-401)
        obs_pos = torch.rand(self._num_obs, 2) * side
        obs = torch.rand(self._num_obs, self.state_dim)
        obs[:, :2] = obs_pos
        obs[:, 2] *= torch.pi * 2
        obs[:, 3] *= self._params["obs_speed_limit"]
        self._obs = obs.to(self.device)
        obs_pos = obs_pos.to(self.device)

        pos = rejection_sample_positions(
            self.num_agents, 2, side, 4 * r,
            avoid=obs_pos.cpu(), avoid_dist=2 * r + 2 * obs_r)
        if self._mode == "demo_2":
            goals2d = self._sample_goals_near(pos, 5 * r)
        else:
            goals2d = rejection_sample_positions(
                self.num_agents, 2, side, 5 * r,
                avoid=obs_pos.cpu(), avoid_dist=2 * r + 2 * obs_r)
        pos = pos.to(self.device)
        goals2d = goals2d.to(self.device)

        states = torch.cat(
            [pos, torch.zeros(self.num_agents, 2, device=self.device)], dim=1)
        states[:, 2] = torch.rand(self.num_agents, device=self.device) \
            * 2 * torch.pi - torch.pi
        goals = torch.cat(
            [goals2d, torch.zeros(self.num_agents, 2, device=self.device)],
            dim=1)
        goals[:, 2] = torch.rand(self.num_agents, device=self.device) \
            * 2 * torch.pi - torch.pi
        self._goal = goals

        data = self._build_data(states)
        self._data = self.add_communication_links(data)

        self._set_plot_limits(
            torch.cat([states[:, :2], goals[:, :2], obs_pos], dim=0))
        return self._data

    def _build_data(self, agent_states: Tensor) -> GraphBatch:
        n_obs = self._obs.shape[0]
        x = torch.cat([
            torch.zeros(self.num_agents, self.node_dim),
            torch.ones(n_obs, self.node_dim)], dim=0).type_as(agent_states)
        states = torch.cat([agent_states, self._obs], dim=0)
        agent_mask = torch.zeros(self.num_agents + n_obs, dtype=torch.bool,
                                 device=self.device)
        agent_mask[:self.num_agents] = True
        g = GraphBatch(x=x, pos=states[:, :2], states=states,
                       agent_mask=agent_mask)
        g.agents_first_n = self.num_agents
        return g

    # ------------------------------------------------------------------ step
    def step(self, action: Tensor) -> Tuple[GraphBatch, Tensor, bool, dict]:
        self._t += 1
        # fused single-kernel path on GPU (ops/hip/env_step.hip)
        out = ops.env_step_fused(
            "dubins", self._data.states, self._goal, action, self.dt,
            self._params["car_radius"], self._params["speed_limit"],
            self._params["dist2goal"], 2.0)
        if out is not None:
            return self._finish_fused_step(out)

        # reference gcbf/env/dubins_car.py:522-615
        reward_action = -torch.norm(action, dim=1).sum() * 0.01
        action = action + self._step_u_ref()
        lower_lim, upper_lim = self.action_lim
        action = torch.clamp(action, lower_lim, upper_lim)
        am = self._data.agent_mask
        prev_reach = torch.less(
            torch.norm(self.data.states[am, :2] - self._goal[:, :2], dim=1),
            self._params["dist2goal"])
        with torch.no_grad():
            state = self.forward(self._data, action)

        data = GraphBatch(
            x=self._data.x, pos=state[:, :2], states=state, agent_mask=am)
        data.agents_first_n = self.num_agents
        self._obs = state[~am]
        self._data = self.add_communication_links(data)

        time_up = self._t >= self.max_episode_steps
        reach = torch.less(
            torch.norm(self.data.states[am, :2] - self._goal[:, :2], dim=1),
            self._params["dist2goal"])
        done = bool(time_up or reach.all())

        collision = self.collision_mask(data)
        reward_step = -0.0001
        reward_collision = -collision.int() * 0.1
        reward_reach = (reach.int() - prev_reach.int()).int() * 10
        reward = reward_reach + reward_collision + reward_step + reward_action

        safe = 1.0 - collision.sum() / self.num_agents
        return self.data, reward.detach(), done, {
            "reach": reach, "collision": collision, "safe": safe}

    def forward_graph(self, data: GraphBatch, action: Tensor) -> GraphBatch:
        # reference gcbf/env/dubins_car.py:617-635
        action = action + self.u_ref(data)
        lower_lim, upper_lim = self.action_lim
        action = torch.clamp(action, lower_lim, upper_lim)
        state = self.forward(data, action)
        return data.replace(
            edge_attr=self.edge_attr(state, data.edge_index),
            pos=state[:, :2],
            states=state,
        )

    # ----------------------------------------------------------------- graph
    def edge_attr(self, state: Tensor, edge_index: Tensor) -> Tensor:
        # relative [x, y, θ, v·cosθ, v·sinθ] (gcbf/env/dubins_car.py:724-728)
        edge_info = torch.cat([
            state[:, :3],
            (state[:, 3] * torch.cos(state[:, 2])).unsqueeze(1),
            (state[:, 3] * torch.sin(state[:, 2])).unsqueeze(1)], dim=1)
        return edge_info.index_select(0, edge_index[0]) - \
            edge_info.index_select(0, edge_index[1])

    # dense builder with agent-only receivers comes from SimpleCar's
    # add_communication_links; the fused GPU kernel computes the 5-dim
    # dubins edge_attr in its fill pass
    _attr_kind = ops.ATTR_DUBINS

    @property
    def state_lim(self) -> Tuple[Tensor, Tensor]:
        low = torch.tensor([self._xy_min[0], self._xy_min[1], -10, -10],
                           device=self.device)
        high = torch.tensor([self._xy_max[0], self._xy_max[1], 10, 10],
                            device=self.device)
        return low, high

    @property
    def action_lim(self) -> Tuple[Tensor, Tensor]:
        upper = torch.ones(2, device=self.device) * 2.0
        return -upper, upper

    # ----------------------------------------------------------------- u_ref
    def u_ref(self, data: GraphBatch) -> Tensor:
        # PID heading/accel controller (reference gcbf/env/dubins_car.py:764-816)
        am = data.agent_mask
        if data.agent_index is not None:   # capture-safe integer indexing
            states = data.states.index_select(0, data.agent_index)
        elif am is not None:
            states = data.states[am]
        else:
            states = data.states
        states = states.reshape(-1, self.num_agents, self.state_dim)
        diff = (states - self._goal).reshape(-1, self.state_dim)
        states = states.reshape(-1, self.state_dim)

        k_omega, k_v, k_a = 0.2, 0.3, 0.6

        dist = torch.norm(diff[:, :2], dim=-1)
        theta_t = (torch.acos(torch.clamp(-diff[:, 0] / (dist + 1e-4), -1, 1))
                   * torch.sign(-diff[:, 1])) % (2 * torch.pi)
        theta = states[:, 2] % (2 * torch.pi)
        theta_diff = theta_t - theta
        agent_dir = torch.stack([torch.cos(theta), torch.sin(theta)], dim=-1)
        inner = (-diff[:, :2] * agent_dir).sum(dim=-1)
        theta_between = torch.acos(torch.clamp(inner / (dist + 1e-4), -1, 1))

        anti = torch.logical_and(theta_diff < torch.pi, theta_diff >= 0)
        small = theta <= torch.pi
        clock_cond = torch.logical_and(theta_diff > -torch.pi, theta_diff <= 0)
        sign = torch.where(
            small,
            torch.where(anti, torch.ones_like(theta), -torch.ones_like(theta)),
            torch.where(clock_cond, -torch.ones_like(theta),
                        torch.ones_like(theta)))
        omega = torch.clamp(sign * k_omega * theta_between, -5.0, 5.0)

        a = -k_a * states[:, 3] + k_v * dist
        sl = self._params["speed_limit"]
        a = torch.where(states[:, 3] > sl, torch.clamp(a, max=0.0), a)
        a = torch.where(states[:, 3] < -sl, torch.clamp(a, min=0.0), a)

        return torch.stack([omega, a], dim=-1).reshape(-1, self.action_dim)

    # ----------------------------------------------------------------- masks
    def _pairwise_agent_rows(self, data: GraphBatch, diag_offset: float
                             ) -> Tuple[Tensor, Tensor]:
        """pos-diff (B, n_agents, N, 2) and distance with diag offset on the
        agent-self entries."""
        B = data.num_graphs
        N = data.nodes_per_graph
        n = self.num_agents
        sv = data.states.view(B, N, -1)
        pd = sv[:, :n, :2].unsqueeze(2) - sv[:, :, :2].unsqueeze(1)
        dist = pd.norm(dim=-1)
        eye = torch.eye(N, device=data.device, dtype=dist.dtype)[:n]
        return pd, dist + eye * diag_offset

    _env_kind = ops.ENV_DUBINS

    def safe_mask(self, data: GraphBatch, return_edge: bool = False) -> Tensor:
        r = self._params["car_radius"]
        if return_edge:
            return data.edge_attr[:, :2].norm(dim=-1) > 4 * r
        m = self._fused_mask(data, "safe")
        if m is not None:
            return m
        # diag offset 4r+1, threshold 3r (reference dubins_car.py:835-838)
        _, dist = self._pairwise_agent_rows(data, 4 * r + 1)
        return (dist > 3 * r).min(dim=2)[0].reshape(-1).bool()

    def unsafe_mask(self, data: GraphBatch, return_edge: bool = False) -> Tensor:
        r = self._params["car_radius"]
        if return_edge:
            return data.edge_attr[:, :2].norm(dim=-1) < 2 * r
        m = self._fused_mask(data, "unsafe")
        if m is not None:
            return m
        pd, dist = self._pairwise_agent_rows(data, 4 * r + 1)
        collision = (dist < 2 * r).max(dim=2)[0]

        warn_zone = dist < 3 * r
        pos_vec = -(pd / (pd.norm(dim=-1, keepdim=True) + 1e-4))
        B, N = pd.shape[0], pd.shape[2]
        n = self.num_agents
        sv = data.states.view(B, N, -1)
        theta = sv[:, :n, 2]
        theta_vec = torch.stack([torch.cos(theta), torch.sin(theta)],
                                dim=-1).unsqueeze(2)
        inner = (pos_vec * theta_vec).sum(dim=-1)
        thr = torch.cos(torch.asin(2 * r / (dist + 1e-7)))
        unsafe = torch.logical_and(inner > thr, warn_zone).max(dim=2)[0]
        return torch.logical_or(collision, unsafe).reshape(-1).bool()

    def collision_mask(self, data: GraphBatch) -> Tensor:
        r = self._params["car_radius"]
        if self._mode not in ("train", "test", "demo_1", "demo_2"):
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
        if self._mode not in ("train", "test", "demo_2"):
            raise RuntimeError("pybullet demo rendering unavailable")
        return_tuple = True
        if traj is None:
            traj = (self.data,)
            return_tuple = False
        r = self._params["car_radius"]
        gif = []
        for data in traj:
            if ax is None:
                fig, ax_ = plt.subplots(1, 1, figsize=(12, 10), dpi=100)
            else:
                ax_ = ax
            plot_graph(ax_, data, radius=r, color="#FF8C00", with_label=False,
                       plot_edge=plot_edge, alpha=0.8)
            goal_data = GraphBatch(x=self._goal, pos=self._goal[:, :2],
                                   states=self._goal)
            plot_graph(ax_, goal_data, radius=r, color="#3CB371",
                       with_label=True, plot_edge=False, alpha=0.8)
            x_int = self._xy_max[0] - self._xy_min[0]
            y_int = self._xy_max[1] - self._xy_min[1]
            ax_.set_xlim(self._xy_min[0], self._xy_min[0] + max(x_int, y_int))
            ax_.set_ylim(self._xy_min[1], self._xy_min[1] + max(x_int, y_int))
            plt.axis("off")
            plt.tight_layout()
            if return_ax:
                return ax_
            gif.append(fig_to_rgb_array(fig))
            plt.close(fig)
        return tuple(gif) if return_tuple else gif[0]
