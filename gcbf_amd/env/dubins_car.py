"""DubinsCar: unicycle-dynamics cars with moving point obstacles.

Behavioral equivalent of the reference DubinsCar (gcbf/env/dubins_car.py):
state [x, y, θ, v], action [ω/10, a], obstacles as graph nodes (node feature
x=1), dense radius graph with agent-only receivers and optional k-nearest
cap, PID-style reference controller.

The demo modes (demo_0/1/3) are fully native: the reference's pybullet
LiDAR/contact/camera machinery (gcbf/env/dubins_car.py:55-382, 637-722,
884-923) is replaced by an analytic oriented-box world
(gcbf_amd/env/demo_world.py) — 32-ray slab-test LiDAR with agent/goal
occluders, SDF contact checks, kinematic box motion, matplotlib top-down
rendering — vectorized in torch and device-aware (runs on the MI355X).

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
from .demo_world import BoxWorld
from .simple_car import SimpleCar
from .utils import plot_graph, fig_to_rgb_array, rejection_sample_positions

N_LIDAR_RAYS = 32  # reference gcbf/env/dubins_car.py:313


class DubinsCar(SimpleCar):

    def __init__(self, num_agents: int, device: torch.device, dt: float = 0.03,
                 params: Optional[dict] = None,
                 max_neighbors: Optional[int] = None):
        super().__init__(num_agents, device, dt, params, max_neighbors)
        self._num_obs = self._params["num_obs"]
        self._params["obs_len_max"] = self._params["area_size"] / 8.0
        self._obs: Optional[Tensor] = None  # obstacle states (n_obs, 4)
        # demo modes: native box world (analytic LiDAR + contacts — the
        # reference used pybullet here, gcbf/env/dubins_car.py:55-382)
        self._world: Optional[BoxWorld] = None

    def demo(self, idx: int):
        super().demo(idx)
        if idx == 3:
            # demo_3 uses small moving obstacles (reference :74-75)
            self._params["obs_len_max"] = self._params["car_radius"] * 2

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

        if self._mode in ("demo_0", "demo_1", "demo_3"):
            return self._reset_demo()

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

    # ------------------------------------------------------- demo modes
    # Native replacements for the reference's pybullet demo paths
    # (gcbf/env/dubins_car.py:55-382, 637-722, 884-923): oriented-box
    # obstacles, analytic 32-ray LiDAR (occluded by other agents/goals),
    # SDF contact checks, kinematic box motion, matplotlib rendering.
    def _init_obs_world(self) -> BoxWorld:
        """Obstacle layout of _init_obs_bullet (reference :147-209):
        alternating vertical/horizontal boxes in an intersection corridor;
        demo_3 adds 4 large static corner blocks and rejects overlapping
        small boxes."""
        import numpy as np
        p = self._params
        area = p["area_size"]
        world = BoxWorld(self.device)
        for i in range(self._num_obs):
            center = np.random.rand(3) * area / 4
            if i % 2 == 0:
                center[0] += area / 2 - area / 8
                center[1] *= 4
                theta = float(np.pi / 2)
            else:
                center[1] += area / 2 - area / 8
                center[0] *= 4
                theta = 0.0
            length = float(np.random.rand()) * p["obs_len_max"] + area / 80
            width = area / 80
            if self._mode == "demo_3" and world.num_boxes:
                # reject boxes overlapping an existing one (closest points
                # within obs_point_r, reference :177-186); the reference's
                # for-loop does not retry rejected indices
                probe = torch.tensor([center[:2]], dtype=torch.float32,
                                     device=self.device)
                # conservative surface gap: center distance minus both
                # half-diagonals
                if float(world.box_distance(probe).min()) < \
                        (length + width) / 2 + p["obs_point_r"]:
                    continue
            speed = float(2 * np.random.rand() - 1) * p["obs_speed_limit"]
            world.add_box(center[:2], (length, width), theta,
                          vel=(theta, speed))
        if self._mode == "demo_3":
            sq = area / 16 * 3
            for cx, cy in ((sq, sq), (sq, area - sq), (area - sq, sq),
                           (area - sq, area - sq)):
                world.add_box((cx, cy), (sq / 2, sq / 4), 0.0, vel=(0.0, 0.0))
        return world

    def _sample_demo_positions(self, kind: str) -> Tensor:
        """Cross/random layout rejection sampling of _init_agent_bullet /
        _init_goal_bullet (reference :211-313)."""
        import numpy as np
        p = self._params
        side = p["area_size"]
        r = p["car_radius"]
        cross = self._mode in ("demo_0", "demo_3")
        min_sep = 4 * r if kind == "agent" else 8 * r
        clearance = 2 * r + 2 * p["obs_point_r"]
        pos = np.zeros((self.num_agents, 2))
        i, guard = 0, 0
        while i < self.num_agents:
            guard += 1
            if guard > 100000:
                raise RuntimeError("demo placement: no valid layout found")
            if cross:
                cand = np.random.rand(2) * side / 8 * 3
                if kind == "agent":
                    if i % 2 == 0:
                        cand[0] += side / 16 * 13 - side / 16 * 3
                    if self._mode == "demo_3" and i % 4 in (3, 0):
                        cand[1] += side / 16 * 13 - side / 16 * 3
                else:
                    if i % 2 == 1:
                        cand[0] += side / 16 * 13
                    cand[1] += side / 16 * 13
                    if self._mode == "demo_3" and i % 4 in (3, 0):
                        cand[1] -= side / 16 * 13
            else:  # demo_1: random placement
                cand = np.random.rand(2) * side
            if i and np.linalg.norm(pos[:i] - cand, axis=1).min() <= min_sep:
                continue
            if self._world.num_boxes:
                probe = torch.tensor(np.array([cand]),
                                     dtype=torch.float32,
                                     device=self.device)
                # surface-to-surface: box SDF minus the circle footprint
                if float(self._world.box_distance(probe).min()) - r \
                        <= clearance:
                    continue
            pos[i] = cand
            i += 1
        return torch.tensor(pos, dtype=torch.float32, device=self.device)

    def _lidar_observe(self, agent_pos: Tensor) -> Tensor:
        """32 rays per agent against the box world, occluded by other
        agents and goal cylinders; hits become the obstacle point cloud
        self._obs = [x, y, heading, speed] (reference _lidar /
        _get_observation_bullet, :313-382).  Returns the (K, 2) points."""
        n = self.num_agents
        dev = self.device
        angles = torch.arange(N_LIDAR_RAYS, device=dev) \
            * (2 * torch.pi / N_LIDAR_RAYS)
        dirs1 = torch.stack([torch.cos(angles), torch.sin(angles)], dim=1)
        hits_pos, hits_vel = [], []
        goal_pos = self._goal[:, :2]
        for a in range(n):
            origins = agent_pos[a].expand(N_LIDAR_RAYS, 2)
            others = torch.cat([agent_pos[:a], agent_pos[a + 1:], goal_pos])
            hit, pts, box = self._world.raycast(
                origins, dirs1, self._params["comm_radius"],
                occluder_centers=others,
                occluder_radius=self._params["car_radius"])
            if hit.any():
                # reference appends in reversed ray order (:330)
                idx = hit.nonzero(as_tuple=True)[0].flip(0)
                hits_pos.append(pts[idx])
                hits_vel.append(self._world.vel[box[idx]])
        if hits_pos:
            obs_pos = torch.cat(hits_pos)
            obs_vel = torch.cat(hits_vel)
        else:
            obs_pos = torch.zeros(0, 2, device=dev)
            obs_vel = torch.zeros(0, 2, device=dev)
        self._obs = torch.cat([obs_pos, obs_vel], dim=1)
        return obs_pos

    def _reset_demo(self) -> GraphBatch:
        import numpy as np
        self._world = self._init_obs_world() \
            if self._mode in ("demo_0", "demo_3") else BoxWorld(self.device)
        agent_pos = self._sample_demo_positions("agent")
        theta = torch.tensor(
            np.arctan2(np.sin(np.random.rand(self.num_agents) * 2 * np.pi),
                       np.cos(np.random.rand(self.num_agents) * 2 * np.pi)),
            dtype=torch.float32, device=self.device)
        states = torch.cat([
            agent_pos, theta.unsqueeze(1),
            torch.zeros(self.num_agents, 1, device=self.device)], dim=1)
        goal_pos = self._sample_demo_positions("goal")
        if self._mode == "demo_3":  # reference shuffles goals (:311-312)
            goal_pos = goal_pos[torch.randperm(self.num_agents,
                                               device=self.device)]
        self._goal = torch.cat(
            [goal_pos, torch.zeros(self.num_agents, 2, device=self.device)],
            dim=1)
        obs_pos = self._lidar_observe(agent_pos)
        data = self._build_data(states)
        self._data = self.add_communication_links(data)
        self._set_plot_limits(
            torch.cat([states[:, :2], goal_pos, obs_pos], dim=0))
        return self._data

    def _step_demo(self, action: Tensor):
        """Demo-mode step (reference :525-615): integrate agents on the
        current graph, advance the boxes kinematically, re-scan LiDAR,
        rebuild the graph from scratch (node count varies with hits)."""
        self._t += 1
        reward_action = -torch.norm(action, dim=1).sum() * 0.01
        action = action + self._step_u_ref()
        lower_lim, upper_lim = self.action_lim
        action = torch.clamp(action, lower_lim, upper_lim)
        am = self._data.agent_mask
        sel = am if am is not None else slice(None)
        prev_reach = torch.less(
            torch.norm(self.data.states[sel, :2] - self._goal[:, :2], dim=1),
            self._params["dist2goal"])
        with torch.no_grad():
            state = self.forward(self._data, action)
        agent_state = state[sel]

        if self._mode in ("demo_0", "demo_3"):
            self._world.advance(self.dt)
            self._lidar_observe(agent_state[:, :2])
        else:  # demo_1: no boxes; keep the (empty) point cloud
            self._lidar_observe(agent_state[:, :2])
        data = self._build_data(agent_state)
        self._data = self.add_communication_links(data)

        time_up = self._t >= self.max_episode_steps
        reach = torch.less(
            torch.norm(agent_state[:, :2] - self._goal[:, :2], dim=1),
            self._params["dist2goal"])
        done = bool(time_up or reach.all())
        collision = self.collision_mask(self._data)
        reward = (reach.int() - prev_reach.int()).int() * 10 \
            - collision.int() * 0.1 - 0.0001 + reward_action
        safe = 1.0 - collision.sum() / self.num_agents
        return self.data, reward.detach(), done, {
            "reach": reach, "collision": collision, "safe": safe}

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
        if self._mode in ("demo_0", "demo_1", "demo_3"):
            return self._step_demo(action)
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
        if self._mode in ("demo_0", "demo_3"):
            # agent-agent circles PLUS contact with the box obstacles
            # (reference :900-920 used pybullet getClosestPoints over the
            # env's CURRENT bodies for every graph in a batch — mirrored:
            # the box term uses the env's current agent positions)
            B = data.num_graphs
            N = data.nodes_per_graph
            n = self.num_agents
            sv = data.states.view(B, N, -1)[:, :n, :2]
            pd = sv.unsqueeze(2) - sv.unsqueeze(1)
            dist = pd.norm(dim=-1) + torch.eye(
                n, device=data.device) * (2 * r + 1)
            coll_agent = (dist < 2 * r).max(dim=2)[0].reshape(-1)
            am = self._data.agent_mask
            cur = self._data.states[am if am is not None else slice(None), :2]
            coll_box = (self._world.min_distance(cur) - r <= 0) \
                if self._world is not None and self._world.num_boxes \
                else torch.zeros(n, dtype=torch.bool, device=data.device)
            return torch.logical_or(
                coll_agent, coll_box.repeat(B)).bool()
        m = self._fused_mask(data, "collision")
        if m is not None:
            return m
        _, dist = self._pairwise_agent_rows(data, 2 * r + 1)
        return (dist < 2 * r).max(dim=2)[0].reshape(-1).bool()

    # ---------------------------------------------------------------- render
    def _render_demo(self, plot_edge: bool = True):
        """Top-down matplotlib frame of the demo world: boxes, agents,
        goals, LiDAR hit points and communication edges — the functional
        equivalent of the reference's pybullet camera image
        (gcbf/env/dubins_car.py:637-722; documented deviation: matplotlib
        instead of a GL render)."""
        import matplotlib.pyplot as plt
        from matplotlib.patches import Circle, Rectangle
        from matplotlib.transforms import Affine2D
        import numpy as np
        r = self._params["car_radius"]
        fig, ax = plt.subplots(1, 1, figsize=(12, 10), dpi=100)
        w = self._world
        for b in range(w.num_boxes):
            cx, cy = w.centers[b].tolist()
            hx, hy = w.half[b].tolist()
            th = float(w.theta[b])
            rect = Rectangle((cx - hx, cy - hy), 2 * hx, 2 * hy,
                             color="#8B0000", alpha=0.9)
            rect.set_transform(
                Affine2D().rotate_around(cx, cy, th) + ax.transData)
            ax.add_patch(rect)
        data = self.data
        am = data.agent_mask
        states = data.states[am if am is not None else slice(None)]
        pts = data.states[~am][:, :2].cpu().numpy() if am is not None \
            else np.zeros((0, 2))
        if len(pts):
            ax.scatter(pts[:, 0], pts[:, 1], s=12, c="#EEA333", zorder=3)
        for i in range(self.num_agents):
            x, y, th = float(states[i, 0]), float(states[i, 1]), \
                float(states[i, 2])
            ax.add_patch(Circle((x, y), r, color="#FF8C00", alpha=0.8,
                                zorder=4))
            ax.plot([x, x + r * np.cos(th)], [y, y + r * np.sin(th)],
                    c="k", lw=1, zorder=5)
            gx, gy = float(self._goal[i, 0]), float(self._goal[i, 1])
            ax.add_patch(Circle((gx, gy), r, color="#3CB371", alpha=0.8,
                                zorder=4))
        if plot_edge and data.num_edges:
            ei = data.edge_index.cpu().numpy()
            pos = data.pos.cpu().numpy()
            for k in range(ei.shape[1]):
                a, b = ei[0, k], ei[1, k]
                ax.plot([pos[a, 0], pos[b, 0]], [pos[a, 1], pos[b, 1]],
                        c="#1A5276", lw=0.5, alpha=0.5, zorder=2)
        x_int = self._xy_max[0] - self._xy_min[0]
        y_int = self._xy_max[1] - self._xy_min[1]
        ax.set_xlim(self._xy_min[0], self._xy_min[0] + max(x_int, y_int))
        ax.set_ylim(self._xy_min[1], self._xy_min[1] + max(x_int, y_int))
        ax.set_aspect("equal")
        plt.axis("off")
        plt.tight_layout()
        frame = fig_to_rgb_array(fig)
        plt.close(fig)
        return frame

    def render(self, traj=None, return_ax: bool = False, plot_edge: bool = True,
               ax=None):
        import matplotlib.pyplot as plt
        if self._mode in ("demo_0", "demo_1", "demo_3") and traj is None:
            return self._render_demo(plot_edge=plot_edge)
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
