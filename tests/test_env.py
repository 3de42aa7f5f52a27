"""Environment semantics tests: each mask/dynamics/u_ref is checked against a
naive per-graph transcription of the reference formulas."""
import math

import pytest
import torch

from gcbf_amd.env import make_env
from gcbf_amd.graph import GraphBatch


def _mk(env_name, n, device="cpu", **params_over):
    env = make_env(env_name, n, torch.device(device))
    if params_over:
        p = env.default_params
        p.update(params_over)
        env = make_env(env_name, n, torch.device(device), params=p)
    env.train()
    return env


# ------------------------------------------------------------------ SimpleCar

def test_simple_car_dims_and_reset():
    env = _mk("SimpleCar", 5)
    data = env.reset()
    assert data.num_nodes == 5
    assert data.states.shape == (5, 4)
    assert (data.states[:, 2:] == 0).all()  # zero initial velocity
    # separation constraint
    d = torch.cdist(data.pos, data.pos) + torch.eye(5) * 10
    assert d.min() > 4 * env.params["car_radius"]
    # goals exist, inside area
    assert env._goal.shape == (5, 2)


def test_simple_car_dynamics():
    env = _mk("SimpleCar", 3)
    data = env.reset()
    u = torch.randn(3, 2)
    xdot = env.dynamics(data, u)
    assert torch.allclose(xdot[:, :2], data.states[:, 2:])
    assert torch.allclose(xdot[:, 2:], u)
    # Euler forward
    nxt = env.forward(data, u)
    assert torch.allclose(nxt, data.states + xdot * env.dt)


def test_simple_car_u_ref_drives_to_goal():
    env = _mk("SimpleCar", 4)
    data = env.reset()
    for _ in range(300):
        data.update(u_ref=env.u_ref(data))
        a = torch.zeros(4, 2)
        data, r, done, info = env.step(a)
        if done:
            break
    dist = torch.norm(data.states[:, :2] - env._goal, dim=1)
    # LQR should get close to goals in an uncluttered 4-agent scene
    assert dist.mean() < 0.5


def test_simple_car_masks_naive():
    env = _mk("SimpleCar", 6)
    data = env.reset()
    # push two agents together to trigger collision/unsafe
    s = data.states.clone()
    s[1, :2] = s[0, :2] + 0.01
    s[0, 2:] = torch.tensor([0.5, 0.0])
    data = GraphBatch(x=data.x, pos=s[:, :2], states=s)
    data = env.add_communication_links(data)

    r = env.params["car_radius"]
    n = 6
    diff = s.unsqueeze(1) - s.unsqueeze(0)
    pd = diff[:, :, :2]
    dist = pd.norm(dim=2)

    # naive safe
    d_safe = dist + torch.eye(n) * (4 * r + 1)
    naive_safe = (d_safe > 4 * r).min(dim=1).values
    assert torch.equal(env.safe_mask(data), naive_safe)

    # naive collision
    d_col = dist + torch.eye(n) * (2 * r + 1)
    naive_col = (d_col < 2 * r).max(dim=1).values
    assert torch.equal(env.collision_mask(data), naive_col)

    # naive unsafe (collision + heading cone)
    d_u = dist + torch.eye(n) * (4 * r + 1)
    coll = (d_u < 2 * r).max(dim=1).values
    warn = d_u < 4 * r
    pos_vec = -(pd / (pd.norm(dim=2, keepdim=True) + 1e-4))
    v = s[:, 2:].norm(dim=1, keepdim=True) + 1e-5
    tv = (s[:, 2:] / v).unsqueeze(1)
    inner = (pos_vec * tv).sum(dim=2)
    thr = torch.cos(torch.asin(2 * r / (d_u + 1e-7)))
    unsafe = ((inner > thr) & warn).max(dim=1).values
    assert torch.equal(env.unsafe_mask(data), coll | unsafe)
    assert env.unsafe_mask(data)[0]  # agent 0 heads into agent 1


def test_simple_car_batched_masks_match_per_graph():
    env = _mk("SimpleCar", 5)
    graphs = []
    for _ in range(3):
        graphs.append(env.reset())
    batch = GraphBatch.from_list(graphs)
    batched = env.safe_mask(batch)
    per = torch.cat([env.safe_mask(g) for g in graphs])
    assert torch.equal(batched, per)
    batched_u = env.unsafe_mask(batch)
    per_u = torch.cat([env.unsafe_mask(g) for g in graphs])
    assert torch.equal(batched_u, per_u)


def test_simple_car_forward_graph_keeps_topology_and_grad():
    env = _mk("SimpleCar", 4)
    data = env.reset()
    action = torch.zeros(4, 2, requires_grad=True)
    nxt = env.forward_graph(data, action)
    assert torch.equal(nxt.edge_index, data.edge_index)
    # gradient flows from next state back to the action
    nxt.states.sum().backward()
    assert action.grad is not None and action.grad.abs().sum() > 0


def test_simple_car_edge_attr_convention():
    env = _mk("SimpleCar", 4)
    data = env.reset()
    if data.num_edges == 0:
        pytest.skip("no edges in this reset")
    src, dst = data.edge_index
    ref = data.states[src] - data.states[dst]
    assert torch.allclose(data.edge_attr, ref)


def test_simple_car_step_rewards_and_done():
    env = _mk("SimpleCar", 3)
    data = env.reset()
    data, r, done, info = env.step(torch.zeros(3, 2))
    assert r.shape == (3,)
    assert isinstance(done, bool)
    assert set(info) >= {"safe", "reach", "collision"}


# ------------------------------------------------------------------ DubinsCar

def test_dubins_reset_with_obstacles():
    env = _mk("DubinsCar", 6, num_obs=3)
    data = env.reset()
    assert data.num_nodes == 9
    assert data.agent_mask.sum() == 6
    assert (data.x[:6] == 0).all() and (data.x[6:] == 1).all()
    # heading in [-pi, pi)
    th = data.states[:6, 2]
    assert (th >= -math.pi - 1e-6).all() and (th <= math.pi + 1e-6).all()


def test_dubins_dynamics_formulas():
    env = _mk("DubinsCar", 4, num_obs=2)
    data = env.reset()
    u = torch.randn(4, 2) * 0.1
    xdot = env.dynamics(data, u)
    s = data.states
    sl = env.params["speed_limit"]
    v_capped = torch.clamp(s[:, 3], max=sl)
    assert torch.allclose(xdot[:, 0], v_capped * torch.cos(s[:, 2]))
    assert torch.allclose(xdot[:, 1], v_capped * torch.sin(s[:, 2]))
    # agents: θ̇ = 10 u₀, v̇ = u₁ (none at goal in a fresh reset)
    assert torch.allclose(xdot[:4, 2], u[:, 0] * 10)
    assert torch.allclose(xdot[:4, 3], u[:, 1])
    # obstacles keep zero heading/accel derivatives
    assert (xdot[4:, 2:] == 0).all()


def test_dubins_freeze_on_reach():
    env = _mk("DubinsCar", 2, num_obs=0)
    data = env.reset()
    # teleport agent 0 onto its goal
    s = data.states.clone()
    s[0, :2] = env._goal[0, :2]
    data = data.replace(states=s, pos=s[:, :2])
    xdot = env.dynamics(data, torch.ones(2, 2))
    assert (xdot[0] == 0).all()
    assert not (xdot[1] == 0).all()


def test_dubins_masks_naive():
    env = _mk("DubinsCar", 5, num_obs=2)
    data = env.reset()
    s = data.states.clone()
    s[1, :2] = s[0, :2] + 0.01  # force proximity
    data = data.replace(states=s, pos=s[:, :2])
    r = env.params["car_radius"]
    n, N = 5, 7
    diff = s.unsqueeze(1) - s.unsqueeze(0)
    pd = diff[:n, :, :2]
    dist = pd.norm(dim=2)
    d_u = dist + torch.eye(N)[:n] * (4 * r + 1)
    coll = (d_u < 2 * r).max(dim=1).values
    warn = d_u < 3 * r
    pos_vec = -(pd / (pd.norm(dim=2, keepdim=True) + 1e-4))
    tv = torch.stack([torch.cos(s[:n, 2]), torch.sin(s[:n, 2])],
                     dim=-1).unsqueeze(1)
    inner = (pos_vec * tv).sum(dim=2)
    thr = torch.cos(torch.asin(2 * r / (d_u + 1e-7)))
    unsafe_ref = coll | ((inner > thr) & warn).max(dim=1).values
    assert torch.equal(env.unsafe_mask(data), unsafe_ref)

    safe_ref = ((dist + torch.eye(N)[:n] * (4 * r + 1)) > 3 * r).min(
        dim=1).values
    assert torch.equal(env.safe_mask(data), safe_ref)

    col_ref = ((dist + torch.eye(N)[:n] * (2 * r + 1)) < 2 * r).max(
        dim=1).values
    assert torch.equal(env.collision_mask(data), col_ref)


def test_dubins_u_ref_turns_toward_goal():
    env = _mk("DubinsCar", 1, num_obs=0)
    env.reset()
    # place agent at origin heading +x, goal straight ahead
    s = torch.tensor([[0.0, 0.0, 0.0, 0.0]])
    env._goal = torch.tensor([[1.0, 0.0, 0.0, 0.0]])
    data = env._build_data(s)
    u = env.u_ref(data)
    # tiny residual from the reference's 1e-4 epsilon in theta_between
    assert abs(u[0, 0]) < 0.01      # ~no turn needed
    assert u[0, 1] > 0              # accelerate toward goal
    # goal to the left (+y): positive omega (anticlockwise)
    env._goal = torch.tensor([[0.0, 1.0, 0.0, 0.0]])
    u = env.u_ref(data)
    assert u[0, 0] > 0


def test_dubins_edge_attr_five_dims():
    env = _mk("DubinsCar", 4, num_obs=1)
    data = env.reset()
    if data.num_edges == 0:
        pytest.skip("no edges")
    assert data.edge_attr.shape[1] == 5
    src, dst = data.edge_index
    s = data.states
    info = torch.cat([s[:, :3],
                      (s[:, 3] * torch.cos(s[:, 2])).unsqueeze(1),
                      (s[:, 3] * torch.sin(s[:, 2])).unsqueeze(1)], dim=1)
    assert torch.allclose(data.edge_attr, info[src] - info[dst], atol=1e-6)


def test_dubins_only_agents_receive_edges():
    env = _mk("DubinsCar", 4, num_obs=3)
    data = env.reset()
    if data.num_edges:
        assert (data.edge_index[1] < 4).all()


def test_dubins_topk_neighbors():
    env = make_env("DubinsCar", 8, torch.device("cpu"), max_neighbors=2)
    env.train()
    data = env.reset()
    if data.num_edges:
        counts = torch.bincount(data.edge_index[1], minlength=8)
        assert counts.max() <= 2


# ---------------------------------------------------------------- SimpleDrone

def test_drone_reset_spawns_num_agents_obstacles():
    # reference quirk: num_obs ignored, always num_agents obstacles
    env = _mk("SimpleDrone", 5)
    data = env.reset()
    assert data.num_nodes == 10
    assert data.agent_mask.sum() == 5


def test_drone_dynamics_linear():
    env = _mk("SimpleDrone", 3)
    data = env.reset()
    u = torch.randn(3, 3) * 0.1
    xdot = env.dynamics(data, u)
    A, B = env._A, env._B
    am = data.agent_mask
    ref_agents = data.states[am] @ A.t() + u @ B.t()
    assert torch.allclose(xdot[am], ref_agents, atol=1e-5)
    assert (xdot[~am] == 0).all()


def test_drone_unsafe_mask_vz_quirk():
    """The velocity cone uses [vx/v, vy/v, vz] — vz unnormalized."""
    env = _mk("SimpleDrone", 2)
    data = env.reset()
    s = data.states.clone()
    r = env.params["drone_radius"]
    # place agent 1 just above agent 0, agent 0 moving up fast
    s[0, :3] = torch.tensor([1.0, 1.0, 1.0])
    s[1, :3] = torch.tensor([1.0, 1.0, 1.0 + 3.5 * r])
    s[0, 3:] = torch.tensor([0.0, 0.0, 2.0])
    data = data.replace(states=s, pos=s[:, :3])
    n, N = 2, 4
    diff = s.unsqueeze(1) - s.unsqueeze(0)
    pd = diff[:n, :, :3]
    dist = pd.norm(dim=2)
    d_u = dist + torch.eye(N)[:n] * (2 * r + 1)
    coll = (d_u < 2 * r).max(dim=1).values
    warn = d_u < 4 * r
    pos_vec = -(pd / (pd.norm(dim=2, keepdim=True) + 1e-4))
    vel = s[:n, 3:]
    v = vel.norm(dim=1, keepdim=True) + 1e-5
    tv = torch.cat([vel[:, 0:1] / v, vel[:, 1:2] / v, vel[:, 2:3]],
                   dim=1).unsqueeze(1)
    inner = (pos_vec * tv).sum(dim=2)
    thr = torch.cos(torch.asin(2 * r / (d_u + 1e-7)))
    ref = coll | ((inner > thr) & warn).max(dim=1).values
    assert torch.equal(env.unsafe_mask(data), ref)
    assert env.unsafe_mask(data)[0]  # flying into neighbor above


def test_drone_u_ref_shapes():
    env = _mk("SimpleDrone", 4)
    data = env.reset()
    u = env.u_ref(data)
    assert u.shape == (4, 3)


# --------------------------------------------------------------- shared paths

@pytest.mark.parametrize("env_name,n", [("SimpleCar", 4), ("DubinsCar", 4),
                                        ("SimpleDrone", 3)])
def test_episode_runs_to_done(env_name, n):
    env = _mk(env_name, n)
    data = env.reset()
    for t in range(600):
        data.update(u_ref=env.u_ref(data))
        act_dim = env.action_dim
        data, r, done, info = env.step(torch.zeros(n, act_dim))
        if done:
            break
    assert done


@pytest.mark.parametrize("env_name", ["SimpleCar", "DubinsCar", "SimpleDrone"])
def test_relink_batched_matches_per_graph(env_name):
    env = _mk(env_name, 4)
    graphs = [env.reset() for _ in range(3)]
    batch = GraphBatch.from_list([g.replace() for g in graphs])
    out = env.add_communication_links_batched(batch.replace())
    parts = [env.add_communication_links(g.replace(edge_index=None,
                                                   edge_attr=None))
             for g in graphs]
    ref = GraphBatch.from_list(parts)
    assert torch.equal(out.edge_index, ref.edge_index)
    assert torch.allclose(out.edge_attr, ref.edge_attr)


def test_demo2_mode_limits_goal_distance():
    env = _mk("SimpleCar", 4)
    env.demo(2)
    data = env.reset()
    d = torch.norm(data.states[:, :2] - env._goal, dim=1)
    assert (d <= env.params["max_distance"] * 1.5).all()
    assert env.max_episode_steps == 2500
