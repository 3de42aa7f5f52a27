"""DubinsCar demo modes (native BoxWorld LiDAR/contacts, no pybullet).

Covers the reference's demo_0/1/3 functional surface
(gcbf/env/dubins_car.py:55-382, 637-722, 884-923): box layouts, analytic
ray casting, obstacle point clouds as graph nodes, contact collision
masks, kinematic obstacle motion, and demo rendering.
"""
import math

import numpy as np
import pytest
import torch

from gcbf_amd.env import make_env
from gcbf_amd.env.demo_world import BoxWorld


def _dubins(n=4, num_obs=4, seed=0):
    torch.manual_seed(seed)
    np.random.seed(seed)
    dev = torch.device("cpu")
    e0 = make_env("DubinsCar", n, dev)
    p = e0.default_params
    p["num_obs"] = num_obs
    return make_env("DubinsCar", n, dev, params=p)


# ------------------------------------------------------------- BoxWorld
def test_raycast_axis_aligned_box():
    w = BoxWorld(torch.device("cpu"))
    w.add_box((2.0, 0.0), (1.0, 1.0), 0.0)   # box spans x in [1.5, 2.5]
    origins = torch.zeros(1, 2)
    dirs = torch.tensor([[1.0, 0.0]])
    hit, pts, box = w.raycast(origins, dirs, max_range=5.0)
    assert bool(hit[0])
    assert torch.allclose(pts[0], torch.tensor([1.5, 0.0]), atol=1e-5)
    assert int(box[0]) == 0
    # out of range
    hit, _, _ = w.raycast(origins, dirs, max_range=1.0)
    assert not bool(hit[0])
    # pointing away
    hit, _, _ = w.raycast(origins, -dirs, max_range=5.0)
    assert not bool(hit[0])


def test_raycast_rotated_box():
    w = BoxWorld(torch.device("cpu"))
    # unit square rotated 45°: along +x its near corner is at
    # x = 2 − √2/2
    w.add_box((2.0, 0.0), (1.0, 1.0), math.pi / 4)
    hit, pts, _ = w.raycast(torch.zeros(1, 2), torch.tensor([[1.0, 0.0]]),
                            max_range=5.0)
    assert bool(hit[0])
    assert abs(float(pts[0, 0]) - (2.0 - math.sqrt(2) / 2)) < 1e-5


def test_raycast_occluder_blocks():
    w = BoxWorld(torch.device("cpu"))
    w.add_box((3.0, 0.0), (1.0, 1.0), 0.0)
    origins = torch.zeros(1, 2)
    dirs = torch.tensor([[1.0, 0.0]])
    occ = torch.tensor([[1.0, 0.0]])
    hit, _, _ = w.raycast(origins, dirs, 5.0, occluder_centers=occ,
                          occluder_radius=0.2)
    assert not bool(hit[0])
    # occluder behind the box does not block
    occ = torch.tensor([[4.0, 0.0]])
    hit, _, _ = w.raycast(origins, dirs, 5.0, occluder_centers=occ,
                          occluder_radius=0.2)
    assert bool(hit[0])


def test_box_distance_sdf():
    w = BoxWorld(torch.device("cpu"))
    w.add_box((0.0, 0.0), (2.0, 1.0), 0.0)   # |x|<=1, |y|<=0.5
    pts = torch.tensor([[2.0, 0.0],    # 1.0 outside along x
                        [0.0, 0.0],    # inside
                        [2.0, 1.5]])   # corner distance sqrt(2)
    d = w.box_distance(pts)[:, 0]
    assert abs(float(d[0]) - 1.0) < 1e-6
    assert float(d[1]) < 0
    assert abs(float(d[2]) - math.sqrt(2)) < 1e-5


def test_advance_moves_boxes():
    w = BoxWorld(torch.device("cpu"))
    w.add_box((0.0, 0.0), (1.0, 1.0), 0.0, vel=(0.0, 2.0))     # +x at 2
    w.add_box((5.0, 5.0), (1.0, 1.0), 0.0, vel=(math.pi / 2, 0.0))  # static
    w.advance(0.5)
    assert torch.allclose(w.centers[0], torch.tensor([1.0, 0.0]), atol=1e-6)
    assert torch.allclose(w.centers[1], torch.tensor([5.0, 5.0]), atol=1e-6)


# ----------------------------------------------------------- demo modes
@pytest.mark.parametrize("idx", [0, 1, 3])
def test_demo_reset_and_steps(idx):
    env = _dubins(n=4, num_obs=4, seed=1)
    env.demo(idx)
    data = env.reset()
    n = env.num_agents
    assert data.states.shape[0] >= n
    if idx == 1:
        assert env._world.num_boxes == 0
        assert data.states.shape[0] == n     # no LiDAR points
    else:
        assert env._world.num_boxes >= (4 if idx == 3 else 1)
    assert env.max_episode_steps == (2500 if idx == 1 else 2000)
    for _ in range(5):
        a = torch.zeros(n, 2)
        data, reward, done, info = env.step(a)
        assert torch.isfinite(data.states).all()
        assert info["collision"].shape[0] == n
        assert info["reach"].shape[0] == n
        if done:
            break


def test_demo0_lidar_points_are_on_box_surfaces():
    env = _dubins(n=4, num_obs=6, seed=3)
    env.demo(0)
    env.reset()
    if env._obs.shape[0] == 0:
        pytest.skip("layout produced no LiDAR hits for this seed")
    pts = env._obs[:, :2]
    d = env._world.box_distance(pts).min(dim=1).values
    assert float(d.abs().max()) < 1e-3   # hits lie on a box surface
    # hit velocities are one of the boxes' velocities
    assert env._obs.shape[1] == 4


def test_demo0_obstacle_nodes_in_graph():
    env = _dubins(n=4, num_obs=6, seed=3)
    env.demo(0)
    data = env.reset()
    n = env.num_agents
    k = env._obs.shape[0]
    assert data.states.shape[0] == n + k
    if k:
        assert bool(data.x[n:].eq(1).all())
        assert bool(data.agent_mask[:n].all())
        assert not bool(data.agent_mask[n:].any())
        # GNN forward over the demo graph works end to end
        from gcbf_amd.algo import make_algo
        algo = make_algo("gcbf", env, n, env.node_dim, env.edge_dim,
                         env.action_dim, torch.device("cpu"))
        data.update(u_ref=env.u_ref(data))
        a = algo.act(data)
        assert a.shape == (n, 2)


def test_demo3_has_static_corner_blocks_and_shuffled_goals():
    env = _dubins(n=4, num_obs=4, seed=5)
    env.demo(3)
    env.reset()
    # the 4 corner blocks are appended last and are static
    assert env._world.num_boxes >= 4
    assert torch.all(env._world.vel[-4:, 1] == 0)
    area = env._params["area_size"]
    sq = area / 16 * 3
    expect = torch.tensor([[sq, sq], [sq, area - sq], [area - sq, sq],
                           [area - sq, area - sq]])
    assert torch.allclose(env._world.centers[-4:], expect, atol=1e-5)


def test_demo_collision_mask_detects_box_contact():
    env = _dubins(n=2, num_obs=0, seed=7)
    env.demo(0)
    env.reset()
    # plant a box directly on agent 0
    p0 = env.data.states[0, :2]
    env._world.add_box((float(p0[0]), float(p0[1])), (0.5, 0.5), 0.0)
    mask = env.collision_mask(env.data)
    assert bool(mask[0])


def test_demo_render_returns_frame():
    env = _dubins(n=3, num_obs=4, seed=9)
    env.demo(0)
    env.reset()
    frame = env.render()
    assert frame.ndim == 3 and frame.shape[2] == 3
    assert frame.shape[0] > 100 and frame.shape[1] > 100


def test_demo1_runs_nominal_episode_segment():
    """demo_1 with the nominal controller: a longer rollout stays finite
    and agents progress toward goals on average."""
    env = _dubins(n=4, num_obs=0, seed=11)
    env.demo(1)
    data = env.reset()
    d0 = torch.norm(data.states[:4, :2] - env._goal[:, :2], dim=1).mean()
    for _ in range(100):
        data, r, done, info = env.step(torch.zeros(4, 2))
        if done:
            break
    d1 = torch.norm(data.states[:4, :2] - env._goal[:, :2], dim=1).mean()
    assert torch.isfinite(data.states).all()
    assert d1 < d0


# --------------------------------------------------- geometry fuzzing
def test_raycast_fuzz_vs_sampled_oracle():
    """Random rays vs a dense sampling oracle: the reported hit distance
    must match the first sampled point inside any box to grid accuracy,
    and misses must have no sampled hit."""
    torch.manual_seed(31)
    w = BoxWorld(torch.device("cpu"))
    g = torch.Generator().manual_seed(31)
    for _ in range(5):
        c = torch.rand(2, generator=g) * 4
        sz = 0.2 + torch.rand(2, generator=g)
        th = float(torch.rand(1, generator=g)) * 6.28
        w.add_box((float(c[0]), float(c[1])),
                  (float(sz[0]), float(sz[1])), th)
    K = 64
    origins = torch.rand(K, 2, generator=g) * 4
    ang = torch.rand(K, generator=g) * 6.28
    dirs = torch.stack([torch.cos(ang), torch.sin(ang)], dim=1)
    # drop rays starting inside a box (pybullet reports no hit with the
    # containing body; the slab test would report the exit face)
    outside = w.min_distance(origins) > 0
    origins, dirs = origins[outside], dirs[outside]
    R = 2.0
    hit, pts, box = w.raycast(origins, dirs, R)
    ts = torch.linspace(1e-3, R, 4001)
    for k in range(origins.shape[0]):
        samples = origins[k] + ts.unsqueeze(1) * dirs[k]
        inside = w.box_distance(samples).min(dim=1).values < 0
        if bool(hit[k]):
            t_hit = float((pts[k] - origins[k]).norm())
            first = float(ts[inside.nonzero()[0, 0]]) if inside.any() \
                else None
            assert first is not None and abs(first - t_hit) < 2e-3, \
                (k, t_hit, first)
        else:
            # no sampled point strictly inside any box before the range
            assert not bool(inside.any()), k


def test_box_distance_fuzz_vs_corner_oracle():
    """Rotated-box SDF vs an explicit polygon distance computation."""
    import math
    torch.manual_seed(33)
    w = BoxWorld(torch.device("cpu"))
    cx, cy, hx, hy, th = 1.0, -0.5, 0.8, 0.3, 0.7
    w.add_box((cx, cy), (2 * hx, 2 * hy), th)
    pts = torch.randn(200, 2) * 2
    d = w.box_distance(pts)[:, 0]
    # oracle: distance from point to the rectangle via corner/edge math in
    # the box frame
    c, s = math.cos(th), math.sin(th)
    rel = pts - torch.tensor([cx, cy])
    bx = rel[:, 0] * c + rel[:, 1] * s
    by = -rel[:, 0] * s + rel[:, 1] * c
    qx = bx.abs() - hx
    qy = by.abs() - hy
    outside = torch.sqrt(qx.clamp(min=0) ** 2 + qy.clamp(min=0) ** 2)
    inside = torch.maximum(qx, qy).clamp(max=0)
    oracle = outside + inside
    assert torch.allclose(d, oracle, atol=1e-6)
