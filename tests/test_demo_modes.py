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
