"""Environment utilities: LQR gain, plotting helpers.

Reference: gcbf/env/utils.py (lqr at :14-36, plot helpers at :39-116).
Rendering is optional — matplotlib/networkx are imported lazily so headless
training boxes without them still work.
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch
from torch import Tensor


def lqr(A: np.ndarray, B: np.ndarray, Q: np.ndarray, R: np.ndarray) -> np.ndarray:
    """Discrete-time LQR gain K for x_{t+1} = A x + B u, u = -K x.

    Same construction as the reference (gcbf/env/utils.py:14-36): solve the
    discrete algebraic Riccati equation, then K = (BᵀXB + R)⁻¹ BᵀXA.
    """
    from scipy.linalg import inv, solve_discrete_are
    X = solve_discrete_are(A, B, Q, R)
    K = inv(B.T @ X @ B + R) @ (B.T @ X @ A)
    return K


def plot_graph(ax, data, radius: float, color: str, with_label: bool = True,
               plot_edge: bool = False, alpha: float = 1.0,
               danger_radius: Optional[float] = None,
               safe_radius: Optional[float] = None,
               obstacle_color: str = "#000000"):
    """2D graph rendering (agents as circles, edges as arrows)."""
    import matplotlib.pyplot as plt
    pos = data.pos.cpu().detach().numpy()
    agent_mask = None
    if getattr(data, "agent_mask", None) is not None:
        agent_mask = data.agent_mask.cpu().numpy()

    def plot_node(i, node_color, node_label=True, r=radius, a=alpha):
        ax.add_patch(plt.Circle((pos[i, 0], pos[i, 1]), radius=r,
                                color=node_color, clip_on=False, alpha=a))
        if node_label:
            ax.text(pos[i, 0], pos[i, 1], f"{i}", size=12, color="k",
                    family="sans-serif", weight="normal",
                    horizontalalignment="center", verticalalignment="center",
                    transform=ax.transData, clip_on=True)
        if danger_radius is not None:
            ax.add_patch(plt.Circle((pos[i, 0], pos[i, 1]),
                                    radius=danger_radius, color="red",
                                    clip_on=False, alpha=a, fill=False))
        if safe_radius is not None:
            ax.add_patch(plt.Circle((pos[i, 0], pos[i, 1]),
                                    radius=safe_radius, color="green",
                                    clip_on=False, alpha=a, fill=False))

    for i in range(pos.shape[0]):
        if agent_mask is not None and not agent_mask[i]:
            plot_node(i, obstacle_color, False, r=0.02, a=1)
        else:
            plot_node(i, color, with_label)
    if plot_edge and getattr(data, "edge_index", None) is not None:
        ei = data.edge_index.cpu().numpy()
        for e in range(ei.shape[1]):
            s, d = ei[0, e], ei[1, e]
            ax.annotate("", xy=(pos[d, 0], pos[d, 1]),
                        xytext=(pos[s, 0], pos[s, 1]),
                        arrowprops=dict(arrowstyle="->", alpha=0.5, lw=1.0))
    return ax


def plot_node_3d(ax, pos: np.ndarray, r: float, color: str, alpha: float,
                 grid: int = 10):
    u = np.linspace(0, 2 * np.pi, grid)
    v = np.linspace(0, np.pi, grid)
    x = r * np.outer(np.cos(u), np.sin(v)) + pos[0]
    y = r * np.outer(np.sin(u), np.sin(v)) + pos[1]
    z = r * np.outer(np.ones(np.size(u)), np.cos(v)) + pos[2]
    ax.plot_surface(x, y, z, color=color, alpha=alpha)
    return ax


def plot_graph_3d(ax, data, radius: float, color: str, with_label: bool = True,
                  plot_edge: bool = False, alpha: float = 1.0):
    pos = data.pos.cpu().detach().numpy()
    for i in range(pos.shape[0]):
        plot_node_3d(ax, pos[i], radius, color, alpha)
        if with_label:
            ax.text(pos[i, 0], pos[i, 1], pos[i, 2], f"{i}", size=12,
                    color="k", family="sans-serif", weight="normal",
                    horizontalalignment="center", verticalalignment="center")
    if plot_edge and getattr(data, "edge_index", None) is not None:
        ei = data.edge_index.cpu().numpy()
        for e in range(ei.shape[1]):
            j, k = ei[0, e], ei[1, e]
            vec = pos[j, :] - pos[k, :]
            x = [pos[j, 0] - 2 * radius * vec[0], pos[k, 0] + 2 * radius * vec[0]]
            y = [pos[j, 1] - 2 * radius * vec[1], pos[k, 1] + 2 * radius * vec[1]]
            z = [pos[j, 2] - 2 * radius * vec[2], pos[k, 2] + 2 * radius * vec[2]]
            ax.plot(x, y, z, linewidth=1.0, color="k")
    return ax


def fig_to_rgb_array(fig) -> np.ndarray:
    """Render a matplotlib figure to an RGB numpy array."""
    fig.canvas.draw()
    buf = np.asarray(fig.canvas.buffer_rgba())
    return buf[:, :, :3].copy()


def rejection_sample_positions(n: int, dim: int, side_length: float,
                               min_sep: float,
                               avoid: Optional[Tensor] = None,
                               avoid_dist: float = 0.0,
                               generator: Optional[torch.Generator] = None
                               ) -> Tensor:
    """Sequential rejection sampling of n points in [0, side]^dim.

    Matches the reference reset loops (gcbf/env/simple_car.py:97-121): points
    are accepted one at a time against a zero-initialized buffer, so early
    candidates must also clear the origin — a reference quirk kept for parity.
    Runs on CPU (host control flow) and is only called at episode resets.
    """
    pts = torch.zeros(n, dim)
    i = 0
    while i < n:
        cand = torch.rand(dim, generator=generator) * side_length
        if torch.norm(pts - cand, dim=1).min() <= min_sep:
            continue
        if avoid is not None and avoid.numel():
            if torch.norm(avoid - cand, dim=1).min() <= avoid_dist:
                continue
        pts[i] = cand
        i += 1
    return pts


def create_point_cloud_surface(vertices: Tensor, r: float) -> Tensor:
    """Point cloud covering a quadrilateral surface (reference
    gcbf/env/utils.py:119-131)."""
    points = []
    length = torch.norm(vertices[:, 1, :] - vertices[:, 0, :])
    width = torch.norm(vertices[:, 2, :] - vertices[:, 1, :])
    for i in range(1, int(length // (2 * r))):
        for j in range(int(width // (2 * r) + 1)):
            points.append(
                vertices[:, 0, :]
                + i * 2 * r * (vertices[:, 1, :] - vertices[:, 0, :]) / length
                + j * 2 * r * (vertices[:, 2, :] - vertices[:, 1, :]) / width)
    for vertex in vertices:
        for i in range(4):
            points.append(vertex[i, :].unsqueeze(0))
    return torch.cat(points, dim=0)


def create_point_cloud(vertices: Tensor, r: float, dim: int = 2) -> Tensor:
    """Sample obstacle boundary points every 2r (reference
    gcbf/env/utils.py:134-150)."""
    points = []
    if dim == 2:
        for i in range(vertices.shape[0]):
            points.append(vertices[i, :])
            j = i + 1 if i < vertices.shape[0] - 1 else 0
            direction = (vertices[j, :] - vertices[i, :]) / torch.norm(
                vertices[j, :] - vertices[i, :])
            while torch.norm(points[-1] - vertices[j, :]) > 2 * r:
                points.append(points[-1] + 2 * r * direction)
        points = torch.stack(points, dim=0)
    elif dim == 3:
        surface_nodes = [[0, 1, 2, 3], [4, 5, 6, 7], [0, 4, 5, 1],
                         [1, 2, 6, 5], [2, 6, 7, 3], [0, 3, 7, 4]]
        points = create_point_cloud_surface(vertices[surface_nodes, :], r)
    else:
        raise NotImplementedError
    return points


def create_rectangle(center: Tensor, length: float, width: float,
                     theta: float) -> Tensor:
    """Rotated rectangle vertices (reference gcbf/env/utils.py:153-161)."""
    vertices = torch.tensor([[length / 2, width / 2],
                             [length / 2, -width / 2],
                             [-length / 2, -width / 2],
                             [-length / 2, width / 2]]).type_as(center)
    rot = torch.tensor([[np.cos(theta), -np.sin(theta)],
                        [np.sin(theta), np.cos(theta)]]).type_as(center)
    return center + vertices @ rot


def create_cuboid(center: Tensor, length: float, width: float, height: float,
                  theta: float) -> Tensor:
    """Rotated cuboid vertices (reference gcbf/env/utils.py:164-175)."""
    signs = [(1, 1, 1), (1, -1, 1), (-1, -1, 1), (-1, 1, 1),
             (1, 1, -1), (1, -1, -1), (-1, -1, -1), (-1, 1, -1)]
    vertices = torch.tensor(
        [[sx * length / 2, sy * width / 2, sz * height / 2]
         for sx, sy, sz in signs]).type_as(center)
    rot = torch.tensor([[np.cos(theta), -np.sin(theta), 0],
                        [np.sin(theta), np.cos(theta), 0],
                        [0, 0, 1]]).type_as(center)
    return center + vertices @ rot
