"""Native 2D demo world: oriented-box obstacles, analytic LiDAR, contacts.

The reference's DubinsCar demo modes (demo_0/1/3) use pybullet for four
things (gcbf/env/dubins_car.py:55-382, 884-923): box obstacle bodies,
batched LiDAR ray casts, closest-point/contact queries, and a GL camera.
This module replaces the first three with closed-form geometry, vectorized
in torch (device-aware, so the demo path runs on the MI355X too):

* ray ∩ oriented box  — slab test in the box frame (`raycast`)
* point→box distance  — oriented-box SDF (`box_distance`)
* kinematic motion    — constant-velocity integration (`advance`)

Rays are also occluded by circles (other agents and goal cylinders), like
pybullet's first-hit semantics; agents/goals are modeled as circles of the
car radius (the reference's racecar URDF footprint — a documented
approximation).  The camera is replaced by matplotlib top-down rendering
in DubinsCar.render.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor


class BoxWorld:
    """A set of oriented rectangular obstacles with per-box velocity."""

    def __init__(self, device: torch.device):
        self.device = device
        self.centers = torch.zeros(0, 2, device=device)
        self.half = torch.zeros(0, 2, device=device)      # half extents (l/2, w/2)
        self.theta = torch.zeros(0, device=device)
        # velocity as [heading, speed] — the reference's _obs_v layout
        self.vel = torch.zeros(0, 2, device=device)

    @property
    def num_boxes(self) -> int:
        return self.centers.shape[0]

    def add_box(self, center, size, theta: float, vel=(0.0, 0.0)):
        """size = (length, width); vel = (heading, speed)."""
        t = lambda v, n: torch.as_tensor(
            v, dtype=torch.float32, device=self.device).reshape(1, n)
        self.centers = torch.cat([self.centers, t(center, 2)])
        self.half = torch.cat([self.half, t(size, 2) / 2])
        self.theta = torch.cat([
            self.theta, torch.tensor([theta], dtype=torch.float32,
                                     device=self.device)])
        self.vel = torch.cat([self.vel, t(vel, 2)])

    def advance(self, dt: float):
        """Constant-velocity kinematic motion (the reference resets body
        positions/velocities every step; box-box contact response is not
        modeled — a documented simplification)."""
        if self.num_boxes == 0:
            return
        h, s = self.vel[:, 0], self.vel[:, 1]
        self.centers = self.centers + torch.stack(
            [s * torch.cos(h), s * torch.sin(h)], dim=1) * dt

    # ------------------------------------------------------------- queries
    def _to_box_frame(self, p: Tensor) -> Tensor:
        """(..., 2) world points -> (..., M, 2) box-frame coordinates."""
        rel = p.unsqueeze(-2) - self.centers            # (..., M, 2)
        c, s = torch.cos(self.theta), torch.sin(self.theta)
        x = rel[..., 0] * c + rel[..., 1] * s
        y = -rel[..., 0] * s + rel[..., 1] * c
        return torch.stack([x, y], dim=-1)

    def box_distance(self, points: Tensor) -> Tensor:
        """Signed distance from each point to each box surface: (P, M).
        Negative inside (oriented-box SDF)."""
        if self.num_boxes == 0:
            return torch.full((points.shape[0], 0), float("inf"),
                              device=self.device)
        q = self._to_box_frame(points).abs() - self.half
        outside = q.clamp(min=0).norm(dim=-1)
        inside = q.max(dim=-1).values.clamp(max=0)
        return outside + inside

    def min_distance(self, points: Tensor) -> Tensor:
        """(P,) distance to the nearest box surface (inf if no boxes)."""
        d = self.box_distance(points)
        return d.min(dim=1).values if self.num_boxes else \
            torch.full((points.shape[0],), float("inf"), device=self.device)

    def raycast(self, origins: Tensor, dirs: Tensor, max_range: float,
                occluder_centers: Optional[Tensor] = None,
                occluder_radius: float = 0.0
                ) -> Tuple[Tensor, Tensor, Tensor]:
        """Batched first-hit ray cast against the boxes.

        origins (K, 2), dirs (K, 2) unit vectors.  Returns
        (hit_mask (K,), hit_points (K, 2), hit_box (K,) long).  A ray whose
        first intersection is an occluder circle (or nothing within
        max_range) reports no hit — pybullet rayTestBatch semantics with
        the reference's `in self._obs_id` filter
        (gcbf/env/dubins_car.py:330-345).
        """
        K = origins.shape[0]
        dev = self.device
        if self.num_boxes == 0 or K == 0:
            return (torch.zeros(K, dtype=torch.bool, device=dev),
                    torch.zeros(K, 2, device=dev),
                    torch.full((K,), -1, dtype=torch.long, device=dev))
        o = self._to_box_frame(origins)                   # (K, M, 2)
        c, s = torch.cos(self.theta), torch.sin(self.theta)
        dx = dirs[:, :1] * c + dirs[:, 1:] * s            # (K, M)
        dy = -dirs[:, :1] * s + dirs[:, 1:] * c
        d = torch.stack([dx, dy], dim=-1)                 # (K, M, 2)

        inv = 1.0 / torch.where(d.abs() < 1e-12,
                                torch.full_like(d, 1e-12).copysign(d), d)
        t1 = (-self.half - o) * inv
        t2 = (self.half - o) * inv
        tmin = torch.minimum(t1, t2).max(dim=-1).values   # (K, M)
        tmax = torch.maximum(t1, t2).min(dim=-1).values
        valid = (tmax >= tmin) & (tmax >= 0) & (tmin > 1e-9) \
            & (tmin <= max_range)
        t_hit = torch.where(valid, tmin,
                            torch.full_like(tmin, float("inf")))
        t_box, box_id = t_hit.min(dim=1)                  # (K,)
        hit = torch.isfinite(t_box)

        if occluder_centers is not None and occluder_centers.numel():
            # ray ∩ circle: |o + t·d − c| = r, smallest positive root
            oc = origins.unsqueeze(1) - occluder_centers  # (K, C, 2)
            b = (oc * dirs.unsqueeze(1)).sum(-1)          # (K, C)
            cterm = (oc * oc).sum(-1) - occluder_radius ** 2
            disc = b * b - cterm
            root = -b - torch.sqrt(disc.clamp(min=0))
            occ_valid = (disc > 0) & (root > 1e-9)
            t_occ = torch.where(occ_valid, root,
                                torch.full_like(root, float("inf")))
            t_occ = t_occ.min(dim=1).values
            hit = hit & (t_box < t_occ)

        pts = origins + t_box.nan_to_num(posinf=0.0).unsqueeze(1) * dirs
        return hit, pts, torch.where(hit, box_id, torch.full_like(box_id, -1))
