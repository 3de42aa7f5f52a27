"""Eager (pure PyTorch) reference implementations of every hot-path op.

These are the golden oracles for the HIP/CDNA4 kernels in ``gcbf_amd/ops/hip``
and the CPU execution path.  Semantics mirror the reference implementation:

* segment softmax / weighted sum  ~ PyG ``AttentionalAggregation``
  (reference: gcbf/nn/gnn.py:17-19 via torch_scatter)
* dense radius graph with optional k-nearest cap
  (reference: gcbf/env/dubins_car.py:730-746, simple_drone.py:316-333)
* batched pairwise masks (reference: gcbf/env/simple_car.py:306-387 et al.,
  which loop over graphs in Python — here they are single batched ops)
"""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor


# --------------------------------------------------------------------------
# segment (per-destination) reductions over dst-sorted edge lists
# --------------------------------------------------------------------------

def segment_softmax(gate: Tensor, dst: Tensor, num_nodes: int) -> Tensor:
    """Softmax of ``gate`` (E, 1) over edges grouped by destination node.

    Matches ``torch_geometric.utils.softmax`` semantics (numerically stabilized
    by the per-segment max; empty segments produce no outputs).
    """
    gate = gate.squeeze(-1)
    # one extra row absorbs sentinel destinations (padded edge lists use
    # dst == num_nodes for invalid entries)
    seg_max = torch.full((num_nodes + 1,), float("-inf"),
                         dtype=gate.dtype, device=gate.device)
    seg_max = seg_max.scatter_reduce(0, dst, gate.detach(), reduce="amax",
                                     include_self=True)
    shifted = gate - seg_max.index_select(0, dst)
    ex = shifted.exp()
    denom = torch.zeros(num_nodes + 1, dtype=gate.dtype, device=gate.device)
    denom = denom.index_add(0, dst, ex)
    att = ex / (denom.index_select(0, dst) + 1e-16)
    return att.unsqueeze(-1)


def segment_sum(values: Tensor, dst: Tensor, num_nodes: int) -> Tensor:
    """Sum of (E, D) edge values into (num_nodes, D) per destination.
    A sentinel row absorbs dst == num_nodes (padded edge lists)."""
    out = torch.zeros(num_nodes + 1, values.shape[1],
                      dtype=values.dtype, device=values.device)
    return out.index_add(0, dst, values)[:num_nodes]


def segment_attn_aggregate(msg: Tensor, gate: Tensor, dst: Tensor,
                           num_nodes: int) -> Tensor:
    """out[n] = sum_{e: dst[e]=n} softmax_seg(gate)[e] * msg[e].

    The fused attention aggregation (gate softmax + weighted scatter-sum) of
    PyG ``AttentionalAggregation`` (reference gcbf/nn/gnn.py:17-19).
    """
    att = segment_softmax(gate, dst, num_nodes)
    return segment_sum(att * msg, dst, num_nodes)


def segment_max(values: Tensor, dst: Tensor, num_nodes: int) -> Tensor:
    """Per-destination max with 0 for empty segments (torch_scatter
    ``scatter_max`` as used by PyG ``aggr='max'``, reference gcbf/nn/gnn.py:117
    — PyG fills empty segments with 0)."""
    out = torch.full((num_nodes, values.shape[1]), float("-inf"),
                     dtype=values.dtype, device=values.device)
    idx = dst.unsqueeze(-1).expand_as(values)
    out = out.scatter_reduce(0, idx, values, reduce="amax", include_self=True)
    return out.masked_fill(out == float("-inf"), 0.0)


# --------------------------------------------------------------------------
# graph construction
# --------------------------------------------------------------------------

def dense_radius_graph(
        pos: Tensor,
        agent_mask: Optional[Tensor],
        comm_radius: float,
        max_neighbors: Optional[int] = None,
        batch: int = 1,
) -> Tensor:
    """Batched dense radius-graph construction.

    pos: (B*N, pos_dim); agent_mask: (B*N,) bool or None.  Only agent nodes
    receive edges.  Returns global ``edge_index`` (2, E) = [src; dst] sorted by
    (graph, dst, src), matching the reference builder's ``nonzero`` ordering
    (gcbf/env/dubins_car.py:730-746: receivers are the first ``n_agents`` rows
    of each graph, self-edges excluded, optional cap to the ``max_neighbors``
    nearest senders).
    """
    B = batch
    N = pos.shape[0] // B
    p = pos.view(B, N, -1)
    # (B, N, N) pairwise distances dist[b, i, j] = |p_i - p_j| via explicit
    # differences (same rounding as the reference's torch.norm and the HIP
    # kernel; cdist's matmul trick rounds differently at the radius boundary)
    dist = (p.unsqueeze(2) - p.unsqueeze(1)).norm(dim=-1)
    big = comm_radius + 1.0
    eye = torch.eye(N, device=pos.device, dtype=dist.dtype)
    if agent_mask is None:
        n_rec = N
        dist = dist + eye * big
    else:
        am = agent_mask.view(B, N)
        n_rec = int(am[0].sum().item())
        # reference convention: agents are the first n_rec rows of each graph
        dist = dist[:, :n_rec, :]
        # reference adds eye(n_rec) to the first n_rec columns (self-exclusion)
        dist = dist + eye[:n_rec] * big

    if max_neighbors is not None and max_neighbors < dist.shape[-1]:
        # keep only each receiver's k nearest senders (reference uses topk then
        # masks the rest, gcbf/env/dubins_car.py:736-740, but with a Python
        # loop per row — this is the batched equivalent).  Tie semantics
        # differ from torch.topk: EVERY sender tied at the k-th smallest
        # distance is kept (can exceed k on exact float ties — measure-zero
        # in practice); the HIP kth_smallest kernel matches this rule.
        kth = dist.topk(max_neighbors, dim=-1, largest=False).values[..., -1:]
        dist = torch.where(dist <= kth, dist, dist + big)

    mask = dist < comm_radius  # (B, n_rec, N)
    nz = mask.nonzero(as_tuple=False)  # rows (b, i, j) in row-major order
    b, i, j = nz[:, 0], nz[:, 1], nz[:, 2]
    src = b * N + j
    dst = b * N + i
    return torch.stack([src, dst], dim=0)


def pairwise_dist_masked(states: Tensor, agent_mask: Optional[Tensor],
                         batch: int, pos_dim: int,
                         diag_offset: float) -> Tensor:
    """(B, n_rec, N) pairwise position distances with ``diag_offset`` added on
    the receiver-self diagonal — the shared core of the safe/unsafe/collision
    masks (reference pattern at gcbf/env/simple_car.py:320-325)."""
    B = batch
    N = states.shape[0] // B
    p = states.view(B, N, -1)[..., :pos_dim]
    dist = (p.unsqueeze(2) - p.unsqueeze(1)).norm(dim=-1)
    eye = torch.eye(N, device=states.device, dtype=dist.dtype)
    if agent_mask is None:
        return dist + eye * diag_offset
    am = agent_mask.view(B, N)
    n_rec = int(am[0].sum().item())
    return dist[:, :n_rec, :] + eye[:n_rec] * diag_offset
