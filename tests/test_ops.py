"""Hot-path op tests: eager implementations vs. naive per-segment loops.

These same tests run against the HIP kernels on GPU (the ops dispatch layer
routes CUDA tensors to gcbf_amd._C), so they double as kernel numerics tests.
"""

import pytest
import torch

from gcbf_amd import ops


def _naive_segment_softmax(gate, dst, n):
    out = torch.zeros_like(gate)
    for seg in range(n):
        idx = (dst == seg).nonzero()[:, 0]
        if idx.numel():
            out[idx] = torch.softmax(gate[idx], dim=0)
    return out


def _rand_edges(E, n, d, device="cpu", seed=0):
    g = torch.Generator().manual_seed(seed)
    dst = torch.sort(torch.randint(0, n, (E,), generator=g)).values.to(device)
    msg = torch.randn(E, d, generator=g).to(device)
    gate = torch.randn(E, 1, generator=g).to(device)
    return msg, gate, dst


def test_segment_softmax_matches_naive():
    msg, gate, dst = _rand_edges(64, 10, 8)
    att = ops.segment_softmax(gate, dst, 10)
    ref = _naive_segment_softmax(gate, dst, 10)
    assert torch.allclose(att, ref, atol=1e-6)


def test_segment_attn_aggregate_matches_naive():
    msg, gate, dst = _rand_edges(64, 10, 8)
    out = ops.segment_attn_aggregate(msg, gate, dst, 10)
    att = _naive_segment_softmax(gate, dst, 10)
    ref = torch.zeros(10, 8)
    for e in range(64):
        ref[dst[e]] += att[e] * msg[e]
    assert torch.allclose(out, ref, atol=1e-5)
    # empty segments are zero
    empty = torch.tensor([seg for seg in range(10)
                          if (dst == seg).sum() == 0])
    if empty.numel():
        assert out[empty].abs().max() == 0


def test_segment_attn_aggregate_grads():
    msg, gate, dst = _rand_edges(40, 8, 4)
    msg = msg.double().requires_grad_(True)
    gate = gate.double().requires_grad_(True)

    def f(m, g):
        return ops.eager.segment_attn_aggregate(m, g, dst, 8)

    assert torch.autograd.gradcheck(f, (msg, gate), atol=1e-6)


def test_segment_attn_custom_backward_matches_eager():
    """The analytic VJP in _SegmentAttnAggregate vs. autograd through eager."""
    from gcbf_amd.ops import _SegmentAttnAggregate
    msg, gate, dst = _rand_edges(50, 9, 6)
    m1 = msg.clone().requires_grad_(True)
    g1 = gate.clone().requires_grad_(True)
    out1 = _SegmentAttnAggregate.apply(m1, g1, dst, 9)
    grad_out = torch.randn_like(out1)
    out1.backward(grad_out)

    m2 = msg.clone().requires_grad_(True)
    g2 = gate.clone().requires_grad_(True)
    out2 = ops.eager.segment_attn_aggregate(m2, g2, dst, 9)
    out2.backward(grad_out)

    assert torch.allclose(out1, out2, atol=1e-6)
    assert torch.allclose(m1.grad, m2.grad, atol=1e-5)
    assert torch.allclose(g1.grad, g2.grad, atol=1e-5)


def test_segment_max_matches_naive():
    msg, _, dst = _rand_edges(64, 10, 8)
    out = ops.segment_max(msg, dst, 10)
    for seg in range(10):
        idx = (dst == seg).nonzero()[:, 0]
        if idx.numel():
            assert torch.allclose(out[seg], msg[idx].max(dim=0).values)
        else:
            assert out[seg].abs().max() == 0


def test_segment_max_backward():
    msg, _, dst = _rand_edges(30, 6, 3)
    m = msg.clone().requires_grad_(True)
    out = ops.segment_max(m, dst, 6)
    out.sum().backward()
    # gradient mass equals number of non-empty (segment, feature) cells
    n_nonempty = sum(int((dst == s).any()) for s in range(6)) * 3
    assert m.grad.sum().item() == pytest.approx(n_nonempty)


def _naive_radius_graph(pos, n_rec, r, max_neighbors):
    """Direct transcription of the reference builder semantics
    (gcbf/env/dubins_car.py:730-746) for one graph."""
    N = pos.shape[0]
    dist = torch.cdist(pos.unsqueeze(0), pos.unsqueeze(0))[0]
    dist = dist[:n_rec, :]
    dist = dist + torch.eye(N)[:n_rec] * (r + 1)
    if max_neighbors is not None and max_neighbors < N:
        _, ids = torch.topk(dist, max_neighbors, dim=-1, largest=False)
        for i in range(n_rec):
            mask = torch.zeros(N, dtype=torch.bool)
            mask[ids[i]] = True
            dist[i, ~mask] += r + 1
    edges = []
    for i in range(n_rec):
        for j in range(N):
            if dist[i, j] < r:
                edges.append((j, i))
    if not edges:
        return torch.zeros(2, 0, dtype=torch.long)
    return torch.tensor(edges, dtype=torch.long).t()


@pytest.mark.parametrize("max_neighbors", [None, 3])
def test_dense_radius_graph_single(max_neighbors):
    torch.manual_seed(1)
    N, n_rec = 12, 8
    pos = torch.rand(N, 2) * 2
    am = torch.zeros(N, dtype=torch.bool)
    am[:n_rec] = True
    ei = ops.dense_radius_graph(pos, am, 1.0, max_neighbors, batch=1)
    ref = _naive_radius_graph(pos, n_rec, 1.0, max_neighbors)
    # same ordering: row-major over (dst, src)
    assert torch.equal(ei, ref)


def test_dense_radius_graph_batched_equals_per_graph():
    torch.manual_seed(2)
    B, N, n_rec = 4, 10, 6
    pos = torch.rand(B * N, 2) * 2
    am = torch.zeros(B, N, dtype=torch.bool)
    am[:, :n_rec] = True
    ei = ops.dense_radius_graph(pos, am.view(-1), 1.0, None, batch=B)
    parts = []
    for b in range(B):
        e = _naive_radius_graph(pos[b * N:(b + 1) * N], n_rec, 1.0, None)
        parts.append(e + b * N)
    ref = torch.cat(parts, dim=1)
    assert torch.equal(ei, ref)


def test_dense_radius_graph_no_agent_mask():
    torch.manual_seed(3)
    pos = torch.rand(8, 2)
    ei = ops.dense_radius_graph(pos, None, 0.7, None, batch=1)
    ref = _naive_radius_graph(pos, 8, 0.7, None)
    assert torch.equal(ei, ref)
    # no self loops, dst-sorted
    assert (ei[0] != ei[1]).all()
    assert (ei[1].diff() >= 0).all()
