"""GraphBatch container tests."""
import torch

from gcbf_amd.graph import GraphBatch


def _rand_graph(n, e_density=0.5, with_mask=False, seed=0):
    g = torch.Generator().manual_seed(seed)
    states = torch.randn(n, 4, generator=g)
    x = torch.zeros(n, 4)
    pos = states[:, :2]
    # dst-sorted edge list
    mask = torch.rand(n, n, generator=g) < e_density
    mask.fill_diagonal_(False)
    nz = mask.nonzero()
    dst, src = nz[:, 0], nz[:, 1]
    ei = torch.stack([src, dst])
    ea = states[src] - states[dst]
    am = None
    if with_mask:
        am = torch.zeros(n, dtype=torch.bool)
        am[: n // 2] = True
    return GraphBatch(x=x, pos=pos, states=states, edge_index=ei,
                      edge_attr=ea, agent_mask=am)


def test_from_list_offsets():
    gs = [_rand_graph(5, seed=i) for i in range(3)]
    b = GraphBatch.from_list(gs)
    assert b.num_graphs == 3
    assert b.num_nodes == 15
    assert b.ptr.tolist() == [0, 5, 10, 15]
    # second graph's edges offset by 5
    e0 = gs[1].edge_index + 5
    n_e0 = gs[0].num_edges
    assert torch.equal(b.edge_index[:, n_e0:n_e0 + gs[1].num_edges], e0)
    assert torch.allclose(b.edge_attr[n_e0:n_e0 + gs[1].num_edges],
                          gs[1].edge_attr)


def test_to_list_roundtrip():
    gs = [_rand_graph(6, seed=i, with_mask=True) for i in range(4)]
    b = GraphBatch.from_list(gs)
    back = b.to_list()
    assert len(back) == 4
    for orig, rec in zip(gs, back):
        assert torch.allclose(orig.states, rec.states)
        assert torch.equal(orig.edge_index, rec.edge_index)
        assert torch.allclose(orig.edge_attr, rec.edge_attr)
        assert torch.equal(orig.agent_mask, rec.agent_mask)


def test_views_uniform():
    gs = [_rand_graph(5, seed=i) for i in range(3)]
    b = GraphBatch.from_list(gs)
    sv = b.states_view()
    assert sv.shape == (3, 5, 4)
    assert torch.allclose(sv[1], gs[1].states)


def test_replace_and_update():
    g = _rand_graph(5)
    g2 = g.replace(states=g.states * 2)
    assert torch.allclose(g2.states, g.states * 2)
    assert torch.allclose(g2.pos, g.pos)  # untouched
    g.update(u_ref=torch.ones(5, 2))
    assert g.u_ref is not None
