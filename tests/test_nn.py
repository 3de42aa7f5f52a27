"""NN layer tests: explicit GNN pipelines vs. naive per-node loops, plus
state-dict key parity with the reference checkpoint layout."""
import torch
import torch.nn as nn

from gcbf_amd.nn import (MLP, CBFGNNLayer, CBFNetLayer, ControllerGNNLayer,
                         MACBFControllerLayer)


def _rand_graph(n=6, nd=4, ed=5, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, nd, generator=g)
    mask = torch.rand(n, n, generator=g) < 0.6
    mask.fill_diagonal_(False)
    nz = mask.nonzero()
    ei = torch.stack([nz[:, 1], nz[:, 0]])  # src, dst (dst-sorted)
    ea = torch.randn(ei.shape[1], ed, generator=g)
    return x, ea, ei


def _naive_layer_forward(layer, x, ea, ei, aggr="attn"):
    """Per-node Python-loop version of the message-passing pipeline."""
    n = x.shape[0]
    src, dst = ei
    msg_in = torch.cat([x[dst], x[src], ea], dim=1)
    msg = layer.phi(msg_in)
    agg = torch.zeros(n, msg.shape[1])
    for node in range(n):
        idx = (dst == node).nonzero()[:, 0]
        if not idx.numel():
            continue
        if aggr == "attn":
            gate = layer.aggr_module.gate_nn(msg[idx])
            att = torch.softmax(gate, dim=0)
            agg[node] = (att * msg[idx]).sum(dim=0)
        elif aggr == "max":
            agg[node] = msg[idx].max(dim=0).values
    if aggr == "max":
        return layer.gamma(agg)
    return layer.gamma(torch.cat([agg, x], dim=1))


def test_cbf_gnn_layer_matches_naive():
    torch.manual_seed(0)
    layer = CBFGNNLayer(node_dim=4, edge_dim=5, output_dim=16, phi_dim=8)
    # shrink MLPs for test speed
    layer.phi = MLP(13, 8, (32, 32), limit_lip=True)
    layer.gamma = MLP(12, 16, (32, 32), limit_lip=True)
    layer.eval()
    x, ea, ei = _rand_graph()
    with torch.no_grad():
        out = layer(x, ea, ei)
        ref = _naive_layer_forward(layer, x, ea, ei)
    assert torch.allclose(out, ref, atol=1e-5)


def test_controller_layer_node_mask():
    torch.manual_seed(0)
    layer = ControllerGNNLayer(node_dim=4, edge_dim=5, output_dim=8,
                               phi_dim=8)
    layer.phi = MLP(13, 8, (16,))
    layer.gamma = MLP(12, 8, (16,))
    x, ea, ei = _rand_graph()
    mask = torch.tensor([True, True, True, False, False, False])
    with torch.no_grad():
        full = layer(x, ea, ei)
        masked = layer(x, ea, ei, node_mask=mask)
    assert torch.allclose(full[mask], masked, atol=1e-6)


def test_macbf_layer_max_aggregation():
    torch.manual_seed(0)
    layer = MACBFControllerLayer(node_dim=4, edge_dim=5, output_dim=4,
                                 phi_dim=8)
    x, ea, ei = _rand_graph()
    with torch.no_grad():
        out = layer(x, ea, ei)
        ref = _naive_layer_forward(layer, x, ea, ei, aggr="max")
    assert torch.allclose(out, ref, atol=1e-5)


def test_cbfnet_layer_per_edge():
    layer = CBFNetLayer(node_dim=4, edge_dim=5, output_dim=1)
    x, ea, ei = _rand_graph()
    out = layer(x, ea, ei)
    assert out.shape == (ei.shape[1], 1)


def test_mlp_spectral_norm_keys():
    m = MLP(8, 4, (16, 16), limit_lip=True)
    keys = set(m.state_dict().keys())
    # old-style spectral norm: weight_orig + power-iteration buffers
    assert "net.0.weight_orig" in keys
    assert "net.0.weight_u" in keys
    assert "net.0.weight_v" in keys
    assert "net.2.weight_orig" in keys
    assert "net.4.weight_orig" in keys
    m2 = MLP(8, 4, (16, 16), limit_lip=False)
    assert "net.0.weight" in m2.state_dict()


def test_mlp_spectral_norm_limits_lipschitz():
    torch.manual_seed(0)
    m = MLP(8, 8, (16,), limit_lip=True)
    # push a weight to large norm, then check the effective weight is normed
    with torch.no_grad():
        m.net[0].weight_orig.mul_(100.0)
    m.eval()
    for _ in range(20):  # power iteration happens in train mode
        m.train()
        m(torch.randn(4, 8))
    m.eval()
    w = m.net[0].weight
    assert torch.linalg.matrix_norm(w, 2) < 1.5


def test_reference_checkpoint_key_layout():
    """CBFGNN / GNNController state-dict keys match the reference layout
    (feat_transformer.module_0.{phi,gamma,aggr_module.gate_nn}.net.N.*)."""
    from gcbf_amd.algo.gcbf import CBFGNN
    from gcbf_amd.controller import GNNController
    cbf = CBFGNN(num_agents=4, node_dim=4, edge_dim=5, phi_dim=8)
    keys = set(cbf.state_dict().keys())
    assert "feat_transformer.module_0.phi.net.0.weight_orig" in keys
    assert "feat_transformer.module_0.gamma.net.0.weight_orig" in keys
    assert "feat_transformer.module_0.aggr_module.gate_nn.net.0.weight" in keys
    assert "feat_2_CBF.net.0.weight" in keys

    actor = GNNController(num_agents=4, node_dim=4, edge_dim=5, phi_dim=8,
                          action_dim=2)
    akeys = set(actor.state_dict().keys())
    assert "feat_transformer.module_0.phi.net.0.weight" in akeys
    assert "feat_2_action.net.0.weight" in akeys


def test_mlp_orthogonal_init():
    torch.manual_seed(0)
    m = MLP(32, 16, (64,), limit_lip=False)
    w = m.net[0].weight  # (64, 32): columns orthonormal
    assert torch.allclose(w.t() @ w, torch.eye(32), atol=1e-5)
    assert (m.net[0].bias == 0).all()


def test_agent_index_matches_boolean_mask():
    """LONG agent_index indexing (hipGraph-capturable) must equal the
    boolean agent_mask path on a batched graph with obstacle rows."""
    import torch
    from gcbf_amd.algo.gcbf import CBFGNN
    from gcbf_amd.controller import GNNController
    from gcbf_amd.graph import GraphBatch

    torch.manual_seed(3)
    B, n, n_obs = 3, 5, 4
    N = n + n_obs
    x = torch.cat([torch.zeros(n, 4), torch.ones(n_obs, 4)]).repeat(B, 1)
    states = torch.randn(B * N, 4)
    # dense-ish random edges into agent receivers, dst-sorted per graph
    ei = []
    for b in range(B):
        for i in range(n):
            for j in range(N):
                if j != i and torch.rand(1).item() < 0.6:
                    ei.append((b * N + j, b * N + i))
    ei = torch.tensor(ei, dtype=torch.long).t()
    ea = torch.randn(ei.shape[1], 5)
    am = torch.zeros(B * N, dtype=torch.bool)
    am[(torch.arange(B * N) % N) < n] = True
    ptr = torch.arange(B + 1, dtype=torch.long) * N

    g_bool = GraphBatch(x=x, pos=states[:, :2], states=states,
                        edge_index=ei, edge_attr=ea, agent_mask=am,
                        u_ref=torch.randn(B * n, 2), ptr=ptr)
    g_idx = GraphBatch(x=x, pos=states[:, :2], states=states,
                       edge_index=ei, edge_attr=ea, agent_mask=am,
                       u_ref=g_bool.u_ref, ptr=ptr)
    g_idx.agent_index = am.nonzero().flatten()

    cbf = CBFGNN(num_agents=n, node_dim=4, edge_dim=5, phi_dim=256).eval()
    actor = GNNController(num_agents=n, node_dim=4, edge_dim=5, phi_dim=256,
                          action_dim=2).eval()
    with torch.no_grad():
        assert torch.equal(cbf(g_bool), cbf(g_idx))
        assert torch.equal(actor(g_bool), actor(g_idx))


def test_sn_weight_reuse_context():
    """Within sn_weight_reuse, W/σ is computed once and shared; outside,
    every call recomputes and the power iteration advances per call."""
    import torch
    from gcbf_amd.nn.mlp import SNLinear, sn_weight_reuse
    torch.manual_seed(0)
    lin = SNLinear(8, 8)
    lin.train()
    with sn_weight_reuse():
        w1 = lin.effective_weight()
        u_after_first = lin.weight_u.clone()
        w2 = lin.effective_weight()
        assert w1 is w2                    # shared tensor, shared autograd
        assert torch.equal(lin.weight_u, u_after_first)  # no 2nd advance
    assert lin._w_cache is None            # cleared on exit
    w3 = lin.effective_weight()
    assert w3 is not w1
    assert not torch.equal(lin.weight_u, u_after_first)  # advanced again
    # gradient flows through both uses of the cached weight
    with sn_weight_reuse():
        x = torch.randn(4, 8)
        y = lin(x).sum() + lin(x * 2).sum()
        y.backward()
    assert lin.weight_orig.grad is not None
    assert torch.isfinite(lin.weight_orig.grad).all()
