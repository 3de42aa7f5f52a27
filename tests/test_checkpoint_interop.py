"""Reference checkpoint interoperability: state-dict keys must match the
reference modules (PyG Sequential + torch.nn.utils.spectral_norm Linears),
so its `cbf.pkl`/`actor.pkl` files load directly."""
import torch
import torch.nn as nn
from torch.nn.utils import spectral_norm

from gcbf_amd.algo.gcbf import CBFGNN
from gcbf_amd.controller import GNNController
from gcbf_amd.graph import GraphBatch


def _ref_mlp_state(prefix, dims, lip, state):
    """Emulate the reference MLP's state-dict entries: Sequential `net` with
    Linears at even indices, old-style spectral norm when lip."""
    idx = 0
    for n_in, n_out in zip(dims[:-1], dims[1:]):
        lin = nn.Linear(n_in, n_out)
        if lip:
            lin = spectral_norm(lin)
        for k, v in lin.state_dict().items():
            state[f"{prefix}.net.{idx}.{k}"] = v
        idx += 2  # activation module between
    return state


def _build_ref_cbf_state(node_dim=4, edge_dim=5, phi_dim=256):
    s = {}
    base = "feat_transformer.module_0"
    _ref_mlp_state(f"{base}.phi", [2 * node_dim + edge_dim, 2048, 2048,
                                   phi_dim], True, s)
    _ref_mlp_state(f"{base}.gamma", [phi_dim + node_dim, 2048, 2048, 1024],
                   True, s)
    _ref_mlp_state(f"{base}.aggr_module.gate_nn", [phi_dim, 128, 128, 1],
                   False, s)
    _ref_mlp_state("feat_2_CBF", [1024, 512, 128, 32, 1], False, s)
    return s


def test_reference_style_cbf_checkpoint_loads():
    torch.manual_seed(0)
    ref_state = _build_ref_cbf_state()
    cbf = CBFGNN(num_agents=4, node_dim=4, edge_dim=5, phi_dim=256)
    missing, unexpected = cbf.load_state_dict(ref_state, strict=True), None
    # strict load must succeed (raises on mismatch) — also run a forward
    n = 6
    states = torch.randn(n, 4)
    ei = torch.tensor([[1, 2, 3], [0, 0, 1]])
    g = GraphBatch(x=torch.zeros(n, 4), pos=states[:, :2], states=states,
                   edge_index=ei, edge_attr=torch.randn(3, 5))
    cbf.eval()
    with torch.no_grad():
        h = cbf(g)
    assert h.shape == (n, 1)
    assert torch.isfinite(h).all()


def test_reference_style_actor_checkpoint_loads():
    torch.manual_seed(0)
    s = {}
    base = "feat_transformer.module_0"
    _ref_mlp_state(f"{base}.phi", [13, 2048, 2048, 256], False, s)
    _ref_mlp_state(f"{base}.gamma", [260, 2048, 2048, 1024], False, s)
    _ref_mlp_state(f"{base}.aggr_module.gate_nn", [256, 128, 128, 1],
                   False, s)
    _ref_mlp_state("feat_2_action", [1026, 512, 128, 32, 2], False, s)
    actor = GNNController(num_agents=4, node_dim=4, edge_dim=5, phi_dim=256,
                          action_dim=2)
    actor.load_state_dict(s, strict=True)


def test_roundtrip_via_torch_save(tmp_path):
    """save -> load across fresh instances keeps outputs identical (the
    reference checkpoint layout: torch.save(state_dict))."""
    torch.manual_seed(0)
    a = CBFGNN(num_agents=4, node_dim=4, edge_dim=5, phi_dim=256)
    p = tmp_path / "cbf.pkl"
    torch.save(a.state_dict(), p)
    b = CBFGNN(num_agents=4, node_dim=4, edge_dim=5, phi_dim=256)
    b.load_state_dict(torch.load(p, weights_only=True))
    n = 5
    states = torch.randn(n, 4)
    ei = torch.tensor([[1, 2], [0, 3]])
    g = GraphBatch(x=torch.zeros(n, 4), pos=states[:, :2], states=states,
                   edge_index=ei, edge_attr=torch.randn(2, 5))
    a.eval(), b.eval()
    with torch.no_grad():
        assert torch.equal(a(g), b(g))
