"""GPU kernel numerics tests: HIP/CDNA4 kernels vs. the eager fp32 oracles.

All tests are @pytest.mark.gpu and run on an MI355X box; the ops dispatch
layer routes CUDA tensors to gcbf_amd._C, so these exercise the native
kernels end to end.
"""
import numpy as np
import pytest
import torch

from gcbf_amd import ops
from gcbf_amd.ops import eager

pytestmark = pytest.mark.gpu


def _rand_edges(E, n, d, seed=0):
    g = torch.Generator().manual_seed(seed)
    dst = torch.sort(torch.randint(0, n, (E,), generator=g)).values
    msg = torch.randn(E, d, generator=g)
    gate = torch.randn(E, 1, generator=g)
    return msg.cuda(), gate.cuda(), dst.cuda()


def test_ext_loads():
    assert ops.hip_available(), "gcbf_amd._C must be importable on GPU"


@pytest.mark.parametrize("E,n,d", [(64, 10, 8), (1000, 77, 256),
                                   (5000, 300, 128), (333, 50, 7)])
def test_segment_attn_fwd_matches_eager(E, n, d):
    msg, gate, dst = _rand_edges(E, n, d)
    out = ops.segment_attn_aggregate(msg, gate, dst, n)
    ref = eager.segment_attn_aggregate(msg.cpu(), gate.cpu(), dst.cpu(), n)
    assert torch.allclose(out.cpu(), ref, atol=1e-5), \
        (out.cpu() - ref).abs().max()


@pytest.mark.parametrize("E,n,d", [(200, 20, 16), (3000, 100, 256)])
def test_segment_attn_bwd_matches_eager(E, n, d):
    msg, gate, dst = _rand_edges(E, n, d)
    m1 = msg.clone().requires_grad_(True)
    g1 = gate.clone().requires_grad_(True)
    out1 = ops.segment_attn_aggregate(m1, g1, dst, n)
    grad_out = torch.randn_like(out1)
    out1.backward(grad_out)

    m2 = msg.cpu().requires_grad_(True)
    g2 = gate.cpu().requires_grad_(True)
    out2 = eager.segment_attn_aggregate(m2, g2, dst.cpu(), n)
    out2.backward(grad_out.cpu())

    assert torch.allclose(m1.grad.cpu(), m2.grad, atol=1e-5)
    assert torch.allclose(g1.grad.cpu(), g2.grad, atol=1e-4), \
        (g1.grad.cpu() - g2.grad).abs().max()


def test_segment_attn_empty_segments():
    # nodes with no incoming edges must produce zero rows
    msg, gate, dst = _rand_edges(16, 4, 8, seed=3)
    out = ops.segment_attn_aggregate(msg, gate, dst, 64)
    assert out.shape == (64, 8)
    empty = torch.ones(64, dtype=torch.bool)
    empty[dst.unique().cpu()] = False
    assert out[empty.cuda()].abs().max().item() == 0.0


@pytest.mark.parametrize("B,N,n_rec,P,topk", [
    (1, 16, 16, 2, None), (1, 20, 12, 2, None), (4, 18, 12, 2, 5),
    (8, 288, 256, 2, None), (2, 10, 6, 3, None)])
def test_build_graph_matches_eager(B, N, n_rec, P, topk):
    torch.manual_seed(B * 100 + N)
    pos = (torch.rand(B * N, P) * 3).cuda()
    states = torch.randn(B * N, 4 if P == 2 else 6).cuda()

    def attr_fn(s, ei):
        return s.index_select(0, ei[0]) - s.index_select(0, ei[1])

    ei, ea = ops.build_graph(pos, states, n_rec, 1.0, topk, B,
                             ops.ATTR_DIFF, states.shape[1], attr_fn)
    am = None
    if n_rec != N:
        am = torch.zeros(B, N, dtype=torch.bool)
        am[:, :n_rec] = True
        am = am.view(-1)
    ref_ei = eager.dense_radius_graph(pos.cpu(), am, 1.0, topk, B)
    assert torch.equal(ei.cpu(), ref_ei), \
        f"edge sets differ: {ei.shape} vs {ref_ei.shape}"
    ref_ea = attr_fn(states.cpu(), ref_ei)
    assert torch.allclose(ea.cpu(), ref_ea, atol=1e-6)


def test_build_graph_dubins_attr():
    torch.manual_seed(7)
    B, N, n_rec = 3, 12, 8
    pos = (torch.rand(B * N, 2) * 3).cuda()
    states = torch.randn(B * N, 4).cuda()

    def dubins_attr(s, ei):
        info = torch.cat([s[:, :3],
                          (s[:, 3] * torch.cos(s[:, 2])).unsqueeze(1),
                          (s[:, 3] * torch.sin(s[:, 2])).unsqueeze(1)], dim=1)
        return info.index_select(0, ei[0]) - info.index_select(0, ei[1])

    ei, ea = ops.build_graph(pos, states, n_rec, 1.0, None, B,
                             ops.ATTR_DUBINS, 5, dubins_attr)
    ref_ei = eager.dense_radius_graph(
        pos.cpu(),
        torch.arange(B * N) % N < n_rec, 1.0, None, B)
    assert torch.equal(ei.cpu(), ref_ei)
    assert torch.allclose(ea.cpu(), dubins_attr(states.cpu(), ref_ei),
                          atol=1e-5)


def test_gnn_forward_gpu_matches_cpu():
    from gcbf_amd.algo.gcbf import CBFGNN
    from gcbf_amd.graph import GraphBatch
    torch.manual_seed(0)
    cbf = CBFGNN(num_agents=8, node_dim=4, edge_dim=5, phi_dim=256)
    n = 12
    x = torch.zeros(n, 4)
    states = torch.randn(n, 4)
    mask = torch.rand(n, n) < 0.5
    mask.fill_diagonal_(False)
    nz = mask.nonzero()
    ei = torch.stack([nz[:, 1], nz[:, 0]])
    ea = torch.randn(ei.shape[1], 5)
    g_cpu = GraphBatch(x=x, pos=states[:, :2], states=states, edge_index=ei,
                       edge_attr=ea)
    cbf.eval()
    with torch.no_grad():
        h_cpu = cbf(g_cpu)
        cbf_gpu = cbf.cuda()
        h_gpu = cbf_gpu(g_cpu.to("cuda"))
    assert torch.allclose(h_gpu.cpu(), h_cpu, atol=1e-4), \
        (h_gpu.cpu() - h_cpu).abs().max()


def test_env_rollout_gpu():
    from gcbf_amd.env import make_env
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    data = env.reset()
    assert data.states.is_cuda
    for _ in range(5):
        data.update(u_ref=env.u_ref(data))
        data, r, done, info = env.step(torch.zeros(16, 2, device=dev))
    assert data.num_edges >= 0


def test_gcbf_update_gpu():
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=32)
    data = env.reset()
    for step in range(1, 33):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.5)
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
        if algo.is_update(step):
            out = algo.update(step, None)
    assert all(0 <= v <= 1 for v in out.values())


def test_gcbf_update_gpu_bf16():
    """bf16 autocast path: SNLinear power iteration + segment kernels must
    cope with mixed dtypes."""
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.trainer.utils import set_seed
    from gcbf_amd.utils.amp import enable_bf16
    set_seed(0)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=32)
    enable_bf16(algo)
    data = env.reset()
    for step in range(1, 33):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.5)
        assert a.dtype == torch.float32  # module boundary casts back
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
        if algo.is_update(step):
            out = algo.update(step, None)
    assert all(0 <= v <= 1 for v in out.values())


@pytest.mark.parametrize("env_name,n,obs", [("SimpleCar", 8, None),
                                            ("DubinsCar", 8, 4),
                                            ("SimpleDrone", 6, None)])
def test_fused_masks_match_eager(env_name, n, obs):
    """HIP fused mask kernel vs. the eager batched math on identical states
    (single graphs and batches)."""
    import os
    from gcbf_amd.env import make_env
    from gcbf_amd.graph import GraphBatch
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    kw = {}
    if obs is not None:
        e0 = make_env(env_name, n, dev)
        p = e0.default_params
        p["num_obs"] = obs
        kw["params"] = p
    env = make_env(env_name, n, dev, **kw)
    env.train()
    graphs = [env.reset() for _ in range(3)]
    # crowd some agents to trigger nontrivial masks
    g0 = graphs[0]
    s = g0.states.clone()
    s[1, :2] = s[0, :2] + 0.01
    graphs[0] = g0.replace(states=s, pos=s[:, : s.shape[1] // 2])
    for data in [graphs[0], GraphBatch.from_list(graphs)]:
        for which, fn in [("safe", env.safe_mask), ("unsafe", env.unsafe_mask),
                          ("collision", env.collision_mask)]:
            got = fn(data)  # GPU -> fused kernel
            os.environ["GCBF_AMD_FORCE_EAGER_MASKS"] = "1"
            try:
                ref = _eager_mask(env, data, which)
            finally:
                del os.environ["GCBF_AMD_FORCE_EAGER_MASKS"]
            assert torch.equal(got.cpu(), ref.cpu()), \
                f"{env_name} {which} mismatch"


def _eager_mask(env, data, which):
    """Run the env's eager mask math by temporarily disabling the kernel."""
    from unittest import mock
    with mock.patch.object(type(env), "_fused_mask",
                           lambda self, d, w: None):
        if which == "safe":
            return env.safe_mask(data)
        if which == "unsafe":
            return env.unsafe_mask(data)
        return env.collision_mask(data)


@pytest.mark.parametrize("env_name,n,obs", [("DubinsCar", 16, 0),
                                            ("DubinsCar", 8, 4),
                                            ("SimpleCar", 8, None),
                                            ("SimpleDrone", 6, None)])
def test_fused_env_step_matches_python(env_name, n, obs):
    """Fused rollout-step kernel vs. the eager python step on identical
    state (states, u_ref_next, reward, reach, collision)."""
    from unittest import mock
    from gcbf_amd.env import make_env
    from gcbf_amd import ops
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    kw = {}
    if obs:
        e0 = make_env(env_name, n, dev)
        p = e0.default_params
        p["num_obs"] = obs
        kw["params"] = p
    env = make_env(env_name, n, dev, **kw)
    env.train()
    data0 = env.reset()
    act = torch.randn(n, env.action_dim, device=dev) * 0.1

    def snapshot():
        obs = getattr(env, "_obs", None)
        return (env._data.replace(), env._t,
                None if obs is None else obs.clone())

    def restore(s):
        env._data, env._t, obs_s = s
        if obs_s is not None:
            env._obs = obs_s

    snap = snapshot()
    d_fused, r_fused, done_fused, info_fused = env.step(act.clone())
    fused_states = d_fused.states.clone()
    fused_uref = d_fused.u_ref.clone()
    fused_ei = d_fused.edge_index.clone()

    restore(snap)
    with mock.patch.object(ops, "env_step_fused", lambda *a, **k: None):
        d_py, r_py, done_py, info_py = env.step(act.clone())
    py_uref = env.u_ref(d_py)

    assert torch.allclose(fused_states, d_py.states, atol=1e-5), \
        (fused_states - d_py.states).abs().max()
    assert torch.allclose(fused_uref, py_uref, atol=1e-4), \
        (fused_uref - py_uref).abs().max()
    assert torch.allclose(r_fused, r_py.float(), atol=1e-4)
    assert done_fused == done_py
    assert torch.equal(info_fused["reach"], info_py["reach"])
    assert torch.equal(info_fused["collision"].cpu(),
                       info_py["collision"].cpu())
    assert torch.equal(fused_ei, d_py.edge_index)


def test_fused_env_step_full_episode():
    """Run a whole DubinsCar episode through the fused step; states must
    stay finite and episodes terminate."""
    from gcbf_amd.env import make_env
    from gcbf_amd.trainer.utils import set_seed
    set_seed(1)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    data = env.reset()
    for t in range(510):
        data.update(u_ref=env.u_ref(data) if data.u_ref is None
                    else data.u_ref)
        data, r, done, info = env.step(
            torch.zeros(16, 2, device=dev))
        assert torch.isfinite(data.states).all()
        if done:
            break
    assert done


def test_rollout_engine_matches_eager_loop():
    """Captured rollout vs. the eager loop: same seeds, same exploration
    stream -> trajectories must agree closely for many steps."""
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.rollout import RolloutEngine, engine_supported
    from gcbf_amd.trainer.utils import set_seed
    import numpy as np
    dev = torch.device("cuda")

    def build():
        set_seed(3)
        env = make_env("DubinsCar", 16, dev)
        env.train()
        algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                         env.action_dim, dev, batch_size=512)
        return env, algo

    # eager loop
    env_a, algo_a = build()
    data = env_a.reset()
    np.random.seed(7)
    states_a = []
    for step in range(40):
        if data.u_ref is None:
            data.update(u_ref=env_a.u_ref(data))
        a = algo_a.step(data, prob=0.5)
        data, r, done, info = env_a.step(a)
        states_a.append(data.states.clone())
        if done:
            break

    # captured loop (same weights by construction: same seed -> same init)
    env_b, algo_b = build()
    env_b.reset()
    assert engine_supported(env_b, algo_b)
    np.random.seed(7)
    eng = RolloutEngine(env_b, algo_b)
    states_b = []
    for step in range(len(states_a)):
        done = eng.step(prob=0.5)
        states_b.append(eng.states[eng.cur].clone())
        if done:
            break

    assert len(states_a) == len(states_b)
    for t, (sa, sb) in enumerate(zip(states_a, states_b)):
        assert torch.allclose(sa, sb, atol=5e-4), \
            (t, (sa - sb).abs().max())
    # buffers got the same number of graphs with matching edge counts
    assert algo_a.buffer.size == algo_b.buffer.size
    for ga, gb in zip(algo_a.buffer.data, algo_b.buffer.data):
        assert ga.num_edges == gb.num_edges


def test_rollout_engine_update_interleave():
    """Engine + updates: weights change between replays and the captured
    graph must pick them up (actor output changes)."""
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.rollout import RolloutEngine
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=32)
    env.reset()
    eng = RolloutEngine(env, algo)
    for step in range(1, 65):
        done = eng.step(prob=0.1)
        if done:
            eng.reload()
        if algo.is_update(step):
            out = algo.update(step, None)
    assert all(0 <= v <= 1 for v in out.values())


class TestFusedLinear:
    """MFMA bf16 GEMM kernel (ops/hip/fused_linear.hip) numerics."""

    def test_identity_layout(self):
        # A = I picks out rows of W^T: catches any fragment-layout mixup
        from gcbf_amd import _C
        M, K, N = 128, 64, 128
        A = torch.zeros(M, K, device="cuda")
        for i in range(K):
            A[i, i] = 1.0
        W = torch.randn(N, K, device="cuda")  # asymmetric
        out = _C.fused_linear(A.bfloat16(), W.bfloat16(), None, 0, True)
        ref = W.t().float()
        assert torch.allclose(out[:K], W.bfloat16().float().t(), atol=1e-3), \
            (out[:K] - ref[:K]).abs().max()
        assert out[K:].abs().max() == 0

    @pytest.mark.parametrize("M,K,N", [(256, 64, 128), (512, 2048, 2048),
                                       (1024, 2048, 256), (256, 320, 2048),
                                       (2048, 256, 128)])
    def test_matches_blas_bf16(self, M, K, N):
        from gcbf_amd import _C
        torch.manual_seed(M + K + N)
        A = torch.randn(M, K, device="cuda")
        W = torch.randn(N, K, device="cuda") / (K ** 0.5)
        b = torch.randn(N, device="cuda")
        out = _C.fused_linear(A.bfloat16(), W.bfloat16(), b, 0, True)
        ref = (A.bfloat16() @ W.bfloat16().t()).float() + b
        err = (out - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err < 2e-2 * max(scale, 1.0), (err, scale)

    def test_relu_tanh_epilogues(self):
        from gcbf_amd import _C
        torch.manual_seed(0)
        A = torch.randn(256, 128, device="cuda")
        W = torch.randn(128, 128, device="cuda") / 12.0
        b = torch.randn(128, device="cuda")
        ref = (A.bfloat16() @ W.bfloat16().t()).float() + b
        out_r = _C.fused_linear(A.bfloat16(), W.bfloat16(), b, 1, True)
        assert torch.allclose(out_r, torch.relu(ref), atol=2e-2)
        out_t = _C.fused_linear(A.bfloat16(), W.bfloat16(), b, 2, True)
        assert torch.allclose(out_t, torch.tanh(ref), atol=2e-2)

    def test_bf16_output(self):
        from gcbf_amd import _C
        A = torch.randn(128, 64, device="cuda")
        W = torch.randn(128, 64, device="cuda")
        out = _C.fused_linear(A.bfloat16(), W.bfloat16(), None, 0, False)
        assert out.dtype == torch.bfloat16

    def test_mlp_fused_plan_matches_eager(self):
        from gcbf_amd.nn import MLP
        torch.manual_seed(0)
        mlp = MLP(13, 256, (2048, 2048), limit_lip=True).cuda()
        mlp.eval()
        x = torch.randn(700, 13, device="cuda")
        with torch.no_grad():
            ref = mlp(x)            # eager fp32
            mlp.fused_mfma = True
            out = mlp(x)            # MFMA bf16 path
        rel = (out - ref).abs().max() / ref.abs().max().clamp_min(1e-3)
        assert rel < 0.05, rel

    def test_fused_backward_grads(self):
        from gcbf_amd.nn.fused import fused_linear_act
        torch.manual_seed(0)
        M, K, N = 256, 128, 128
        x = torch.randn(M, K, device="cuda", requires_grad=True)
        w = (torch.randn(N, K, device="cuda") / 10).requires_grad_(True)
        b = torch.randn(N, device="cuda", requires_grad=True)
        out = fused_linear_act(x, w, b, 1, True)
        g = torch.randn_like(out)
        out.backward(g)

        # reference gradients USING THE SAME relu mask as the fused output
        # (bf16 rounding flips marginal pre-activations, so masks from an
        # fp32 recomputation differ on a few elements — that is inherent to
        # bf16 training, not a kernel bug)
        gm = g * (out > 0)
        gm16 = gm.bfloat16()
        dx_ref = (gm16 @ w.detach().bfloat16()).float()
        dw_ref = (gm16.t() @ x.detach().bfloat16()).float()
        db_ref = gm.sum(0)
        assert torch.allclose(x.grad, dx_ref, atol=1e-3), \
            (x.grad - dx_ref).abs().max()
        assert torch.allclose(w.grad, dw_ref, atol=1e-3), \
            (w.grad - dw_ref).abs().max()
        assert torch.allclose(b.grad, db_ref, atol=1e-3)


@pytest.mark.parametrize("env_name,n,obs,algo_name,mn", [
    ("SimpleCar", 16, None, "macbf", 12),     # BASELINE config 5
    ("SimpleDrone", 16, None, "gcbf", None),  # config 3 (single-GPU slice)
])
def test_baseline_configs_train_gpu(env_name, n, obs, algo_name, mn):
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    env = make_env(env_name, n, dev, max_neighbors=mn)
    env.train()
    algo = make_algo(algo_name, env, n, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=32)
    data = env.reset()
    for step in range(1, 65):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.6)
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
        if algo.is_update(step):
            out = algo.update(step, None)
    assert all(np.isfinite(v) for v in out.values())


@pytest.mark.timeout(600)
def test_stress_config_n256_updates():
    """BASELINE config 4: DubinsCar n=256 + 32 obstacles — one update must
    run within memory/time budget on one GPU."""
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    e0 = make_env("DubinsCar", 256, dev)
    p = e0.default_params
    p["num_obs"] = 32
    # keep the n=16 packing density: 256 agents need a 16x16 area (the
    # default 4x4 is near the random-sequential-packing jamming limit and
    # the reference's rejection sampler would not terminate either)
    p["area_size"] = 16.0
    env = make_env("DubinsCar", 256, dev, params=p)
    env.train()
    algo = make_algo("gcbf", env, 256, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=32)
    from gcbf_amd.utils.amp import enable_bf16
    enable_bf16(algo)
    data = env.reset()
    assert data.num_nodes == 288
    for step in range(1, 33):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.9)
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
        if algo.is_update(step):
            out = algo.update(step, None)
    assert all(np.isfinite(v) for v in out.values())


def test_segment_attn_deterministic():
    """No atomics, fixed CSR reduction order: repeated runs are bitwise
    identical (SURVEY.md §5.2 determinism requirement)."""
    msg, gate, dst = _rand_edges(5000, 300, 256, seed=11)
    out1 = ops.segment_attn_aggregate(msg, gate, dst, 300)
    out2 = ops.segment_attn_aggregate(msg, gate, dst, 300)
    assert torch.equal(out1, out2)
    m = msg.clone().requires_grad_(True)
    g = gate.clone().requires_grad_(True)
    o = ops.segment_attn_aggregate(m, g, dst, 300)
    go = torch.randn_like(o)
    o.backward(go)
    g1m, g1g = m.grad.clone(), g.grad.clone()
    m.grad = None; g.grad = None
    o = ops.segment_attn_aggregate(m, g, dst, 300)
    o.backward(go)
    assert torch.equal(m.grad, g1m)
    assert torch.equal(g.grad, g1g)


def test_fused_linear_deterministic():
    from gcbf_amd import _C
    A = torch.randn(512, 2048, device="cuda").bfloat16()
    W = torch.randn(2048, 2048, device="cuda").bfloat16()
    b = torch.randn(2048, device="cuda")
    o1 = _C.fused_linear(A, W, b, 1, True)
    o2 = _C.fused_linear(A, W, b, 1, True)
    assert torch.equal(o1, o2)


def test_segment_max_hip_matches_eager():
    msg, _, dst = _rand_edges(500, 40, 128, seed=5)
    out = ops.segment_max(msg, dst, 40)
    ref = eager.segment_max(msg.cpu(), dst.cpu(), 40)
    assert torch.allclose(out.cpu(), ref, atol=1e-6)
    # backward: gradient lands on the argmax edges only
    m = msg.clone().requires_grad_(True)
    o = ops.segment_max(m, dst, 40)
    o.sum().backward()
    m2 = msg.cpu().requires_grad_(True)
    o2 = eager.segment_max(m2, dst.cpu(), 40)
    o2.sum().backward()
    assert torch.allclose(m.grad.cpu(), m2.grad, atol=1e-6)


def test_rollout_engine_soft_capacity_overflow_fallback():
    """Soft-capacity engine: with an edge capacity barely above the initial
    edge count, steps must overflow into the eager fallback and later resume
    captured replay, with the buffer/state staying consistent throughout."""
    from gcbf_amd.env import make_env
    from gcbf_amd.algo import make_algo
    from gcbf_amd.rollout import RolloutEngine, engine_supported
    from gcbf_amd.trainer.utils import set_seed
    set_seed(5)
    dev = torch.device("cuda")
    e0 = make_env("DubinsCar", 64, dev)
    p = e0.default_params
    p["num_obs"] = 8
    p["area_size"] = 8.0
    env = make_env("DubinsCar", 64, dev, params=p)
    env.train()
    algo = make_algo("gcbf", env, 64, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=512)
    data = env.reset()
    assert engine_supported(env, algo)
    eng = RolloutEngine(env, algo, edge_capacity=data.num_edges + 8)
    saw_eager = saw_captured = False
    for t in range(120):
        was_eager = eng._eager
        saw_eager |= was_eager
        saw_captured |= not was_eager
        done = eng.step(prob=0.5)
        assert torch.isfinite(eng.states[eng.cur]).all()
        if not eng._eager:
            assert torch.isfinite(eng.ea[eng.cur][: eng.E]).all()
        if done:
            eng.reload()
    assert algo.buffer.size == 120
    assert saw_captured
    # with capacity this tight the moving scene must overflow at least once
    assert saw_eager


@pytest.mark.skipif(
    __import__("os").environ.get("GCBF_AMD_UPDATE_CAPTURE") != "1",
    reason="captured update engine is experimental (opt-in via "
           "GCBF_AMD_UPDATE_CAPTURE=1)")
def test_update_engine_matches_eager():
    """Captured update engine vs eager: one inner iteration's gradients on
    the SAME sampled batch must agree per-parameter (cosine + norm ratio),
    and the loss scalars must match to bf16 noise.

    Grad cosines are the right criterion: strict weight comparison over 10
    Adam iterations compounds inherent bf16 reduction-order noise and the
    one-step spectral-norm power-iteration offset (the engine re-links
    before the doubled CBF forward; eager after), while genuine capture
    corruption shows up as cosine ~0 with norms off by orders of magnitude.
    """
    from gcbf_amd.env import make_env
    from gcbf_amd.algo import make_algo
    from gcbf_amd.graph import GraphBatch
    from gcbf_amd.rollout import RolloutEngine
    from gcbf_amd.trainer.utils import set_seed
    from gcbf_amd.utils.amp import enable_bf16

    set_seed(11)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=512)
    enable_bf16(algo)
    env.reset()
    reng = RolloutEngine(env, algo)
    for _ in range(512):
        if reng.step(prob=0.7):
            reng.reload()
    # freeze the spectral-norm power iteration BEFORE the engine captures:
    # in train mode each path advances u/v at different points (engine:
    # relinked CBF then doubled CBF inside the capture; eager: doubled
    # then relinked), so σ differs by one power step between the
    # differentiable passes — the ReLU-gated ḣ loss amplifies that by
    # 1/dt into gradient-support flips on small-norm params (measured:
    # losses match to bf16 noise, tiny-norm cosines scatter).  eval()
    # pins σ so this test isolates what it is meant to verify: the
    # CAPTURE mechanics (gather/build/replay/backward correctness).
    algo.cbf.eval()
    algo.update(512)
    e = algo._upd_engine
    assert e is not None, "update engine must build on GPU"
    for _ in range(512):
        if reng.step(prob=0.7):
            reng.reload()

    params = [p for p in algo.cbf.parameters() if p.requires_grad] + \
             [p for p in algo.actor.parameters() if p.requires_grad]
    names = [f"cbf.{n}" for n, p in algo.cbf.named_parameters()
             if p.requires_grad] + \
            [f"actor.{n}" for n, p in algo.actor.named_parameters()
             if p.requires_grad]

    def eager_loss(gl):
        p = algo.params
        eps, alpha = p["eps"], p["alpha"]
        # buffer snapshots are metadata-only under the ring-backed rollout;
        # materialize the batch from the ring (bitwise equal to from_list —
        # test_ring_batch_matches_from_list_gpu)
        graphs = algo._ring.batch(gl)
        actions = algo.actor(graphs)
        graphs_next = env.forward_graph(graphs, actions)
        both = GraphBatch.from_list([graphs, graphs_next])
        h_both = algo.cbf(both)
        n_ag = h_both.shape[0] // 2
        h, h_next = h_both[:n_ag], h_both[n_ag:]
        hv = h[:, 0]
        um = env.unsafe_mask(graphs).to(hv.dtype)
        any_u = (um.sum() > 0).to(hv.dtype)
        loss_unsafe = any_u * (torch.relu(hv + eps) * um).sum() \
            / um.sum().clamp(min=1)
        sm = env.safe_mask(graphs).to(hv.dtype)
        any_s = (sm.sum() > 0).to(hv.dtype)
        loss_safe = any_s * (torch.relu(-hv + eps) * sm).sum() \
            / sm.sum().clamp(min=1)
        with torch.no_grad():
            relinked = env.add_communication_links_batched(
                graphs_next.detach())
            h_new = algo.cbf(relinked)
        h_dot = (h_next - h) / env.dt
        residue = ((h_new - h) / env.dt - h_dot).detach()
        h_dot = residue + h_dot
        loss_h_dot = torch.mean(torch.relu(-h_dot - alpha * h + eps))
        loss_action = torch.mean(torch.square(actions).sum(dim=1))
        return (p["loss_unsafe_coef"] * loss_unsafe +
                p["loss_safe_coef"] * loss_safe +
                p["loss_h_dot_coef"] * loss_h_dot +
                p["loss_action_coef"] * loss_action,
                torch.stack([loss_unsafe, loss_safe, loss_h_dot,
                             loss_action]).detach())

    gl = e._sample_for_warmup()
    e._fill_inputs(gl)
    e.gFront.replay()
    assert int(e._ecounts.max().cpu()) <= e.E_cap
    lcap, log7 = e._graphed(e._nodes, e._uref, e._ei, e._ea, e._seg,
                            e._h_new, e.w_dev)
    e._zero_grads()
    lcap.backward()
    g_cap = [p.grad.clone() for p in params]
    log_cap = log7[:4].clone()

    loss, log_eag = eager_loss(gl)
    g_eag = torch.autograd.grad(loss, params)

    # loss scalars: bf16 noise only (h_dot sees the /dt-amplified residue
    # noise, allow more)
    rel = (log_cap - log_eag).abs() / log_eag.abs().clamp(min=1e-3)
    assert float(rel[0]) < 0.15 and float(rel[1]) < 0.15 \
        and float(rel[3]) < 0.15, (log_cap, log_eag)
    assert float(rel[2]) < 0.35, (log_cap, log_eag)

    bad = []
    for n, a, b in zip(names, g_cap, g_eag):
        bn = b.float().norm().item()
        if bn < 1e-6:
            continue  # numerically-zero eager grad: ratio is meaningless
        an = a.float().norm().item()
        cos = torch.nn.functional.cosine_similarity(
            a.float().flatten(), b.float().flatten(), dim=0).item()
        if cos < 0.97 or not (1 / 3 <= an / bn <= 3):
            bad.append(f"{n}: cos={cos:+.4f} |cap|={an:.3e} |eag|={bn:.3e}")
    assert not bad, "\n".join(bad)

    # and a full engine iteration must run end to end with finite logs
    out = e.try_iter(e._sample_for_warmup())
    assert out is not None and torch.isfinite(out).all()


@pytest.mark.skipif(
    __import__("os").environ.get("GCBF_AMD_UPDATE_CAPTURE") != "1",
    reason="captured update engine is experimental (opt-in via "
           "GCBF_AMD_UPDATE_CAPTURE=1)")
def test_update_engine_with_obstacles():
    """Captured update engine on a scene WITH obstacle nodes (static
    agent-index layout): engine must build and run finite iterations."""
    from gcbf_amd.env import make_env
    from gcbf_amd.algo import make_algo
    from gcbf_amd.rollout import RolloutEngine
    from gcbf_amd.trainer.utils import set_seed
    from gcbf_amd.utils.amp import enable_bf16

    set_seed(3)
    dev = torch.device("cuda")
    e0 = make_env("DubinsCar", 16, dev)
    p = e0.default_params
    p["num_obs"] = 4
    env = make_env("DubinsCar", 16, dev, params=p)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=256)
    enable_bf16(algo)
    env.reset()
    eng = RolloutEngine(env, algo)
    for _ in range(256):
        if eng.step(prob=0.7):
            eng.reload()
    out = algo.update(256)
    assert algo._upd_engine is not None, "engine must build with obstacles"
    assert all(0 <= v <= 1 for v in out.values())


@pytest.mark.parametrize("obs", [0, 6])
def test_ring_batch_matches_from_list_gpu(obs):
    """RingStore.batch (HIP batched rebuild) must equal from_list on GPU:
    the stored edges came from the same deterministic kernels."""
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.graph import GraphBatch
    from gcbf_amd.rollout import RolloutEngine
    from gcbf_amd.trainer.utils import set_seed

    set_seed(7)
    dev = torch.device("cuda")
    e0 = make_env("DubinsCar", 16, dev)
    p = e0.default_params
    p["num_obs"] = obs
    env = make_env("DubinsCar", 16, dev, params=p)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=64)
    # eager stepping: the buffer keeps REAL graph tensors, so from_list is
    # an independent ground truth for the ring rebuild
    data = env.reset()
    for _ in range(64):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.6)
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
    algo._make_ring()
    assert algo._ring is not None
    gl = algo.buffer.sample(12, 3)
    assert algo._ring.usable(gl)
    fast = algo._ring.batch(gl)
    ref = GraphBatch.from_list(gl)
    assert torch.equal(fast.states, ref.states)
    assert torch.equal(fast.edge_index, ref.edge_index)
    assert torch.allclose(fast.edge_attr, ref.edge_attr, atol=1e-6), \
        (fast.edge_attr - ref.edge_attr).abs().max()
    assert torch.equal(fast.u_ref, ref.u_ref)


def test_build_graph_padded_exact_capacity():
    """Adversarial: E_max exactly equal to the true edge count — every
    edge present, e_count exact, no pad rows."""
    from gcbf_amd import _C
    from gcbf_amd import ops
    torch.manual_seed(0)
    dev = torch.device("cuda")
    B, N, n_rec = 3, 12, 12
    pos = torch.rand(B * N, 2, device=dev) * 2.0
    states = torch.cat([pos, torch.randn(B * N, 2, device=dev)], dim=1)
    ei_exact, ea_exact = _C.build_graph(
        pos.contiguous(), states.contiguous(), B, n_rec, 1.0, -1,
        int(ops.ATTR_DIFF), 4)
    E = ei_exact.shape[1]
    if E == 0:
        pytest.skip("no edges for this layout")
    ei, seg, ea, ecount = _C.build_graph_padded(
        pos.contiguous(), states.contiguous(), B, n_rec, 1.0, -1,
        int(ops.ATTR_DIFF), 4, E)   # capacity == exact count
    assert int(ecount[0]) == E
    assert torch.equal(ei[:, :E], ei_exact)
    assert torch.allclose(ea[:E], ea_exact, atol=1e-6)
    assert (seg[:E] < B * N).all()          # no sentinel inside real rows


def test_build_graph_padded_overflow_is_flagged_not_oob():
    """Adversarial: capacity BELOW the true edge count — e_count reports
    the true count (callers detect overflow), buffers stay well-formed
    (indices in range, attrs finite), and no out-of-bounds write occurs
    (validated under AMD_SERIALIZE_KERNEL in the sanitizer pass)."""
    from gcbf_amd import _C
    from gcbf_amd import ops
    torch.manual_seed(1)
    dev = torch.device("cuda")
    B, N, n_rec = 2, 16, 16
    pos = torch.rand(B * N, 2, device=dev) * 0.5   # dense: many edges
    states = torch.cat([pos, torch.randn(B * N, 2, device=dev)], dim=1)
    ei_exact, _ = _C.build_graph(
        pos.contiguous(), states.contiguous(), B, n_rec, 1.0, -1,
        int(ops.ATTR_DIFF), 4)
    E = ei_exact.shape[1]
    assert E > 8, "layout should be dense"
    E_max = E // 2
    # guard tensors around the allocation to catch OOB writes
    canary_lo = torch.full((256,), 7.0, device=dev)
    ei, seg, ea, ecount = _C.build_graph_padded(
        pos.contiguous(), states.contiguous(), B, n_rec, 1.0, -1,
        int(ops.ATTR_DIFF), 4, E_max)
    canary_hi = torch.full((256,), 9.0, device=dev)
    torch.cuda.synchronize()
    assert int(ecount[0]) == E          # true count reported -> overflow
    assert ei.shape[1] == E_max
    assert (ei >= 0).all() and (ei < B * N).all()
    assert torch.isfinite(ea).all()
    assert (seg >= 0).all() and (seg <= B * N).all()
    assert (canary_lo == 7.0).all() and (canary_hi == 9.0).all()


def test_kth_smallest_topk_degenerate_ties():
    """Adversarial: ALL pairwise distances identical (regular polygon
    center + duplicated radius) — the k-th-smallest cap must keep ties
    without crashing, matching the eager rule (>= k edges per row kept,
    every kept distance <= kth)."""
    from gcbf_amd import _C
    from gcbf_amd import ops
    dev = torch.device("cuda")
    n = 8
    # all agents at distance exactly 0.5 from each other is impossible in
    # 2D for n>3; instead: co-located pairs produce exact zero-distance
    # ties plus identical cross distances
    base = torch.tensor([[0.0, 0.0], [0.3, 0.0], [0.0, 0.3], [0.3, 0.3]],
                        device=dev)
    pos = torch.cat([base, base])       # each point duplicated (ties)
    states = torch.cat([pos, torch.zeros(n, 2, device=dev)], dim=1)
    k = 3
    ei, ea = _C.build_graph(pos.contiguous(), states.contiguous(), 1, n,
                            1.0, k, int(ops.ATTR_DIFF), 4)
    torch.cuda.synchronize()
    assert torch.isfinite(ea).all()
    # every receiver keeps at least k senders; ties may exceed k
    dst = ei[1]
    counts = torch.bincount(dst, minlength=n)
    assert (counts >= k).all()
    # parity with the eager oracle (same tie semantics)
    from gcbf_amd.ops import eager
    ei_ref = eager.dense_radius_graph(pos, None, 1.0, k, 1)
    have = set(map(tuple, ei.t().tolist()))
    want = set(map(tuple, ei_ref.t().tolist()))
    assert have == want
