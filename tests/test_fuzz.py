"""Property-based fuzzing of the hot-path ops (hypothesis)."""
import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from gcbf_amd import ops
from gcbf_amd.graph import GraphBatch
from gcbf_amd.ops import eager


@settings(max_examples=25, deadline=None)
@given(E=st.integers(1, 200), n=st.integers(1, 40), d=st.integers(1, 32),
       seed=st.integers(0, 10_000))
def test_fuzz_segment_attn(E, n, d, seed):
    g = torch.Generator().manual_seed(seed)
    dst = torch.sort(torch.randint(0, n, (E,), generator=g)).values
    msg = torch.randn(E, d, generator=g)
    gate = torch.randn(E, 1, generator=g)
    out = eager.segment_attn_aggregate(msg, gate, dst, n)
    assert out.shape == (n, d)
    assert torch.isfinite(out).all()
    # each row is a convex combination of its segment's messages
    for seg in range(n):
        idx = (dst == seg).nonzero()[:, 0]
        if idx.numel():
            lo = msg[idx].min(dim=0).values - 1e-4
            hi = msg[idx].max(dim=0).values + 1e-4
            assert (out[seg] >= lo).all() and (out[seg] <= hi).all()
        else:
            assert (out[seg] == 0).all()


@settings(max_examples=20, deadline=None)
@given(N=st.integers(2, 24), n_rec=st.integers(1, 24),
       r=st.floats(0.1, 2.0), seed=st.integers(0, 10_000),
       topk=st.one_of(st.none(), st.integers(1, 8)))
def test_fuzz_radius_graph_invariants(N, n_rec, r, seed, topk):
    n_rec = min(n_rec, N)
    g = torch.Generator().manual_seed(seed)
    pos = torch.rand(N, 2, generator=g) * 3
    am = None
    if n_rec != N:
        am = torch.zeros(N, dtype=torch.bool)
        am[:n_rec] = True
    ei = eager.dense_radius_graph(pos, am, r, topk, 1)
    src, dst = ei
    assert (src != dst).all()                      # no self loops
    assert (dst < n_rec).all()                     # receivers are agents
    if ei.numel():
        assert (dst.diff() >= 0).all()             # dst-sorted
        d = (pos[src] - pos[dst]).norm(dim=1)
        assert (d < r + 1e-5).all()                # within radius
        if topk is not None:
            counts = torch.bincount(dst, minlength=n_rec)
            assert counts.max() <= max(topk, 1)


@settings(max_examples=10, deadline=None)
@given(B=st.integers(1, 4), n=st.integers(2, 8), seed=st.integers(0, 5000))
def test_fuzz_batched_masks_match_per_graph(B, n, seed):
    from gcbf_amd.env import make_env
    torch.manual_seed(seed)
    env = make_env("DubinsCar", n, torch.device("cpu"))
    env.train()
    graphs = []
    for _ in range(B):
        torch.manual_seed(seed + len(graphs))
        graphs.append(env.reset())
    batch = GraphBatch.from_list([g.replace() for g in graphs])
    for fn in (env.safe_mask, env.unsafe_mask, env.collision_mask):
        batched = fn(batch)
        per = torch.cat([fn(g) for g in graphs])
        assert torch.equal(batched, per)
