"""GPU micro-benchmarks: GEMM dtype comparison, rollout breakdown.

Run on an MI355X box:  python tools/micro_bench.py
Prints per-op timings to locate where rollout/update time goes and whether
bf16 GEMMs are healthy on this torch build.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def gemm_suite():
    print("== GEMM F.linear timings (ms) ==")
    shapes = [(16384, 13, 2048), (16384, 2048, 2048), (16384, 2048, 256),
              (4900, 260, 2048), (4900, 2048, 2048), (4900, 2048, 1024),
              (4900, 1024, 512), (16384, 256, 1)]
    for M, K, N in shapes:
        x32 = torch.randn(M, K, device="cuda")
        w32 = torch.randn(N, K, device="cuda")
        b32 = torch.randn(N, device="cuda")
        x16, w16, b16 = x32.bfloat16(), w32.bfloat16(), b32.bfloat16()
        t_fp32 = timeit(lambda: F.linear(x32, w32, b32))
        t_bf16 = timeit(lambda: F.linear(x16, w16, b16))

        def ac():
            with torch.autocast("cuda", dtype=torch.bfloat16):
                return F.linear(x32, w32, b32)
        t_ac = timeit(ac)
        tf = 2 * M * K * N / 1e9  # GFLOP
        print(f"  M={M:6d} K={K:5d} N={N:5d}: fp32 {t_fp32:7.3f} "
              f"({tf / t_fp32:6.0f} GF/s/ms={tf/t_fp32:6.0f} TF) | "
              f"bf16 {t_bf16:7.3f} ({tf / t_bf16:6.0f} TF) | "
              f"autocast {t_ac:7.3f}")


def relu_cost():
    print("== elementwise epilogue cost (ms) ==")
    x = torch.randn(16384, 2048, device="cuda")
    print(f"  relu  16K x 2048 fp32: {timeit(lambda: torch.relu(x)):.3f}")
    xb = x.bfloat16()
    print(f"  relu  16K x 2048 bf16: {timeit(lambda: torch.relu(xb)):.3f}")
    print(f"  cast  fp32->bf16:      {timeit(lambda: x.bfloat16()):.3f}")


def sn_cost():
    print("== SNLinear overhead (ms) ==")
    from gcbf_amd.nn.mlp import SNLinear
    m = SNLinear(2048, 2048).cuda()
    x = torch.randn(8192, 2048, device="cuda")
    m.train()
    print(f"  SNLinear train fwd:      {timeit(lambda: m(x)):.3f}")
    def ac():
        with torch.autocast('cuda', dtype=torch.bfloat16):
            return m(x)
    print(f"  SNLinear train fwd ac:   {timeit(ac):.3f}")
    lin = torch.nn.Linear(2048, 2048).cuda()
    print(f"  plain Linear fwd:        {timeit(lambda: lin(x)):.3f}")


def update_breakdown():
    print("== GCBF update step breakdown ==")
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.graph import GraphBatch
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=512)
    data = env.reset()
    for step in range(1, 513):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.9)
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
    graph_list = algo.buffer.sample(512 // 5, 3)
    graphs = GraphBatch.from_list(graph_list)
    print(f"  batch: {graphs.num_graphs} graphs, {graphs.num_nodes} nodes, "
          f"{graphs.num_edges} edges")

    t = timeit(lambda: algo.cbf(graphs), iters=20)
    print(f"  cbf forward fp32:   {t:.2f} ms")
    t = timeit(lambda: algo.actor(graphs), iters=20)
    print(f"  actor forward fp32: {t:.2f} ms")

    def fwd_bwd():
        h = algo.cbf(graphs)
        a = algo.actor(graphs)
        (h.sum() + a.sum()).backward()
    t = timeit(fwd_bwd, iters=10)
    print(f"  fwd+bwd fp32:       {t:.2f} ms")

    from gcbf_amd.utils.amp import enable_bf16
    enable_bf16(algo)
    t = timeit(lambda: algo.cbf(graphs), iters=20)
    print(f"  cbf forward bf16:   {t:.2f} ms")

    def fwd_bwd16():
        h = algo.cbf(graphs)
        a = algo.actor(graphs)
        (h.sum() + a.sum()).backward()
    t = timeit(fwd_bwd16, iters=10)
    print(f"  fwd+bwd bf16:       {t:.2f} ms")

    t = timeit(lambda: env.unsafe_mask(graphs), iters=20)
    print(f"  unsafe_mask batch:  {t:.3f} ms")
    t = timeit(lambda: env.add_communication_links_batched(
        graphs.replace()), iters=20)
    print(f"  relink batch:       {t:.3f} ms")


def rollout_breakdown():
    print("== rollout step breakdown (DubinsCar n=16) ==")
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.trainer.utils import set_seed
    set_seed(0)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=512)
    data = env.reset()
    data.update(u_ref=env.u_ref(data))

    print(f"  u_ref:           {timeit(lambda: env.u_ref(data)):.3f} ms")
    print(f"  actor fwd:       {timeit(lambda: algo.act(data)):.3f} ms")
    print(f"  unsafe_mask:     {timeit(lambda: env.unsafe_mask(data)):.3f} ms")
    a = torch.zeros(16, 2, device=dev)
    print(f"  env.step:        {timeit(lambda: env.step(a)):.3f} ms")
    print(f"  relink single:   "
          f"{timeit(lambda: env.add_communication_links(data.replace())):.3f}"
          f" ms")


def shape_churn():
    """Is bf16 slow when the GEMM M dimension changes every call
    (hipBLASLt algorithm search per novel shape)?"""
    print("== shape churn: K=2048 N=2048, M varies per call ==")
    w32 = torch.randn(2048, 2048, device="cuda")
    w16 = w32.bfloat16()
    for dt, w in [("fp32", w32), ("bf16", w16)]:
        xs = [torch.randn(9000 + 37 * i, 2048, device="cuda",
                          dtype=w.dtype) for i in range(40)]
        # first pass: novel shapes every call
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for x in xs:
            F.linear(x, w)
        torch.cuda.synchronize()
        t_novel = (time.perf_counter() - t0) / len(xs) * 1e3
        # second pass: same shapes again (cached)
        t0 = time.perf_counter()
        for x in xs:
            F.linear(x, w)
        torch.cuda.synchronize()
        t_cached = (time.perf_counter() - t0) / len(xs) * 1e3
        print(f"  {dt}: novel shapes {t_novel:7.3f} ms/call, "
              f"repeat shapes {t_cached:7.3f} ms/call")


if __name__ == "__main__":
    assert torch.cuda.is_available()
    shape_churn()
    gemm_suite()
    relu_cost()
    sn_cost()
    rollout_breakdown()
    update_breakdown()
