"""Diagnose hipGraph capture: try capturing each rollout component alone."""
import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from gcbf_amd import ops
from gcbf_amd.algo import make_algo
from gcbf_amd.env import make_env
from gcbf_amd.graph import GraphBatch
from gcbf_amd.trainer.utils import set_seed


def try_capture(name, fn):
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                fn()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        g.replay()
        torch.cuda.synchronize()
        print(f"[OK]   {name}")
        return True
    except Exception:
        print(f"[FAIL] {name}")
        traceback.print_exc(limit=8)
        torch.cuda.synchronize()
        return False


def main():
    set_seed(0)
    dev = torch.device("cuda")
    env = make_env("DubinsCar", 16, dev)
    env.train()
    algo = make_algo("gcbf", env, 16, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=512)
    data = env.reset()
    data.update(u_ref=env.u_ref(data))
    from gcbf_amd import _C

    N, n = data.num_nodes, 16
    E_max = n * (N - 1)
    states = data.states.clone().contiguous()
    goal = env._goal.clone().contiguous()
    action = torch.zeros(n, 2, device=dev)

    # 1. fused env step kernel
    try_capture("env_step_fused", lambda: ops.env_step_fused(
        "dubins", states, goal, action, env.dt, 0.05, 0.8, 0.05, 2.0))

    # 2. padded graph build
    try_capture("build_graph_padded", lambda: _C.build_graph_padded(
        states[:, :2].contiguous(), states, 1, n, 1.0, -1, 1, 5, E_max))

    # 3. fused masks
    try_capture("fused_masks", lambda: _C.fused_masks(
        states, 1, n, 0.05, 1, False, True, False))

    # 4. actor forward on padded graph
    ei, seg, ea, ec = _C.build_graph_padded(
        states[:, :2].contiguous(), states, 1, n, 1.0, -1, 1, 5, E_max)
    am = data.agent_mask
    x = data.x.clone()
    u_ref = env.u_ref(data).clone()

    def actor_fwd():
        d = GraphBatch(x=x, pos=states[:, :2], states=states, edge_index=ei,
                       edge_attr=ea, agent_mask=am, u_ref=u_ref)
        d.seg_dst = seg
        with torch.no_grad():
            return algo.actor(d)

    try_capture("actor_forward", actor_fwd)

    # 5. segment op alone
    def seg_op():
        msg = torch.randn(E_max, 8, device=dev)
        gate = torch.randn(E_max, 1, device=dev)
        return ops.segment_attn_aggregate(msg, gate, seg, N)

    try_capture("segment_attn", seg_op)

    # 6. whole engine body
    from gcbf_amd.rollout import RolloutEngine
    try:
        eng = RolloutEngine(env, algo)
        for _ in range(5):
            done = eng.step(0.5)
        torch.cuda.synchronize()
        print("[OK]   RolloutEngine end-to-end")
    except Exception:
        print("[FAIL] RolloutEngine")
        traceback.print_exc(limit=12)


if __name__ == "__main__":
    main()
