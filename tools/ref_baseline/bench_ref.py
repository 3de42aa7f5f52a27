"""Reference-implementation throughput baseline (BASELINE.md protocol).

Drives the UNMODIFIED reference (env/algo classes from /root/reference via
the pygshim dependency shim) through EXACTLY the same steady-state
measurement protocol as /root/repo/bench.py: prefill through a full update
cycle, exploration Bernoulli at the 500k-schedule midpoint (prob 0.5),
natural updates inline in the timed window, fractional owed update charged
from a measured steady-state update.  The rollout step reproduces the
reference trainer's loop body verbatim (gcbf/trainer/trainer.py:60-73).

    python tools/ref_baseline/bench_ref.py --steps 512 --env DubinsCar -n 16
"""
import argparse
import json
import os
import sys
import time

HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    sys.path.insert(0, HERE)
    import pygshim
    pygshim.install()
    sys.path.insert(0, "/root/reference")

    import numpy as np
    import torch
    from gcbf.algo import make_algo
    from gcbf.env import make_env
    from gcbf.trainer.utils import read_params, set_seed
    from torch_geometric.data import Data

    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=256)
    p.add_argument("--warmup", type=int, default=16)
    p.add_argument("--env", type=str, default="DubinsCar")
    p.add_argument("-n", "--num-agents", type=int, default=16)
    p.add_argument("--obs", type=int, default=0)
    p.add_argument("--batch-size", type=int, default=512)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--cpu", action="store_true")
    args = p.parse_args()

    set_seed(args.seed)
    use_cuda = torch.cuda.is_available() and not args.cpu
    device = torch.device("cuda" if use_cuda else "cpu")

    env = make_env(args.env, args.num_agents, device)
    params = env.default_params
    params["num_obs"] = args.obs
    env = make_env(args.env, args.num_agents, device, params=params)
    env.train()
    hyper = read_params(args.env, "gcbf")
    algo = make_algo("gcbf", env, args.num_agents, env.node_dim,
                     env.edge_dim, env.action_dim, device,
                     args.batch_size, hyperparams=hyper)

    prob = 0.5
    state = {"data": env.reset(), "step": 0}

    def sync():
        if use_cuda:
            torch.cuda.synchronize()

    def rollout_step():
        # reference trainer loop body (gcbf/trainer/trainer.py:60-70)
        state["step"] += 1
        data = state["data"]
        data.update(Data(u_ref=env.u_ref(data)))
        action = algo.step(data, prob=1.0 if np.random.rand() < prob
                           else 0.0)
        next_data, reward, done, info = env.step(action)
        state["data"] = env.reset() if done else next_data

    # the reference update logs scalars unconditionally (.item() host syncs
    # per inner iter are part of its published cost)
    writer = pygshim.SummaryWriter()

    def run_update():
        algo.update(state["step"], writer)

    for _ in range(args.warmup):
        rollout_step()
    while len(algo.buffer.data) < args.batch_size:
        rollout_step()
    run_update()
    while len(algo.buffer.data) < args.batch_size:
        rollout_step()
    sync()
    t0 = time.perf_counter()
    run_update()
    sync()
    t_update = time.perf_counter() - t0

    in_region = 0
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        rollout_step()
        if len(algo.buffer.data) >= args.batch_size:
            run_update()
            in_region += 1
    sync()
    region = time.perf_counter() - t0
    owed = max(0.0, args.steps / args.batch_size - in_region)
    elapsed = region + owed * t_update

    print(json.dumps({
        "metric": "env-steps/sec (whole node)",
        "value": round(args.steps / elapsed, 2),
        "unit": "env-steps/s",
        "source": "reference (MIT-REALM/gcbf-pytorch via dependency shim)",
        "n_gpus": 1 if use_cuda else 0,
        "steps": args.steps,
        "ms_per_step": round(elapsed / args.steps * 1000, 3),
        "dtype": "fp32",
        "config": {"env": args.env, "num_agents": args.num_agents,
                   "num_obs": args.obs, "global_batch": args.batch_size},
        "detail": {"update_s": round(t_update, 4),
                   "updates_in_region": in_region,
                   "updates_amortized": round(owed, 4),
                   "rollout_region_s": round(region, 4)},
    }), flush=True)


if __name__ == "__main__":
    main()
