"""Rollout-only microbenchmark: captured engine step cost, split by path.

    python tools/rollout_bench.py [--env DubinsCar] [-n 16] [--obs 0]

Times engine.step for forced-policy, forced-explore and mixed (prob 0.5)
streams, without updates — isolates rollout regressions from update noise
in the full bench.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from gcbf_amd.algo import make_algo
from gcbf_amd.env import make_env
from gcbf_amd.rollout import RolloutEngine
from gcbf_amd.trainer.utils import set_seed
from gcbf_amd.utils.amp import enable_bf16


def run_steps(eng, prob, n):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        if eng.step(prob=prob):
            eng.reload()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--env", type=str, default="DubinsCar")
    p.add_argument("-n", "--num-agents", type=int, default=16)
    p.add_argument("--obs", type=int, default=0)
    p.add_argument("--steps", type=int, default=1500)
    args = p.parse_args()

    set_seed(0)
    dev = torch.device("cuda")
    e0 = make_env(args.env, args.num_agents, dev)
    params = e0.default_params
    params["num_obs"] = args.obs
    env = make_env(args.env, args.num_agents, dev, params=params)
    env.train()
    algo = make_algo("gcbf", env, args.num_agents, env.node_dim,
                     env.edge_dim, env.action_dim, dev, batch_size=512)
    enable_bf16(algo)
    env.reset()
    eng = RolloutEngine(env, algo)
    for _ in range(300):                      # warmup
        if eng.step(prob=0.5):
            eng.reload()
    t_pol = run_steps(eng, 0.0, args.steps)   # always policy graph
    t_exp = run_steps(eng, 1.0, args.steps)   # always explore graph
    t_mix = run_steps(eng, 0.5, args.steps)
    print(f"rollout ms/step: policy {t_pol:.4f}  explore {t_exp:.4f}  "
          f"mixed(0.5) {t_mix:.4f}   [{args.env} n={args.num_agents} "
          f"obs={args.obs}]")


if __name__ == "__main__":
    main()
