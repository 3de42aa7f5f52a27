"""Train the UNMODIFIED reference through the shim, without mid-training
eval episodes (the reference train.py hardcodes eval_epi=3, which
dominates short CPU baseline runs; Trainer.train(eval_epi=0) is the
reference's own no-eval path — checkpoints still save every
eval_interval).  Build sequence mirrors /root/reference/train.py:11-72.

    python tools/ref_baseline/train_ref.py --env SimpleCar -n 4 \
        --steps 2000 --log-path results/refruns
"""
import argparse
import os
import sys
import time

HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    sys.path.insert(0, HERE)
    import pygshim
    pygshim.install()
    sys.path.insert(0, "/root/reference")

    import torch
    from gcbf.algo import make_algo
    from gcbf.env import make_env
    from gcbf.trainer import Trainer
    from gcbf.trainer.utils import init_logger, read_params, set_seed

    p = argparse.ArgumentParser()
    p.add_argument("--env", type=str, required=True)
    p.add_argument("-n", "--num-agents", type=int, required=True)
    p.add_argument("--steps", type=int, default=2000)
    p.add_argument("--batch-size", type=int, default=512)
    p.add_argument("--algo", type=str, default="gcbf")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--obs", type=int, default=None)
    p.add_argument("--area-size", type=float, default=None)
    p.add_argument("--log-path", type=str, default="results/refruns")
    args = p.parse_args()

    set_seed(args.seed)
    device = torch.device("cpu")

    env = make_env(args.env, args.num_agents, device)
    params = env.default_params
    if args.area_size is not None:
        params["area_size"] = args.area_size
    if args.obs is not None:
        params["num_obs"] = args.obs
    mn = 12 if args.algo == "macbf" else None
    env = make_env(args.env, args.num_agents, device, params=params,
                   max_neighbors=mn)
    env.train()
    env_test = make_env(args.env, args.num_agents, device, params=params,
                        max_neighbors=mn)
    env_test.train()

    hyper = read_params(args.env, args.algo)
    log_path = init_logger(args.log_path, args.env, args.algo, args.seed,
                           vars(args), hyper_params=hyper)
    algo = make_algo(args.algo, env, args.num_agents, env.node_dim,
                     env.edge_dim, env.action_dim, device, args.batch_size,
                     hyperparams=hyper)
    trainer = Trainer(env, env_test, algo, log_path)
    t0 = time.time()
    trainer.train(args.steps, eval_interval=max(args.steps // 10, 1),
                  eval_epi=0)
    wall = time.time() - t0
    print(f"REF_TRAIN_DONE wall_s={wall:.1f} "
          f"env_steps_per_s={args.steps / wall:.2f} log={log_path}",
          flush=True)


if __name__ == "__main__":
    main()
