"""Training CLI (same flag surface as the reference train.py:75-98).

Single GPU:   python train.py --env DubinsCar -n 16 --steps 500000
Multi-GPU DP: python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
                  --master-addr 127.0.0.1 train.py --env DubinsCar -n 16 \
                  --steps 500000
Each rank owns an env replica + buffers; gradients are all-reduced over
RCCL/xGMI every inner iteration.
"""
import argparse
import os

import torch

from gcbf_amd.algo import make_algo
from gcbf_amd.env import make_env
from gcbf_amd.parallel import (broadcast_modules,
                               cleanup_distributed, init_distributed,
                               make_grad_synchronizer)
from gcbf_amd.trainer import Trainer
from gcbf_amd.trainer.utils import init_logger, read_params, set_seed


def train(args):
    rank, world_size, local_rank = init_distributed()
    # per-rank seed offset: decorrelated env replicas, identical schedules
    set_seed(args.seed + rank)

    use_cuda = torch.cuda.is_available() and not args.cpu
    if use_cuda:
        dev_idx = (local_rank % torch.cuda.device_count()
                   if world_size > 1 else args.gpu)
        device = torch.device("cuda", dev_idx)
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    if rank == 0:
        print(f"> Training with {device} (world size {world_size})")

    params = make_env(args.env, args.num_agents, device).default_params
    if args.area_size is not None:
        params["area_size"] = args.area_size
    if args.obs is not None:
        params["num_obs"] = args.obs
    max_neighbors = 12 if args.algo == "macbf" else None
    env = make_env(args.env, args.num_agents, device, params=params,
                   max_neighbors=max_neighbors)
    env.train()
    env_test = make_env(args.env, args.num_agents, device, params=params,
                        max_neighbors=max_neighbors)
    env_test.train()

    hyper = read_params(args.env, args.algo)
    if hyper is None or args.cus:
        hyper = {
            "alpha": 1.0,
            "eps": 0.02,
            "inner_iter": 10,
            "loss_action_coef": 0.001 if args.action_coef is None
            else args.action_coef,
            "loss_unsafe_coef": 1.0,
            "loss_safe_coef": 1.0,
            "loss_h_dot_coef": 0.2 if args.h_dot_coef is None
            else args.h_dot_coef,
        }
        if rank == 0:
            print("> Using custom hyper-parameters")
    elif rank == 0:
        print("> Using pre-defined hyper-parameters")

    if args.resume is not None:
        log_path = args.resume
    elif rank == 0:
        log_path = init_logger(args.log_path, args.env, args.algo, args.seed,
                               vars(args), hyper_params=hyper)
    else:
        log_path = os.path.join(args.log_path, args.env, args.algo,
                                f"rank{rank}")

    algo = make_algo(args.algo, env, args.num_agents, env.node_dim,
                     env.edge_dim, env.action_dim, device, args.batch_size,
                     hyperparams=hyper)

    dtype = args.dtype or ("bf16" if use_cuda else "fp32")
    if dtype == "bf16" and use_cuda and args.algo in ("gcbf", "macbf"):
        from gcbf_amd.utils.amp import enable_bf16
        enable_bf16(algo)
        if rank == 0:
            print("> bf16 compute enabled (MFMA fused linears + autocast)")

    if world_size > 1:
        broadcast_modules([algo.cbf, algo.actor])
        algo.grad_sync = make_grad_synchronizer([algo.cbf, algo.actor])

    trainer = Trainer(env=env, env_test=env_test, algo=algo, log_dir=log_path,
                      rank=rank, world_size=world_size)
    start_step = 1
    if args.resume is not None:
        model_dir = os.path.join(log_path, "models")
        ckpts = sorted(int(d.split("_")[1]) for d in os.listdir(model_dir)
                       if d.startswith("step_"))
        if ckpts:
            algo.load(os.path.join(model_dir, f"step_{ckpts[-1]}"))
        start_step = trainer.load_trainer_state()
        if rank == 0:
            print(f"> Resuming from step {start_step}")
    trainer.train(args.steps, eval_interval=max(args.steps // 10, 1),
                  eval_epi=args.eval_epi, start_step=start_step)
    cleanup_distributed()


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    # required (reference train.py:78-81)
    parser.add_argument("--env", type=str, required=True)
    parser.add_argument("-n", "--num-agents", type=int, required=True)
    parser.add_argument("--steps", type=int, required=True)
    # custom
    parser.add_argument("--area-size", type=float, default=None)
    parser.add_argument("--obs", type=int, default=0)
    parser.add_argument("--algo", type=str, default="gcbf")
    parser.add_argument("--gpu", type=int, default=0)
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--cus", action="store_true", default=False)
    parser.add_argument("--h-dot-coef", type=float, default=None)
    parser.add_argument("--action-coef", type=float, default=None)
    # default
    parser.add_argument("--cpu", action="store_true", default=False)
    parser.add_argument("--log-path", type=str, default="./logs")
    parser.add_argument("--batch-size", type=int, default=512)
    # additions over the reference
    parser.add_argument("--dtype", type=str, default=None,
                        choices=[None, "bf16", "fp32"])
    parser.add_argument("--resume", type=str, default=None,
                        help="path of a previous run's log dir to resume")
    parser.add_argument("--eval-epi", type=int, default=3)
    train(parser.parse_args())
