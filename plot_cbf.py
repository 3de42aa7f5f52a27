"""CBF contour plotting CLI (reference plot_cbf.py:16-128).

Loads a trained GCBF checkpoint, simulates episodes, and writes the CBF
contour + attention plot for one agent at each step to
``<path>/figs/agent_<agent>/epi_<i>/<t>.pdf``.
"""
import argparse
import os
import shutil

import numpy as np
import torch

from gcbf_amd.algo import make_algo
from gcbf_amd.env import make_env
from gcbf_amd.trainer.utils import (plot_cbf_contour, read_settings,
                                    set_seed)


def plot_cbf(args):
    set_seed(args.seed)
    use_cuda = torch.cuda.is_available() and not args.cpu
    if use_cuda:
        os.environ.setdefault("CUDA_VISIBLE_DEVICES", str(args.gpu))
    device = torch.device("cuda" if use_cuda else "cpu")

    try:
        settings = read_settings(args.path)
    except TypeError:
        raise TypeError("Cannot find configuration file in the path")

    env_name = settings["env"] if args.env is None else args.env
    num_agents = settings["num_agents"] if args.num_agents is None \
        else args.num_agents
    params = make_env(env_name, num_agents, device).default_params
    params["area_size"] = args.area_size
    params["num_obs"] = args.obs
    env = make_env(env_name, num_agents, device, params=params,
                   max_neighbors=12 if settings["algo"] == "macbf" else None)
    env.test()

    algo = make_algo(settings["algo"], env, num_agents, env.node_dim,
                     env.edge_dim, env.action_dim, device,
                     hyperparams=settings.get("hyper_params"))
    model_path = os.path.join(args.path, "models")
    if args.iter is not None:
        algo.load(os.path.join(model_path, f"step_{args.iter}"))
    else:
        names = [i for i in os.listdir(model_path) if "step" in i]
        steps = sorted(int(i.split("step_")[1].split(".")[0]) for i in names)
        algo.load(os.path.join(model_path, f"step_{steps[-1]}"))

    if not hasattr(algo, "cbf"):
        raise KeyError("The algorithm must has a CBF function")

    fig_root = os.path.join(args.path, "figs")
    os.makedirs(fig_root, exist_ok=True)
    fig_path = os.path.join(fig_root, f"agent_{args.agent}")
    if os.path.exists(fig_path):
        shutil.rmtree(fig_path)
    os.makedirs(fig_path)

    import matplotlib.pyplot as plt
    for i_epi in range(args.epi):
        set_seed(np.random.randint(100000))
        data = env.reset()
        epi_dir = os.path.join(fig_path, f"epi_{i_epi}")
        os.makedirs(epi_dir)
        t = 0
        while True:
            data.update(u_ref=env.u_ref(data))
            action = algo.apply(data)
            ax = plot_cbf_contour(algo.cbf, data, env, args.agent,
                                  args.x_dim, args.y_dim, attention=True)
            plt.savefig(os.path.join(epi_dir, f"{t}.pdf"))
            plt.close("all")
            data, reward, done, _ = env.step(action)
            t += 1
            if done or (args.max_steps and t >= args.max_steps):
                break
    print(f"> Figures saved under {fig_path}")


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    # custom (reference plot_cbf.py:107-123)
    parser.add_argument("--obs", type=int, default=0)
    parser.add_argument("--area-size", type=float, required=True)
    parser.add_argument("-n", "--num-agents", type=int, default=None)
    parser.add_argument("--path", type=str, default=None)
    parser.add_argument("--env", type=str, default=None)
    parser.add_argument("--iter", type=int, default=None)
    parser.add_argument("--epi", type=int, default=5)
    parser.add_argument("--agent", type=int, default=0)
    parser.add_argument("--x-dim", type=int, default=0)
    parser.add_argument("--y-dim", type=int, default=1)
    parser.add_argument("--gpu", type=int, default=0)
    # default
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--cpu", action="store_true", default=False)
    # addition: cap steps per episode (0 = episode end)
    parser.add_argument("--max-steps", type=int, default=0)
    plot_cbf(parser.parse_args())
