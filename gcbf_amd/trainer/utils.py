"""Trainer utilities: seeding, log-dir init, settings I/O, episode eval,
CBF contour plots (reference gcbf/trainer/utils.py)."""
from __future__ import annotations

import copy
import datetime
import os
import random
from typing import Callable, Optional, Tuple

import numpy as np
import torch
import yaml

from ..env.base import MultiAgentEnv
from ..graph import GraphBatch


def set_seed(seed: int):
    # reference gcbf/trainer/utils.py:20-25
    os.environ["PYTHONHASHSEED"] = str(seed)
    torch.manual_seed(seed)
    np.random.seed(seed)
    random.seed(seed)
    torch.cuda.manual_seed_all(seed)


def init_logger(log_path: str, env_name: str, algo_name: str, seed: int,
                args: Optional[dict] = None,
                hyper_params: Optional[dict] = None) -> str:
    """Create logs/<env>/<algo>/seed<seed>_<timestamp>/ and write
    settings.yaml (reference gcbf/trainer/utils.py:28-105)."""
    start_time = datetime.datetime.now().strftime("%Y%m%d%H%M%S")
    log_dir = os.path.join(log_path, env_name, algo_name,
                           f"seed{seed}_{start_time}")
    os.makedirs(log_dir, exist_ok=True)

    with open(os.path.join(log_dir, "settings.yaml"), "w") as log:
        if args is not None:
            for key in args.keys():
                log.write(f"{key}: {args[key]}\n")
            if "algo" not in args.keys():
                log.write(f"algo: {algo_name}\n")
        if hyper_params is not None:
            log.write("hyper_params:\n")
            for key1, val1 in hyper_params.items():
                if isinstance(val1, dict):
                    log.write(f"  {key1}: \n")
                    for key2, val2 in val1.items():
                        log.write(f"    {key2}: {val2}\n")
                else:
                    log.write(f"  {key1}: {val1}\n")
        else:
            log.write("hyper_params: using default hyper-parameters")
    return log_dir


def read_settings(path: str) -> dict:
    with open(os.path.join(path, "settings.yaml")) as f:
        return yaml.load(f, Loader=yaml.FullLoader)


def read_params(env: str, algo: str) -> Optional[dict]:
    """Pre-defined training hyper-parameters per (env, algo)
    (reference gcbf/trainer/utils.py:317-340)."""
    path = os.path.join(os.path.dirname(__file__), "hyperparams.yaml")
    with open(path) as f:
        params = yaml.safe_load(f)
    if env in params and algo in params[env]:
        return params[env][algo]
    return None


def eval_ctrl_epi(controller: Callable, env: MultiAgentEnv, seed: int = 0,
                  make_video: bool = True, plot_edge: bool = True,
                  verbose: bool = True) -> Tuple[float, float, tuple, dict]:
    """Evaluate a controller for one episode
    (reference gcbf/trainer/utils.py:127-223)."""
    set_seed(seed)
    epi_length = 0.0
    epi_reward = 0.0
    video = []
    data = env.reset()
    reach = torch.zeros(env.num_agents).bool()
    safe_agent = torch.ones(env.num_agents).bool()
    success_agent = torch.zeros(env.num_agents).bool()
    safe_data = []
    states = []
    while True:
        data.update(u_ref=env.u_ref(data))
        action = controller(data)
        if data.agent_mask is not None:
            states.append(data.states[data.agent_mask].unsqueeze(0).cpu())
        else:
            states.append(data.states.unsqueeze(0).cpu())
        next_data, reward, done, info = env.step(action)
        epi_length += 1
        epi_reward += float(reward.float().mean())
        if "collision" in info:
            safe_agent[info["collision"].cpu()] = False
            safe = torch.ones(env.num_agents).bool()
            safe[info["collision"].cpu()] = False
            safe_data.append(safe.unsqueeze(0))
        if "reach" in info:
            reach = info["reach"].cpu()
        if make_video:
            video.append(env.render(plot_edge=plot_edge))
        data = next_data
        if done:
            if "reach" in info:
                success_agent = torch.logical_and(reach, safe_agent)
            if verbose:
                msg = (f"n: {env.num_agents}, reward: {epi_reward:.2f}, "
                       f"length: {epi_length}")
                if "collision" in info:
                    msg += (f", safe: "
                            f"{safe_agent.sum().item() / env.num_agents:.2f}")
                    sd = torch.cat(safe_data, dim=0).numpy()
                    msg += f", safe state: {sd.mean():.2f}"
                if "reach" in info:
                    msg += (f", reach: "
                            f"{reach.sum().item() / env.num_agents:.2f}")
                    msg += (f", success: "
                            f"{success_agent.sum().item() / env.num_agents:.2f}")
                print(msg)
            break

    states = torch.cat(states, dim=0)
    return epi_reward, epi_length, tuple(video), {
        "safe": safe_agent.sum().item() / env.num_agents,
        "reach": reach.sum().item() / env.num_agents,
        "success": success_agent.sum().item() / env.num_agents,
        "states": states}


def plot_cbf_contour(cbf_fun, data: GraphBatch, env: MultiAgentEnv,
                     agent_id: int, x_dim: int, y_dim: int,
                     attention: bool = True):
    """Contour of the learned CBF around one agent
    (reference gcbf/trainer/utils.py:226-298)."""
    import matplotlib.pyplot as plt
    n_mesh = 30
    low_lim, high_lim = env.state_lim
    x, y = np.meshgrid(
        np.linspace(low_lim[x_dim].cpu(), high_lim[x_dim].cpu(), n_mesh),
        np.linspace(low_lim[y_dim].cpu(), high_lim[y_dim].cpu(), n_mesh))
    plot_data = []
    for i in range(n_mesh):
        for j in range(n_mesh):
            state = copy.deepcopy(data.states)
            state[agent_id, x_dim] = float(x[i, j])
            state[agent_id, y_dim] = float(y[i, j])
            plot_data.append(GraphBatch(
                x=data.x, pos=state[:, :2], states=state,
                edge_index=data.edge_index,
                edge_attr=env.edge_attr(state, data.edge_index),
                agent_mask=data.agent_mask))
    batch = GraphBatch.from_list(plot_data)
    cbf = cbf_fun(batch).view(n_mesh, n_mesh, env.num_agents)[
        :, :, agent_id].detach().cpu()
    fig, ax = plt.subplots(1, 1, figsize=(12, 10), dpi=100)
    plt.contourf(x, y, cbf, cmap="magma", levels=15, alpha=0.5, linewidths=3)
    plt.colorbar()
    plt.contour(x, y, cbf, levels=[0.0], colors="blue", linewidths=6)
    ax = env.render(return_ax=True, ax=ax)
    if attention and hasattr(cbf_fun, "__self__"):
        ax = plot_attention(ax, cbf_fun.__self__.attention, data, agent_id)
    plt.tight_layout()
    plt.xlabel(f"dim: {x_dim}")
    plt.ylabel(f"dim: {y_dim}")
    return ax


def plot_attention(ax, attention_fun: Callable, data: GraphBatch,
                   agent_id: int):
    # reference gcbf/trainer/utils.py:301-314
    attention = attention_fun(data).cpu().detach().numpy()
    pos = data.pos.cpu().detach().numpy()
    edge_index = data.edge_index.cpu().detach().numpy()
    edge_centers = (pos[edge_index[0], :] + pos[edge_index[1], :]) / 2
    ax.scatter(pos[agent_id, 0], pos[agent_id, 1], s=100, c="black",
               marker="d", alpha=1)
    for i, text_point in enumerate(edge_centers):
        if edge_index[1, i] == agent_id:
            ax.text(text_point[0], text_point[1], f"{attention[i, 0]:.2f}",
                    size=18, color="black", weight="bold",
                    horizontalalignment="center",
                    verticalalignment="center", clip_on=True)
    return ax
