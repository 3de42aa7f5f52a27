"""Evaluation CLI (same flag surface as the reference test.py:181-204).

Runs N evaluation episodes with the refined controller, reports
safe/reach/success rates, optionally writes video (mp4 via OpenCV when
available, else animated GIF via PIL) and .mat trajectories.
"""
import argparse
import os
import time

import numpy as np
import torch

from gcbf_amd.algo import make_algo
from gcbf_amd.env import make_env
from gcbf_amd.trainer.utils import eval_ctrl_epi, read_settings, set_seed


def _write_video(frames, path_mp4: str):
    """mp4 via cv2 if available, else GIF via PIL, else frame archive."""
    try:
        import cv2
        out = cv2.VideoWriter(path_mp4, cv2.VideoWriter_fourcc(*"mp4v"), 25,
                              (frames[-1].shape[1], frames[-1].shape[0]))
        for fig in frames:
            out.write(np.uint8(fig[:, :, ::-1]))
        out.release()
        return path_mp4
    except ImportError:
        pass
    try:
        from PIL import Image
        gif_path = path_mp4.replace(".mp4", ".gif")
        imgs = [Image.fromarray(np.uint8(f)) for f in frames[::2]]
        imgs[0].save(gif_path, save_all=True, append_images=imgs[1:],
                     duration=80, loop=0)
        return gif_path
    except ImportError:
        npz_path = path_mp4.replace(".mp4", ".npz")
        np.savez_compressed(npz_path, frames=np.stack(frames))
        return npz_path


def test(args):
    set_seed(args.seed)
    use_cuda = torch.cuda.is_available() and not args.cpu
    if use_cuda:
        os.environ.setdefault("CUDA_VISIBLE_DEVICES", str(args.gpu))
    device = torch.device("cuda" if use_cuda else "cpu")

    try:
        settings = read_settings(args.path)
    except TypeError:
        # nominal-controller runs need no checkpoint dir (reference
        # test.py:29-30); fall back to CLI values with sane defaults
        settings = {"algo": "nominal",
                    "num_agents": args.num_agents or 16}

    env_name = settings.get("env") if args.env is None else args.env
    num_agents = settings["num_agents"] if args.num_agents is None \
        else args.num_agents
    max_neighbors = 12 if settings["algo"] == "macbf" else None

    params = make_env(env_name, num_agents, device).default_params
    if args.area_size is not None:
        params["area_size"] = args.area_size
    if args.obs is not None:
        params["num_obs"] = args.obs
    if args.sense_radius is not None:
        params["comm_radius"] = args.sense_radius
    env = make_env(env_name, num_agents, device, params=params,
                   max_neighbors=max_neighbors)
    if args.demo is None:
        env.test()
    else:
        env.demo(args.demo)

    algo = make_algo(settings["algo"], env, num_agents, env.node_dim,
                     env.edge_dim, env.action_dim, device,
                     hyperparams=settings.get("hyper_params"))
    dtype = args.dtype or ("bf16" if use_cuda else "fp32")
    if dtype == "bf16" and use_cuda and settings["algo"] in ("gcbf", "macbf"):
        from gcbf_amd.utils.amp import enable_bf16
        enable_bf16(algo)

    if args.path is None:
        assert args.env is not None and args.num_agents is not None
        args.path = f"./logs/{args.env}"
        os.makedirs(os.path.join(args.path, "nominal"), exist_ok=True)
        video_path = os.path.join(args.path, "nominal", "videos")
    else:
        model_path = os.path.join(args.path, "models")
        if args.iter is not None:
            algo.load(os.path.join(model_path, f"step_{args.iter}"))
        else:
            names = [i for i in os.listdir(model_path) if "step" in i]
            steps = sorted(int(i.split("step_")[1].split(".")[0])
                           for i in names)
            algo.load(os.path.join(model_path, f"step_{steps[-1]}"))
        video_path = os.path.join(args.path, "videos")

    if not args.no_video:
        os.makedirs(video_path, exist_ok=True)

    def apply_fn(data):
        return algo.apply(data, rand=args.rand)

    start_time = time.time()
    results = []
    for i in range(args.epi):
        print(f"epi: {i}")
        results.append(eval_ctrl_epi(apply_fn, env,
                                     np.random.randint(100000),
                                     not args.no_video,
                                     plot_edge=not args.no_edge))
    rewards, lengths, video, info = zip(*results)
    video = sum(video, ())

    safe_rates, reach_rates, success_rates = [], [], []
    n_traj = 0
    for i in info:
        if "safe" in i:
            safe_rates.append(float(i["safe"]))
            n_traj += 1
        if "reach" in i:
            reach_rates.append(float(i["reach"]))
        if "success" in i:
            success_rates.append(float(i["success"]))

    if args.write_traj == "mat":
        from scipy.io import savemat
        os.makedirs(os.path.join(args.path, "trajs"), exist_ok=True)
        for i, i_info in enumerate(info):
            savemat(os.path.join(
                args.path, "trajs",
                f"demo{args.demo}_seed{args.seed}_agent{env.num_agents}"
                f"_size_{args.area_size}_safe{np.mean(safe_rates)}"
                f"_reach{np.mean(reach_rates)}"
                f"_success{np.mean(success_rates)}"
                f"_reward{np.mean(rewards):.2f}_traj{i}.mat"),
                {"states": i_info["states"].cpu().numpy()})

    if not args.no_video and video:
        print("> Making video...")
        out_path = _write_video(list(video), os.path.join(
            video_path,
            f"demo{args.demo}_seed{args.seed}_agent{env.num_agents}"
            f"_size_{args.area_size}_safe{np.mean(safe_rates)}"
            f"_reach{np.mean(reach_rates)}_success{np.mean(success_rates)}"
            f"_reward{np.mean(rewards):.2f}.mp4"))
        print(f"> Video written to {out_path}")

    verbose = (f"average reward: {np.mean(rewards):.2f}, "
               f"average length: {np.mean(lengths):.2f}")
    if n_traj > 0:
        verbose += (f", safe rate: {np.mean(safe_rates)} +/- "
                    f"{np.std(safe_rates)}, reach rate: "
                    f"{np.mean(reach_rates)} +/- {np.std(reach_rates)}, "
                    f"success rate: {np.mean(success_rates)} +/- "
                    f"{np.std(success_rates)}")
    print(verbose)
    with open(os.path.join(args.path, "test_log.csv"), "a") as f:
        f.write(f"{env.num_agents},{args.obs},{args.epi},{args.area_size},"
                f"{np.mean(safe_rates)},{np.std(safe_rates)},"
                f"{np.mean(reach_rates)},{np.std(reach_rates)},"
                f"{np.mean(success_rates)},{np.std(success_rates)}\n")
    print(f"> Done in {time.time() - start_time:.0f}s")


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--path", type=str, default=None)
    parser.add_argument("--obs", type=int, default=None)
    parser.add_argument("--sense-radius", type=float, default=None)
    parser.add_argument("--area-size", type=float, default=None)
    parser.add_argument("-n", "--num-agents", type=int, default=None)
    parser.add_argument("--demo", type=int, default=None)
    parser.add_argument("--env", type=str, default=None)
    parser.add_argument("--iter", type=int, default=None)
    parser.add_argument("--epi", type=int, default=5)
    parser.add_argument("--no-video", action="store_true", default=False)
    parser.add_argument("--gpu", type=int, default=0)
    parser.add_argument("--no-edge", action="store_true", default=False)
    parser.add_argument("--write_traj", type=str, default=None)
    parser.add_argument("--rand", type=float, default=30)
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--cpu", action="store_true", default=False)
    parser.add_argument("--dtype", type=str, default=None,
                        choices=[None, "bf16", "fp32"])
    test(parser.parse_args())
