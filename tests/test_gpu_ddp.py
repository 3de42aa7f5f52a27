"""2-rank data parallelism on ONE GPU (gpu-marked).

RCCL refuses two ranks on one device ("Duplicate GPU detected", measured
on MI355X), so the single-box DP smoke runs the gloo backend on CUDA
tensors — same wire format, same GradSynchronizer code path, different
transport.  The real RCCL-over-xGMI path is exercised by the driver's
8-GPU scaling run; this test pins broadcast + bucketed all-reduce +
per-rank rollout/update on device.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _free_port() -> int:
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, world, q, port):
    try:
        _worker_body(rank, world, q, port)
    except Exception:
        import traceback
        q.put((rank, "ERROR:" + traceback.format_exc()))
        raise


def _worker_body(rank, world, q, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.parallel import broadcast_modules, make_grad_synchronizer
    from gcbf_amd.trainer.utils import set_seed
    from gcbf_amd.utils.amp import enable_bf16

    set_seed(100 + rank)
    dev = torch.device("cuda", 0)
    env = make_env("DubinsCar", 8, dev)
    env.train()
    algo = make_algo("gcbf", env, 8, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=32)
    broadcast_modules([algo.cbf, algo.actor])
    algo.grad_sync = make_grad_synchronizer([algo.cbf, algo.actor])
    enable_bf16(algo)
    data = env.reset()
    for step in range(1, 33):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.5)
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
        if algo.is_update(step):
            algo.update(step, None)
    w = algo.actor.feat_2_action.net[0].weight.detach().float().cpu()
    # numpy copy: tensor queue reduction uses shared-memory FDs and races
    # worker exit (ConnectionReset)
    q.put((rank, w.numpy().copy()))
    dist.barrier()
    dist.destroy_process_group()


def test_two_rank_dp_on_one_gpu():
    if not torch.cuda.is_available():
        pytest.skip("GPU required")
    last = None
    for attempt in range(2):
        # two processes sharing one GPU occasionally die in ROCm context
        # setup (ConnectionReset on the result queue, no Python
        # traceback); retry once — a real logic failure reproduces
        try:
            results = _launch_pair()
            break
        except (ConnectionResetError, EOFError, OSError) as e:
            last = e
    else:
        raise last
    # averaged grads from identical broadcast weights keep ranks in
    # lockstep (bf16 numerics are deterministic per rank pair here)
    import numpy as np
    assert np.allclose(results[0], results[1], atol=1e-5), \
        np.abs(results[0] - results[1]).max()


def _launch_pair():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    ps = [ctx.Process(target=_worker, args=(r, 2, q, port))
          for r in range(2)]
    for p in ps:
        p.start()
    try:
        results = {}
        for _ in range(2):
            rank, w = q.get(timeout=600)
            assert not (isinstance(w, str) and w.startswith("ERROR:")), w
            results[rank] = w
        for p in ps:
            p.join(timeout=600)
            assert p.exitcode == 0
        return results
    finally:
        for p in ps:
            if p.is_alive():
                p.terminate()
