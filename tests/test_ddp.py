"""Data-parallel layer tests over gloo (world_size 2, CPU).

Covers the RCCL-over-xGMI code path structure without a GPU: flat-buffer
gradient all-reduce, initial weight broadcast, and a 2-rank GCBF update
whose gradients must agree with the average of the per-rank gradients.
"""
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _np(t):
    """Queue-safe copy: torch.multiprocessing reduces tensors through
    shared-memory FDs, and a worker exiting right after q.put races the
    parent's mapping (ConnectionReset, seen under load) — numpy arrays
    pickle inline."""
    return t.detach().cpu().numpy().copy()

from gcbf_amd.parallel import GradSynchronizer, broadcast_modules


def _free_port() -> int:
    """OS-assigned free TCP port (PID-derived ports collided across
    sequential pytest runs via TIME_WAIT reuse)."""
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)



def _spawn_collect(worker, world=2, timeout=600, tries=2):
    """Launch `world` spawn processes and collect one queue item each.

    Retries once on OS-level spawn/queue races (FileNotFoundError /
    ConnectionResetError observed only under heavy CPU contention)."""
    last = None
    for _ in range(tries):
        try:
            ctx = mp.get_context("spawn")
            q = ctx.Queue()
            port = _free_port()
            ps = [ctx.Process(target=worker, args=(r, world, q, port))
                  for r in range(world)]
            for p in ps:
                p.start()
            results = {}
            for _ in range(world):
                item = q.get(timeout=timeout)
                results[item[0]] = item[1:]
            for p in ps:
                p.join(timeout=timeout)
                assert p.exitcode == 0
            return results
        except (FileNotFoundError, ConnectionResetError) as e:
            last = e
            for p in ps:
                if p.is_alive():
                    p.terminate()
    raise last


def _worker_grad_sync(rank, world, q, port):
    _init(rank, world, port)
    torch.manual_seed(rank)  # different grads per rank
    m = torch.nn.Linear(4, 3)
    # identical weights
    broadcast_modules([m])
    x = torch.randn(8, 4)
    loss = m(x).pow(2).sum()
    loss.backward()
    local_grad = m.weight.grad.clone()
    sync = GradSynchronizer([m])
    sync()
    q.put((rank, _np(local_grad), _np(m.weight.grad)))
    dist.barrier()
    dist.destroy_process_group()


def test_grad_synchronizer_averages():
    results = _spawn_collect(_worker_grad_sync, timeout=120)
    import numpy as np
    mean = (results[0][0] + results[1][0]) / 2
    assert np.allclose(results[0][1], mean, atol=1e-6)
    assert np.allclose(results[1][1], mean, atol=1e-6)


def _worker_gcbf_dp(rank, world, q, port):
    _init(rank, world, port)
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.trainer.utils import set_seed
    set_seed(100 + rank)  # decorrelated envs
    dev = torch.device("cpu")
    env = make_env("SimpleCar", 4, dev)
    env.train()
    algo = make_algo("gcbf", env, 4, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=20)
    broadcast_modules([algo.cbf, algo.actor])
    algo.grad_sync = GradSynchronizer([algo.cbf, algo.actor])
    data = env.reset()
    for step in range(1, 21):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.5)
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
        if algo.is_update(step):
            algo.update(step, None)
    # after synced updates from identical init, weights must match
    q.put((rank, _np(algo.actor.feat_2_action.net[0].weight)))
    dist.barrier()
    dist.destroy_process_group()


def test_gcbf_dp_two_ranks_stay_in_sync():
    import numpy as np
    results = _spawn_collect(_worker_gcbf_dp)
    # identical optimizer trajectories (same averaged grads every step)
    assert np.allclose(results[0][0], results[1][0], atol=1e-6)


def _worker_bucketed(rank, world, q, port):
    _init(rank, world, port)
    from gcbf_amd.parallel import BucketedGradSynchronizer
    torch.manual_seed(rank)
    m = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.ReLU(),
                            torch.nn.Linear(32, 8), torch.nn.Linear(8, 2))
    broadcast_modules([m])
    # tiny bucket size forces several buckets (layout still rank-invariant)
    sync = BucketedGradSynchronizer([m], bucket_bytes=512)
    sync.install()
    x = torch.randn(4, 16)
    loss = m(x).pow(2).sum()
    loss.backward()
    local = [_np(p.grad) for p in m.parameters()]
    sync()
    q.put((rank, local, [_np(p.grad) for p in m.parameters()]))
    dist.barrier()
    dist.destroy_process_group()


def test_bucketed_grad_synchronizer_averages():
    import numpy as np
    results = _spawn_collect(_worker_bucketed, timeout=120)
    for g0, g1, s0, s1 in zip(results[0][0], results[1][0],
                              results[0][1], results[1][1]):
        mean = (g0 + g1) / 2
        assert np.allclose(s0, mean, atol=1e-6)
        assert np.allclose(s1, mean, atol=1e-6)


class _CollectWriter:
    def __init__(self):
        self.scalars = []

    def add_scalar(self, name, value, step):
        self.scalars.append((name, value, step))


def _worker_gcbf_bucketed_logs(rank, world, q, port):
    _init(rank, world, port)
    from gcbf_amd.algo import make_algo
    from gcbf_amd.env import make_env
    from gcbf_amd.parallel import make_grad_synchronizer
    from gcbf_amd.trainer.utils import set_seed
    set_seed(100 + rank)
    dev = torch.device("cpu")
    env = make_env("SimpleCar", 4, dev)
    env.train()
    algo = make_algo("gcbf", env, 4, env.node_dim, env.edge_dim,
                     env.action_dim, dev, batch_size=20)
    broadcast_modules([algo.cbf, algo.actor])
    algo.grad_sync = make_grad_synchronizer([algo.cbf, algo.actor])
    writer = _CollectWriter()
    data = env.reset()
    for step in range(1, 21):
        data.update(u_ref=env.u_ref(data))
        a = algo.step(data, prob=0.5)
        data, r, done, info = env.step(a)
        if done:
            data = env.reset()
        if algo.is_update(step):
            algo.update(step, writer)
    q.put((rank, _np(algo.actor.feat_2_action.net[0].weight),
           writer.scalars))
    dist.barrier()
    dist.destroy_process_group()


def test_gcbf_dp_bucketed_sync_and_global_logs():
    """Bucketed (overlapped) all-reduce keeps ranks in lockstep AND the
    logged update scalars are all-reduced so every rank logs the same
    global-batch values."""
    import numpy as np
    results = _spawn_collect(_worker_gcbf_bucketed_logs)
    assert np.allclose(results[0][0], results[1][0], atol=1e-6)
    s0, s1 = results[0][1], results[1][1]
    assert len(s0) == len(s1) > 0
    for (n0, v0, t0), (n1, v1, t1) in zip(s0, s1):
        assert n0 == n1 and t0 == t1
        assert abs(v0 - v1) < 1e-6, (n0, v0, v1)


def test_env_world_defaults():
    from gcbf_amd.parallel import env_world
    rank, world, local = env_world()
    assert world >= 1
