"""hipGraph-captured rollout engine.

The rollout loop is launch/Python-bound on GPU (~25 kernels for the actor
GNN + graph rebuild + env step per env step).  This engine captures ONE
whole training rollout step into a hipGraph (torch.cuda.CUDAGraph == hipGraph
on ROCm) and replays it per step:

    actor forward (over padded, fixed-capacity edge buffers)
    × exploration gate (host-written device scalar)
    → fused env-step kernel (dynamics + rewards + next u_ref)
    → capture-safe padded graph rebuild (fixed E_max buffers)
    → flag publication ([edge_count, reach_all, unsafe_any] in one int32
      tensor; the host reads it once per step)

Static shapes come from padding: edge buffers hold E_max = n·(N−1) entries,
pad entries carry src=dst=0 (valid gather index) and a segment sentinel
id == num_nodes so the CSR attention aggregation skips them; the GEMMs run
over E_max rows always (tiny at rollout sizes).  Exploration, episode
termination and replay-buffer appends stay on the host.

Weights are read by address, so Adam steps between replays take effect
without re-capture.
"""
from __future__ import annotations



import numpy as np
import torch

from . import ops
from .graph import GraphBatch


def engine_supported(env, algo) -> bool:
    from .algo.gcbf import GCBF
    from .algo.macbf import MACBF
    if not torch.cuda.is_available() or not ops.hip_available():
        return False
    if not isinstance(algo, GCBF) or isinstance(algo, MACBF):
        return False
    if env._mode != "train" or env._max_neighbors is not None:
        return False
    data = env.data if env.data is not None else env.reset()
    n, N = env.num_agents, data.num_nodes
    # sanity bound only: the engine uses a soft edge capacity with an eager
    # fallback, so even large scenes (the n=256 stress config) capture
    return n * (N - 1) <= 262144


class RolloutEngine:

    def __init__(self, env, algo, edge_capacity: int | None = None):
        self.env = env
        self.algo = algo
        self.device = env.device
        self.n = env.num_agents
        self.pos_dim = 3 if env.state_dim == 6 else 2
        self.edge_dim = env.edge_dim

        data = env.data if env.data is not None else env.reset()
        self.N = data.num_nodes
        # soft edge capacity: full n*(N-1) for small scenes; for large ones
        # 8x the observed density (overflowing steps fall back to the eager
        # path until the edge count fits again)
        full = self.n * (self.N - 1)
        if edge_capacity is not None:  # test hook
            self.E_max = min(full, edge_capacity)
        else:
            self.E_max = min(full, max(4096, 8 * max(data.num_edges, 1)))
        self._eager = False

        from gcbf_amd import _C
        self._ext = _C

        kind_by_class = {"SimpleCar": "car", "DubinsCar": "dubins",
                         "SimpleDrone": "drone"}
        self.kind = kind_by_class[type(env).__name__]

        dev = self.device
        # ping-pong buffer PAIRS: each captured replay reads set i and the
        # kernels write set 1-i directly (out= bindings) — no copy-back
        # inside the graph (the copies were ~25% of the replay, rocprof r02)
        self.states = [data.states.clone().contiguous(),
                       data.states.clone().contiguous()]
        self.x = data.x.clone()
        self.agent_mask = (None if data.agent_mask is None
                           else data.agent_mask.clone())
        self.goal = env._goal.clone().contiguous()
        u0 = env.u_ref(data).clone().contiguous()
        self.u_ref = [u0, u0.clone()]
        self.ei = [torch.zeros(2, self.E_max, dtype=torch.long, device=dev)
                   for _ in range(2)]
        self.seg = [torch.full((self.E_max,), self.N, dtype=torch.long,
                               device=dev) for _ in range(2)]
        self.ea = [torch.zeros(self.E_max, self.edge_dim, device=dev)
                   for _ in range(2)]
        self.cur = 0
        self.zero_action = torch.zeros(self.n, env.action_dim, device=dev)
        self.flags = torch.zeros(3, dtype=torch.int32, device=dev)
        # persistent step outputs (pure outputs, not ping-ponged)
        self.reward_buf = torch.zeros(self.n, device=dev)
        self.reach_buf = torch.zeros(self.n, dtype=torch.bool, device=dev)
        self.coll_buf = torch.zeros(self.n, dtype=torch.bool, device=dev)
        self.ecount_buf = torch.zeros(1, dtype=torch.int32, device=dev)
        self.E = 0

        if self.kind in ("car", "drone"):
            self.K = env._get_K_tensor().clone().contiguous()

        self._load_graph_from_env()
        self._capture()
        # attach the ring store up front so per-step snapshots can be
        # metadata-only from the first step (GCBF builds it lazily at the
        # first update otherwise)
        if getattr(algo, "_ring", None) is None \
                and hasattr(algo, "_make_ring") \
                and not getattr(algo, "_ring_tried", True):
            algo._make_ring()

    # ----------------------------------------------------------------- body
    def _step_args(self, action, i: int):
        p = self.env.params
        r_key = "drone_radius" if self.kind == "drone" else "car_radius"
        act_lim = {"car": 10.0, "dubins": 2.0, "drone": 10.0}[self.kind]
        base = [self.states[i], self.goal, action]
        if self.kind in ("car", "drone"):
            base.append(self.K)
        return base + [self.env.dt, p[r_key], p["speed_limit"],
                       p["dist2goal"], act_lim]

    def _body(self, explore: bool, i: int):
        o = 1 - i
        data = GraphBatch(x=self.x, pos=self.states[i][:, :self.pos_dim],
                          states=self.states[i], edge_index=self.ei[i],
                          edge_attr=self.ea[i], agent_mask=self.agent_mask,
                          u_ref=self.u_ref[i])
        data.seg_dst = self.seg[i]
        data.agents_first_n = self.n if self.agent_mask is not None else None
        with torch.no_grad():
            if explore:
                # exploration step: the action is zeroed (pure nominal
                # control, reference gcbf/algo/gcbf.py:131-132) — the
                # reference still runs the actor and discards the output;
                # this graph skips it entirely (observably identical)
                action = self.zero_action
            else:
                action = self.algo.actor(data)
            unsafe_any = self.env.unsafe_mask(data).any()
            # kernels write set o directly (out= buffers), so the replay
            # has no copy-back; ecount lands straight in flags[0]
            ops.env_step_fused(
                self.kind, *self._step_args(action, i),
                out=[self.states[o], self.u_ref[o], self.reward_buf,
                     self.reach_buf, self.coll_buf])
            self._ext.build_graph_padded(
                self.states[o][:, :self.pos_dim].contiguous(),
                self.states[o], 1,
                self.n if self.agent_mask is not None else self.N,
                self.env.params["comm_radius"], -1,
                self.env._attr_kind, self.edge_dim, self.E_max,
                out=[self.ei[o], self.seg[o], self.ea[o], self.flags[0:1]])
            self.flags[1].copy_(self.reach_buf.all().to(torch.int32))
            self.flags[2].copy_(unsafe_any.to(torch.int32))

    def _capture(self):
        saved = (self.states[0].clone(), self.u_ref[0].clone(),
                 self.ei[0].clone(), self.seg[0].clone(),
                 self.ea[0].clone())
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for k in range(4):
                self._body(False, k % 2)
            self._body(True, 0)
            self._body(True, 1)
        torch.cuda.current_stream().wait_stream(s)
        # four graphs over the paired buffers: policy/explore × phase;
        # exploration graphs skip the actor entirely
        self.g = {}
        pool = None
        for explore in (False, True):
            for i in (0, 1):
                g = torch.cuda.CUDAGraph()
                if pool is None:
                    with torch.cuda.graph(g):
                        self._body(explore, i)
                    pool = g.pool()
                else:
                    with torch.cuda.graph(g, pool=pool):
                        self._body(explore, i)
                self.g[(explore, i)] = g
        # restore pre-warmup state into the current (phase-0) set
        self.cur = 0
        self.states[0].copy_(saved[0])
        self.u_ref[0].copy_(saved[1])
        self.ei[0].copy_(saved[2])
        self.seg[0].copy_(saved[3])
        self.ea[0].copy_(saved[4])

    # ---------------------------------------------------------- env plumbing
    def _load_graph_from_env(self):
        """Refresh the CURRENT buffer set from the env's graph (after
        reset)."""
        data = self.env.data
        c = self.cur
        self.states[c].copy_(data.states)
        self.goal.copy_(self.env._goal)
        self.u_ref[c].copy_(self.env.u_ref(data))
        ei, seg, ea, ecount = self._ext.build_graph_padded(
            data.pos.contiguous(), data.states.contiguous(), 1,
            self.n if self.agent_mask is not None else self.N,
            self.env.params["comm_radius"], -1,
            self.env._attr_kind, self.edge_dim, self.E_max)
        self.ei[c].copy_(ei)
        self.seg[c].copy_(seg)
        self.ea[c].copy_(ea)
        self.E = int(ecount.item())
        self._eager = self.E > self.E_max
        if self._eager:
            self.E = 0

    def reload(self):
        self.env.reset()
        self._load_graph_from_env()

    # ----------------------------------------------------- overflow fallback
    def _sync_env_from_buffers(self):
        """Materialize the env's graph from the engine state (exact edge
        build) when entering eager fallback."""
        c = self.cur
        data = GraphBatch(x=self.x,
                          pos=self.states[c][:, :self.pos_dim].clone(),
                          states=self.states[c].clone(),
                          agent_mask=self.agent_mask)
        if self.agent_mask is not None:
            data.agents_first_n = self.n
        self.env._data = self.env.add_communication_links(data)
        self.env._data.u_ref = self.u_ref[c].clone()

    def _eager_step(self, prob: float) -> bool:
        """Plain (non-captured) training step, used while the scene's edge
        count exceeds the captured buffers' capacity."""
        env, algo = self.env, self.algo
        data = env.data
        if data.u_ref is None:
            data.update(u_ref=env.u_ref(data))
        action = algo.step(data, prob)
        next_data, reward, done, info = env.step(action)
        if done:
            return True
        if next_data.num_edges <= self.E_max:
            # fits again: reload the captured buffers and resume replaying
            self._load_graph_from_env()
        return False

    # ----------------------------------------------------------------- step
    def step(self, prob: float) -> bool:
        """One training env step.  Returns done."""
        if self._eager:
            return self._eager_step(prob)
        # snapshot the CURRENT graph for the replay buffer before the replay
        # overwrites the static buffers.  With the ring store attached the
        # snapshot is metadata-only (2 copies into the ring instead of 5
        # tensor clones per step — update batches are rebuilt from the ring,
        # never from stored graph tensors; rocprof r02 measured the clones
        # at ~25% of the rollout step)
        E = self.E
        c = self.cur
        ring = getattr(self.algo, "_ring", None)
        if ring is not None and self.algo.buffer.on_append is ring.push:
            rid = ring.push_raw(self.states[c], self.u_ref[c])
            snap = GraphBatch(
                x=self.x, pos=None, states=None,
                agent_mask=self.agent_mask)
            snap.ring_id = rid
            snap._edge_count = E
        else:
            snap = GraphBatch(
                x=self.x,  # static content, shared
                pos=self.states[c][:, :self.pos_dim].clone(),
                states=self.states[c].clone(),
                edge_index=self.ei[c][:, :E].clone(),
                edge_attr=self.ea[c][:E].clone(),
                agent_mask=self.agent_mask,
                u_ref=self.u_ref[c].clone())

        self.g[(np.random.rand() < prob, c)].replay()
        self.cur = c = 1 - c

        flags = self.flags.cpu()  # ONE host sync per step
        self.E = int(flags[0])
        reach_all = bool(flags[1])
        is_safe = not bool(flags[2])

        self.algo.buffer.append(snap, is_safe)

        self.env._t += 1
        done = self.env._t >= self.env.max_episode_steps or reach_all
        if self.E > self.E_max:
            # soft-capacity overflow: the state advance / rewards / flags are
            # exact (they don't depend on the rebuilt graph) but the new edge
            # buffers are truncated — rebuild exactly and step eagerly until
            # the scene thins out again
            self.E = 0
            self._eager = True
            if not done:
                self._sync_env_from_buffers()
            return done
        if not done:
            # keep the env object's view of the world consistent (cheap:
            # shares the engine's buffers; env methods are not used for
            # captured stepping)
            self.env._data = GraphBatch(
                x=self.x, pos=self.states[c][:, :self.pos_dim],
                states=self.states[c], edge_index=self.ei[c],
                edge_attr=self.ea[c],
                agent_mask=self.agent_mask, u_ref=self.u_ref[c])
            self.env._data.seg_dst = self.seg[c]
        return done
