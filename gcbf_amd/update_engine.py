"""hipGraph-captured training update engine.

At small scenes (the headline DubinsCar n=16 config) one eager inner
update iteration costs ~15-30 ms of which most is kernel-launch/Python
gaps: ~300 tiny graphs are re-batched in Python, then ~200 small kernels
run for the four GNN forwards, the losses and the backward.  This engine
replays the whole iteration as two hipGraphs over fixed-capacity buffers:

    graph FRONT (no_grad): gather sampled graphs from a device ring
        (index_select) → padded batched radius-graph build (fixed E_cap)
        → actor forward → env.forward_graph → padded RE-LINK build of the
        next graph → CBF forward of the re-linked graph (the ḣ residue)
        → publish both edge counts
    one 8-byte host read of the two edge counts (overflow → exact eager
        fallback; FRONT touches no weights, so fallback is always safe)
    graph BACK (via torch.cuda.make_graphed_callables): actor forward →
        forward_graph → doubled-batch CBF (h and h_next under one
        spectral-norm σ) → weighted-sum losses; ``loss.backward()``
        replays its captured backward graph
    eager tail: grad all-reduce (DP) + clip + Adam + bf16 mirror refresh

Why BACK recomputes the actor: autograd must not cross capture
boundaries (backward kernels are launched on the stream their forward was
captured on, so a backward recorded in a later graph silently leaks out of
the capture — observed as stale gradients).  Hand-capturing
``torch.autograd.grad`` in one self-contained graph ALSO failed subtly:
the returned grad tensors were pool-allocated over freed forward
temporaries and bias gradients came back aliased (one grad had norm
exactly 1.0 — a spectral-norm power-iteration temporary).
``make_graphed_callables`` is torch's supported fwd+bwd capture and owns
its static input/output/grad surfaces.  FRONT's actor/next-state kernels
are the same recorded kernel sequence on the same buffers, so its re-link
residue is bitwise consistent with BACK's differentiable path.

Design points:
* The replay buffer's graphs are mirrored into a device ring of states +
  u_ref at append time (``Buffer.on_append``).  Node features x and the
  agent layout are static for a given env config, and edges are an exact
  deterministic function of positions (the same HIP builder that built
  them at rollout time), so the ring fully reconstructs any sampled graph.
* Static shapes come from padding: the graph count is padded to G_cap
  (3 · batch_size/5, the sampler's maximum) with weight-0 copies of a real
  graph, and edge buffers hold E_cap entries with the rollout engine's
  sentinel-segment scheme.  Every loss/accuracy mean becomes a weighted
  sum, exactly equal on the real rows.
* ``loss.backward()`` on the graphed callable replays the captured
  backward and routes the static parameter grads through eager
  AccumulateGrad into ``param.grad``.  Optimizer, clipping and
  communication stay eager — no capturable-optimizer or
  captured-collective requirements, and the engine composes with data
  parallelism unchanged.

Reference behavior preserved (gcbf/algo/gcbf.py:140-230): sampling,
balanced replay, the ḣ re-link residue trick, loss forms and coefficients,
and the u_ref-from-current-goal behavior of replayed graphs
(``forward_graph`` reads the env's live goal, kept address-stable here).
"""
from __future__ import annotations

import torch

from .graph import GraphBatch


class UpdateEngine:
    WARMUP_ITERS = 3

    def __init__(self, algo, env):
        from gcbf_amd import _C
        self._ext = _C
        self.algo = algo
        self.env = env
        self.device = env.device

        data = env.data
        assert data is not None
        self.N = data.num_nodes            # nodes per graph (incl. obstacles)
        self.n = env.num_agents            # agent rows come first per graph
        self.S = env.state_dim
        self.pd = 3 if env.state_dim == 6 else 2
        self.nd = env.node_dim
        self.ad = env.action_dim
        self.ed = env.edge_dim

        seg_len = 3
        self.G_cap = seg_len * (algo.batch_size // 5)
        self.Ntot = self.G_cap * self.N
        self.nA = self.G_cap * self.n      # agent rows in the padded batch
        if algo.buffer.size == 0:
            raise RuntimeError("empty buffer at engine init")

        # edge capacity: a sampled batch of G_cap graphs concentrates hard
        # around G_cap·mean_e edges (~300 draws), so a 1.35x headroom keeps
        # overflow (exact eager fallback) rare while padded GEMM rows — pure
        # wasted FLOPs — stay modest (4x padding measured 3x slower updates)
        mean_e = max(1.0, sum(g.num_edges for g in algo.buffer.data)
                     / algo.buffer.size)
        full = self.G_cap * self.n * (self.N - 1)   # agents receive only
        want = int(1.35 * mean_e * self.G_cap) + 1024
        self.E_cap = min(full, (want + 2047) // 2048 * 2048)

        dev = self.device
        # ---- shared device ring (owned by the algo; also feeds the fast
        # eager batcher, gcbf_amd/ring.py)
        self.ring = algo._ring
        if self.ring is None:
            raise RuntimeError("ring store unavailable")
        assert self.ring.N == self.N and self.ring.n == self.n

        # ---- static tiles / batch skeleton
        self.x_tile = data.x.repeat(self.G_cap, 1).contiguous()
        self.x2_tile = data.x.repeat(2 * self.G_cap, 1).contiguous()
        self.ptr = torch.arange(self.G_cap + 1, dtype=torch.long,
                                device=dev) * self.N
        self.ptr2 = torch.arange(2 * self.G_cap + 1, dtype=torch.long,
                                 device=dev) * self.N
        # static agent layout: rows [g*N, g*N + n) are the agents of graph g
        base = torch.arange(self.G_cap, device=dev).unsqueeze(1) * self.N
        self.agent_index = (base + torch.arange(self.n, device=dev)
                            ).reshape(-1)
        self.agent_index2 = torch.cat([self.agent_index,
                                       self.agent_index + self.Ntot])
        if self.N == self.n:
            self.agent_mask_tile = None
        else:
            am = torch.zeros(self.N, dtype=torch.bool, device=dev)
            am[: self.n] = True
            self.agent_mask_tile = am.repeat(self.G_cap)

        # ---- per-iteration inputs (content updated before each replay)
        self.idx_host = torch.empty(self.G_cap, dtype=torch.long,
                                    pin_memory=True)
        self.w_host = torch.empty(self.G_cap, pin_memory=True)
        self.idx_dev = torch.zeros(self.G_cap, dtype=torch.long, device=dev)
        self.w_dev = torch.zeros(self.G_cap, device=dev)

        # goal must be address-stable: forward_graph reads env._goal (the
        # reference's replayed-graph u_ref uses the CURRENT goal)
        self.goal_static = env._goal.clone().contiguous()
        env._goal = self.goal_static

        self._build()

    # ------------------------------------------------------------- bodies
    def _gather(self):
        nodes = self.ring.states.index_select(
            0, self.idx_dev).reshape(self.Ntot, self.S)
        uref = self.ring.uref.index_select(
            0, self.idx_dev).reshape(self.nA, self.ad)
        return nodes, uref

    def _build_cur(self, nodes, uref):
        env = self.env
        ei, seg, ea, ecount = self._ext.build_graph_padded(
            nodes[:, :self.pd].contiguous(), nodes, self.G_cap, self.n,
            env.params["comm_radius"], -1, env._attr_kind, self.ed,
            self.E_cap)
        gcur = GraphBatch(x=self.x_tile, pos=nodes[:, :self.pd],
                          states=nodes, edge_index=ei, edge_attr=ea,
                          agent_mask=self.agent_mask_tile,
                          u_ref=uref, ptr=self.ptr)
        gcur.seg_dst = seg
        gcur.agent_index = self.agent_index
        return gcur, ecount

    def _front(self):
        """no_grad: gather + build + actor + next states + padded re-link
        + CBF of the re-linked graph (the ḣ residue value)."""
        env, algo = self.env, self.algo
        with torch.no_grad():
            nodes, uref = self._gather()
            gcur, ecount = self._build_cur(nodes, uref)
            actions = algo.actor(gcur)
            gnext = env.forward_graph(gcur, actions)
            ns = gnext.states
            ei2, seg2, ea2, ecount2 = self._ext.build_graph_padded(
                ns[:, :self.pd].contiguous(), ns.contiguous(), self.G_cap,
                self.n, env.params["comm_radius"], -1, env._attr_kind,
                self.ed, self.E_cap)
            grel = GraphBatch(x=self.x_tile, pos=ns[:, :self.pd], states=ns,
                              edge_index=ei2, edge_attr=ea2,
                              agent_mask=self.agent_mask_tile, ptr=self.ptr)
            grel.seg_dst = seg2
            grel.agent_index = self.agent_index
            h_new = algo.cbf(grel)
            ecounts = torch.stack([ecount[0], ecount2[0]])
        return (gcur.states, gcur.u_ref, gcur.edge_index, gcur.edge_attr,
                gcur.seg_dst, h_new, ecounts)

    def _back(self, nodes, uref, ei, ea, seg, h_new, w_dev):
        """Differentiable stage: actor → forward_graph → doubled CBF →
        weighted losses.  Captured via torch.cuda.make_graphed_callables
        (hand-capturing the backward measured corrupted bias gradients —
        grad buffers aliased freed forward temporaries in the shared pool)."""
        algo, env = self.algo, self.env
        p = algo.params
        eps, alpha = p["eps"], p["alpha"]

        gcur = GraphBatch(x=self.x_tile, pos=nodes[:, :self.pd],
                          states=nodes, edge_index=ei, edge_attr=ea,
                          agent_mask=self.agent_mask_tile,
                          u_ref=uref, ptr=self.ptr)
        gcur.seg_dst = seg
        gcur.agent_index = self.agent_index
        actions = algo.actor(gcur)
        gnext = env.forward_graph(gcur, actions)
        gnext.agent_index = self.agent_index

        states2 = torch.cat([gcur.states, gnext.states], dim=0)
        ea2 = torch.cat([gcur.edge_attr, gnext.edge_attr], dim=0)
        ei2 = torch.cat([gcur.edge_index, gcur.edge_index + self.Ntot],
                        dim=1)
        # pad sentinel N_tot must stay past the LAST node of the doubled
        # batch (it would otherwise alias the second half's first node)
        pad = seg == self.Ntot
        seg2 = torch.cat([torch.where(pad, 2 * self.Ntot, seg),
                          seg + self.Ntot])
        both = GraphBatch(x=self.x2_tile, pos=states2[:, :self.pd],
                          states=states2, edge_index=ei2, edge_attr=ea2,
                          agent_mask=None if self.agent_mask_tile is None
                          else self.agent_mask_tile.repeat(2),
                          ptr=self.ptr2)
        both.seg_dst = seg2
        both.agent_index = self.agent_index2
        h_both = algo.cbf(both)
        h, h_next = h_both[:self.nA], h_both[self.nA:]
        hv = h[:, 0]

        # per-AGENT-row weights (h/actions/masks are per agent)
        w_node = w_dev.view(self.G_cap, 1).expand(
            self.G_cap, self.n).reshape(self.nA)
        cw = w_node.sum()

        # identical weighted-sum forms as GCBF._iter_eager, with the pad
        # rows carrying weight 0
        wu = env.unsafe_mask(gcur).to(hv.dtype) * w_node
        cu = wu.sum()
        cu1 = cu.clamp(min=1)
        any_u = (cu > 0).to(hv.dtype)
        loss_unsafe = any_u * (torch.relu(hv + eps) * wu).sum() / cu1
        acc_unsafe = (any_u * ((hv < 0).to(hv.dtype) * wu).sum() / cu1
                      + (1 - any_u))

        ws = env.safe_mask(gcur).to(hv.dtype) * w_node
        cs = ws.sum()
        cs1 = cs.clamp(min=1)
        any_s = (cs > 0).to(hv.dtype)
        loss_safe = any_s * (torch.relu(-hv + eps) * ws).sum() / cs1
        acc_safe = (any_s * ((hv >= 0).to(hv.dtype) * ws).sum() / cs1
                    + (1 - any_s))

        h_dot = (h_next - h) / env.dt
        h_dot_new_link = (h_new - h) / env.dt
        residue = (h_dot_new_link - h_dot).detach()
        h_dot = residue + h_dot
        hd = h_dot[:, 0]
        ha = hv * alpha
        loss_h_dot = (torch.relu(-hd - ha + eps) * w_node).sum() / cw
        acc_h_dot = (((hd + ha) >= 0).to(hv.dtype) * w_node).sum() / cw

        loss_action = (torch.square(actions).sum(dim=1) * w_node).sum() / cw

        loss = (p["loss_unsafe_coef"] * loss_unsafe +
                p["loss_safe_coef"] * loss_safe +
                p["loss_h_dot_coef"] * loss_h_dot +
                p["loss_action_coef"] * loss_action)

        log7 = torch.stack([
            loss_unsafe.detach(), loss_safe.detach(), loss_h_dot.detach(),
            loss_action.detach(), acc_unsafe.detach(), acc_safe.detach(),
            acc_h_dot.detach()])
        return loss, log7

    # ------------------------------------------------------------ capture
    def _fill_inputs(self, graph_list):
        L = len(graph_list)
        cap = self.ring.CAP
        ids = [g.ring_id for g in graph_list]
        self.idx_host[:L] = torch.tensor([i % cap for i in ids],
                                         dtype=torch.long)
        self.idx_host[L:] = ids[0] % cap
        self.w_host[:L] = 1.0
        self.w_host[L:] = 0.0
        self.idx_dev.copy_(self.idx_host, non_blocking=True)
        self.w_dev.copy_(self.w_host, non_blocking=True)

    def _sample_for_warmup(self):
        algo = self.algo
        seg_len = 3
        if algo.memory.size == 0:
            return algo.buffer.sample(algo.batch_size // 5, seg_len)
        curr = algo.buffer.sample(algo.batch_size // 10, seg_len, True)
        prev = algo.memory.sample(
            algo.batch_size // 5 - algo.batch_size // 10, seg_len, True)
        return curr + prev

    def _opt_tail(self):
        algo = self.algo
        if algo.grad_sync is not None:
            algo.grad_sync()
        torch.nn.utils.clip_grad_norm_(algo.cbf.parameters(), 1e-3)
        torch.nn.utils.clip_grad_norm_(algo.actor.parameters(), 1e-3)
        algo.optim_cbf.step()
        algo.optim_actor.step()
        # captured graphs (rollout AND this engine's) read the bf16
        # mirrors by address — refresh after every weight change
        from .nn.fused import sync_bf16_mirrors
        sync_bf16_mirrors(algo.actor)
        sync_bf16_mirrors(algo.cbf)

    def _zero_grads(self):
        self.algo.optim_cbf.zero_grad(set_to_none=True)
        self.algo.optim_actor.zero_grad(set_to_none=True)

    def _build(self):
        algo = self.algo
        self._back_mod = _BackCallable(self)

        # the whole engine (both captures + warmup) runs the MLPs on plain
        # autocast GEMMs: (a) the custom fused-linear Function measured
        # corrupted gradients under graphed capture, and (b) FRONT and BACK
        # must be numerically IDENTICAL per row — the ḣ residue divides an
        # h_new−h_next difference by dt (~33x amplification), so a
        # fused-vs-hipBLASLt mismatch between the two stages shifts the
        # re-link residue.  Launch cost is amortized by the capture anyway;
        # the fused kernels stay active for rollout and eager paths.
        from .nn.mlp import MLP
        self._mlps = [m for mod in (algo.cbf, algo.actor)
                      for m in mod.modules()
                      if isinstance(m, MLP) and getattr(m, "fused_mfma",
                                                        False)]
        for m in self._mlps:
            m.fused_mfma = False
        try:
            self._build_captures()
        finally:
            # restore the fused path for rollout / eager consumers (the
            # captured graphs recorded the plain-GEMM kernels already)
            for m in self._mlps:
                m.fused_mfma = True

    def _build_captures(self):
        # warmup: real iterations over the engine's padded shapes (warms
        # hipBLASLt shape caches, materializes Adam state + bf16 mirrors)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self.WARMUP_ITERS):
                self._fill_inputs(self._sample_for_warmup())
                front = self._front()
                if int(front[6].max().cpu()) > self.E_cap:
                    raise RuntimeError("edge overflow during warmup")
                loss, _ = self._back_mod(*front[:6], self.w_dev)
                self._zero_grads()
                loss.backward()
                self._opt_tail()
        torch.cuda.current_stream().wait_stream(s)

        self._fill_inputs(self._sample_for_warmup())
        self.gFront = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.gFront):
            (self._nodes, self._uref, self._ei, self._ea, self._seg,
             self._h_new, self._ecounts) = self._front()
        # capture records without executing — replay once so the static
        # buffers hold real contents before the back stage warms/captures
        self.gFront.replay()
        # the differentiable stage is captured by torch's supported fwd+bwd
        # capture (its own pool, its own static grad buffers)
        self._graphed = torch.cuda.make_graphed_callables(
            self._back_mod,
            (self._nodes, self._uref, self._ei, self._ea, self._seg,
             self._h_new, self.w_dev))

    # -------------------------------------------------------------- iter
    def try_iter(self, graph_list):
        """One captured inner iteration; returns the 7-scalar log stack,
        or None to request the exact eager fallback (no weights touched)."""
        if len(graph_list) > self.G_cap:
            return None
        if any(g.ring_id is None for g in graph_list):
            return None
        # goal address-stability across env resets
        if self.env._goal is not self.goal_static:
            self.goal_static.copy_(self.env._goal)
            self.env._goal = self.goal_static
        self._fill_inputs(graph_list)
        self.gFront.replay()
        if int(self._ecounts.max().cpu()) > self.E_cap:
            return None
        loss, log7 = self._graphed(self._nodes, self._uref, self._ei,
                                   self._ea, self._seg, self._h_new,
                                   self.w_dev)
        self._zero_grads()
        loss.backward()
        self._opt_tail()
        return log7.clone()


class _BackCallable(torch.nn.Module):
    """nn.Module wrapper so make_graphed_callables tracks the CBF and actor
    parameters; forwards to the engine's differentiable stage."""

    def __init__(self, engine: UpdateEngine):
        super().__init__()
        self.cbf = engine.algo.cbf
        self.actor = engine.algo.actor
        self._engine = engine

    def forward(self, nodes, uref, ei, ea, seg, h_new, w_dev):
        return self._engine._back(nodes, uref, ei, ea, seg, h_new, w_dev)
