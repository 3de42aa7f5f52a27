"""Device ring of replayed graph states + fast batched re-batching.

The reference re-batches ~300 sampled graphs per inner update iteration in
Python (``Batch.from_data_list``, gcbf/algo/gcbf.py:159); our ``from_list``
vectorizes the edge offsets but still concatenates ~4 tensors per graph
(~1200 small device ops per iteration — host-bound).  Since every graph an
environment produces has (a) a static node-feature/agent layout and (b)
edges that are an exact deterministic function of positions (the same
builder that made them at rollout time), a sampled batch can instead be
materialized as:

    index_select over a ring of per-step states/u_ref (2 kernels)
    → one batched exact radius-graph build

which is bitwise identical to concatenating the stored graphs.

``RingStore`` owns the ring and is shared by the eager update path
(``GCBF._iter_eager``) and the captured update engine.
"""
from __future__ import annotations

import torch

from .graph import GraphBatch


class RingStore:

    def __init__(self, env, capacity: int):
        data = env.data
        assert data is not None
        self.env = env
        self.device = env.device
        self.N = data.num_nodes
        self.n = env.num_agents
        self.S = env.state_dim
        self.pd = 3 if env.state_dim == 6 else 2
        self.ad = env.action_dim
        self.CAP = capacity
        dev = self.device
        self.states = torch.zeros(self.CAP, self.N, self.S, device=dev)
        self.uref = torch.zeros(self.CAP, self.n, self.ad, device=dev)
        self.next_id = 0
        self._x1 = data.x.clone()
        am = data.agent_mask
        self._am1 = None if am is None else am.clone()

    # ------------------------------------------------------------- append
    def push(self, g: GraphBatch):
        """Mirror an appended graph into the ring (Buffer.on_append hook)."""
        if g.ring_id is not None:
            return
        slot = self.next_id % self.CAP
        self.states[slot].copy_(g.states, non_blocking=True)
        self.uref[slot].copy_(g.u_ref, non_blocking=True)
        g.ring_id = self.next_id
        self.next_id += 1

    def push_raw(self, states, u_ref) -> int:
        """Copy raw state/u_ref tensors into the next slot; returns the
        ring id.  Used by the rollout engine to append WITHOUT cloning a
        full graph snapshot first (the clone would be dead weight: batches
        are rebuilt from the ring, never from the stored tensors)."""
        slot = self.next_id % self.CAP
        self.states[slot].copy_(states, non_blocking=True)
        self.uref[slot].copy_(u_ref, non_blocking=True)
        rid = self.next_id
        self.next_id += 1
        return rid

    def resident(self, ring_id: int) -> bool:
        return ring_id is not None and self.next_id - ring_id <= self.CAP

    # -------------------------------------------------------------- batch
    def batch(self, graph_list) -> GraphBatch:
        """Materialize a sampled batch: ring gather + one exact batched
        radius-graph build.  Exactly equals ``GraphBatch.from_list`` on the
        stored graphs (the builder is deterministic and the stored edges
        came from the same builder on the same states)."""
        L = len(graph_list)
        ids = torch.tensor([g.ring_id % self.CAP for g in graph_list],
                           dtype=torch.long)
        idx = ids.to(self.device, non_blocking=True)
        nodes = self.states.index_select(0, idx).reshape(L * self.N, self.S)
        uref = self.uref.index_select(0, idx).reshape(L * self.n, self.ad)
        ptr = torch.arange(L + 1, dtype=torch.long,
                           device=self.device) * self.N
        am = None if self._am1 is None else self._am1.repeat(L)
        g = GraphBatch(x=self._x1.repeat(L, 1),
                       pos=nodes[:, :self.pd], states=nodes,
                       agent_mask=am, ptr=ptr)
        g = self.env.add_communication_links_batched(g)
        g.u_ref = uref
        return g

    def usable(self, graph_list) -> bool:
        return all(g.ring_id is not None and self.resident(g.ring_id)
                   for g in graph_list)
