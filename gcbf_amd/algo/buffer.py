"""Replay buffer of graphs with balanced safe/unsafe segment sampling.

Behavioral equivalent of the reference ``Buffer`` (gcbf/algo/buffer.py:11-95):
graphs are stored as a list (on device — 288 GB HBM3E holds the full 100k-graph
window comfortably), with index lists of safe/unsafe steps; sampling draws
balanced safe/unsafe indices and expands each into a symmetric window of
consecutive steps, deduplicated by clamping to the previous window's end.
"""
from __future__ import annotations

import random
from typing import List

import numpy as np

from ..graph import GraphBatch


class Buffer:
    MAX_SIZE = 100000

    def __init__(self):
        self._data: List[GraphBatch] = []
        self.safe_data: List[int] = []
        self.unsafe_data: List[int] = []
        # (index, 0-dim bool tensor) flags not yet classified — the per-step
        # is_safe device→host sync is deferred and resolved in ONE transfer
        # at sample/merge time (the reference syncs every rollout step,
        # gcbf/algo/gcbf.py:133-137)
        self._pending: List[tuple] = []
        # optional hook: called with each appended graph (the captured
        # update engine uses it to mirror states into its device ring)
        self.on_append = None

    @property
    def data(self) -> List[GraphBatch]:
        return self._data

    @property
    def size(self) -> int:
        return len(self._data)

    def _resolve(self):
        if not self._pending:
            return
        import torch
        vals = torch.stack([t for _, t in self._pending]).cpu()
        for (idx, _), v in zip(self._pending, vals):
            (self.safe_data if bool(v) else self.unsafe_data).append(idx)
        self._pending.clear()

    def append(self, data: GraphBatch, is_safe):
        if self.on_append is not None:
            self.on_append(data)
        self._data.append(data)
        import torch
        if torch.is_tensor(is_safe):
            self._pending.append((self.size - 1, is_safe))
        else:
            (self.safe_data if is_safe
             else self.unsafe_data).append(self.size - 1)
        if self.size > self.MAX_SIZE:
            self._resolve()
            del self._data[0]
            if 0 in self.safe_data:
                self.safe_data.remove(0)
            else:
                self.unsafe_data.remove(0)
            self.safe_data = [i - 1 for i in self.safe_data]
            self.unsafe_data = [i - 1 for i in self.unsafe_data]

    def merge(self, other: "Buffer"):
        other._resolve()
        self._resolve()
        size_init = self.size
        self._data += other.data
        self.safe_data.extend(i + size_init for i in other.safe_data)
        self.unsafe_data.extend(i + size_init for i in other.unsafe_data)
        if self.size > self.MAX_SIZE:
            overflow = self.size - self.MAX_SIZE
            for i in range(overflow):
                if i in self.safe_data:
                    self.safe_data.remove(i)
                else:
                    self.unsafe_data.remove(i)
            self.safe_data = [i - overflow for i in self.safe_data]
            self.unsafe_data = [i - overflow for i in self.unsafe_data]
            del self._data[:overflow]

    def clear(self):
        self._data.clear()
        self.safe_data = []
        self.unsafe_data = []

    def sample(self, n: int, m: int = 1,
               balanced_sampling: bool = False) -> List[GraphBatch]:
        """Sample n trajectory segments of up to m consecutive graphs.

        Matches reference gcbf/algo/buffer.py:61-95: indices are drawn (50/50
        from safe/unsafe lists when balanced), sorted, and each expanded to
        [i - m//2, i + m//2] with the lower bound clamped to the previous
        segment's upper bound (avoids duplicated graphs).
        """
        self._resolve()
        assert self.size >= max(n, m)
        if not balanced_sampling:
            index = np.sort(np.random.randint(0, self.size, n))
        else:
            index_unsafe, index_safe = [], []
            if self.unsafe_data:
                index_unsafe = random.choices(self.unsafe_data, k=n // 2)
            if self.safe_data:
                index_safe = random.choices(self.safe_data, k=n // 2)
            index = sorted(index_safe + index_unsafe)

        data_list: List[GraphBatch] = []
        ub = 0
        for i in index:
            lb = max(i - m // 2, ub)
            ub = min(i + m // 2 + 1, self.size)
            data_list.extend(self._data[lb:ub])
        return data_list


class RolloutBuffer:
    """RL-style (state, action, reward, done, log_pi, next_state) ring
    buffer.  API-parity port of the reference's unused RolloutBuffer
    (gcbf/algo/buffer.py:98-204); kept for downstream RL extensions."""

    def __init__(self, num_agents: int, buffer_size: int, action_dim: int,
                 device):
        import torch
        self._n = 0
        self._p = 0
        self.device = device
        self.buffer_size = buffer_size
        self.num_agents = num_agents
        self.data: List[GraphBatch] = [None] * buffer_size
        self.next_data: List[GraphBatch] = [None] * buffer_size
        self.actions = torch.empty(buffer_size, num_agents, action_dim,
                                   dtype=torch.float, device=device)
        self.rewards = torch.empty(buffer_size, num_agents,
                                   dtype=torch.float, device=device)
        self.dones = torch.empty(buffer_size, 1, dtype=torch.float,
                                 device=device)
        self.log_pis = torch.empty(buffer_size, num_agents,
                                   dtype=torch.float, device=device)

    def append(self, data: GraphBatch, action, reward, done: bool, log_pi,
               next_data: GraphBatch):
        import torch
        if action.ndim == 3:
            action = action.squeeze(0)
        self.data[self._p] = data.replace()
        self.actions[self._p].copy_(action)
        self.rewards[self._p].copy_(
            torch.as_tensor(reward, device=self.device))
        self.dones[self._p] = float(done)
        self.log_pis[self._p].copy_(
            torch.as_tensor(log_pi, device=self.device))
        self.next_data[self._p] = next_data.replace()
        self._p = (self._p + 1) % self.buffer_size
        self._n = min(self._n + 1, self.buffer_size)

    def get(self):
        assert self._p % self.buffer_size == 0
        idx = slice(0, self.buffer_size)
        return (self.data[idx], self.actions[idx], self.rewards[idx],
                self.dones[idx], self.log_pis[idx], self.next_data[idx])

    def sample(self, batch_size: int):
        idxes = np.random.randint(low=0, high=self._n, size=batch_size)
        return ([self.data[i] for i in idxes], self.actions[idxes],
                self.rewards[idxes], self.dones[idxes], self.log_pis[idxes],
                [self.next_data[i] for i in idxes])
