"""Scalar metrics writer.

Drop-in replacement for the subset of ``torch.utils.tensorboard
.SummaryWriter`` the framework uses (``add_scalar``).  Writes JSONL per run
(always, cheap and greppable) and mirrors to real TensorBoard when the
optional ``tensorboard`` package is importable.  Single-writer discipline
for data-parallel runs: only rank 0 should construct one.
"""
from __future__ import annotations

import json
import os
import time
from typing import Optional


class SummaryWriter:

    def __init__(self, log_dir: str):
        self.log_dir = log_dir
        os.makedirs(log_dir, exist_ok=True)
        self._f = open(os.path.join(log_dir, "scalars.jsonl"), "a",
                       buffering=1024 * 64)
        self._tb = None
        try:
            from torch.utils.tensorboard import SummaryWriter as TBWriter
            self._tb = TBWriter(log_dir=log_dir)
        except Exception:
            pass
        self._t0 = time.time()

    def add_scalar(self, tag: str, value: float, step: Optional[int] = None):
        self._f.write(json.dumps(
            {"tag": tag, "value": float(value), "step": step,
             "t": round(time.time() - self._t0, 3)}) + "\n")
        if self._tb is not None:
            self._tb.add_scalar(tag, value, step)

    def flush(self):
        self._f.flush()
        if self._tb is not None:
            self._tb.flush()

    def close(self):
        self.flush()
        self._f.close()
        if self._tb is not None:
            self._tb.close()


class NullWriter:
    """No-op writer for non-zero DP ranks."""

    def add_scalar(self, tag, value, step=None):
        pass

    def flush(self):
        pass

    def close(self):
        pass
