"""Profiler annotation ranges (roctx via torch.cuda.nvtx on ROCm).

`rocprofv3 --marker-trace` (and the hip trace viewers) pick these up, so
rollout / update / collective phases are attributable in traces.  No-ops on
CPU.
"""
from __future__ import annotations

from contextlib import contextmanager

import torch

_ENABLED = torch.cuda.is_available()


@contextmanager
def trace_range(name: str):
    if _ENABLED:
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield
