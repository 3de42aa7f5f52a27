"""Mixed precision: bf16 compute in the GNN GEMMs, fp32 everywhere else.

The BASELINE config trains bf16 on MI355X.  Policy: master weights and all
loss/ḣ finite-difference arithmetic stay fp32; the module forwards (where
>99% of the FLOPs are — the φ/γ 2048-wide GEMMs) run under autocast bf16
with fp32 accumulation, and outputs are cast back to fp32 at the module
boundary.
"""
from __future__ import annotations

import functools

import torch


def _wrap_forward(module: torch.nn.Module):
    orig = module.forward

    @functools.wraps(orig)
    def wrapped(*args, **kwargs):
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = orig(*args, **kwargs)
        return out.float()

    module.forward = wrapped


def enable_bf16(algo):
    """Enable bf16 autocast on the algorithm's networks (GPU only)."""
    if not torch.cuda.is_available():
        return algo
    for m in (algo.cbf, algo.actor):
        _wrap_forward(m)
    return algo
