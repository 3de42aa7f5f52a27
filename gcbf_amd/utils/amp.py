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
        # cache_enabled=False: the autocast weight-cast cache is documented
        # as incompatible with CUDA/hipGraph capture (cached casts go stale
        # across replays); the bf16 weight mirrors already dedup the hot
        # casts, so the cache buys nothing here
        with torch.autocast("cuda", dtype=torch.bfloat16,
                            cache_enabled=False):
            out = orig(*args, **kwargs)
        return out.float()

    module.forward = wrapped


def enable_bf16(algo, fused_mfma=None):
    """Enable bf16 compute on the algorithm's networks (GPU only):
    autocast at the module boundary, plus the hand-written MFMA fused
    linear kernels for the big MLP layers (GCBF_AMD_FUSED=0 disables)."""
    import os

    from .. import ops
    from ..nn.mlp import MLP

    if not torch.cuda.is_available():
        return algo
    for m in (algo.cbf, algo.actor):
        _wrap_forward(m)
    if fused_mfma is None:
        fused_mfma = os.environ.get("GCBF_AMD_FUSED", "1") == "1"
    if fused_mfma and ops.hip_available():
        for mod in (algo.cbf, algo.actor):
            for m in mod.modules():
                if isinstance(m, MLP):
                    m.fused_mfma = True
    return algo
