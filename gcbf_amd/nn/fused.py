"""Fused MFMA linear layers: autograd wrapper, bf16 weight mirrors, and the
MLP execution plan.

Forward runs the hand-written CDNA4 MFMA kernel (ops/hip/fused_linear.hip):
bf16 GEMM with fp32 accumulation and the bias+activation epilogue fused in.
Backward uses bf16 rocBLAS matmuls on the saved bf16 activations (the same
precision autocast training uses).

Weight casts are the silent cost of autocast-style training: a rocprof
capture showed per-call fp32→bf16 weight casts were >25% of all GPU time.
Plain Linears therefore keep a version-tracked bf16 *mirror* of their master
weights, re-cast only when the master changes (once per optimizer step).
Captured hipGraphs read the mirror storage by address, so a refresh
propagates into replays without re-capture (``sync_bf16_mirrors`` runs after
the optimizer steps).  Spectral-norm layers recompute W/σ every forward
(power-iteration semantics) and keep the differentiable cast path.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F
from torch import Tensor

ACT_NONE, ACT_RELU, ACT_TANH = 0, 1, 2

# Measured dispatch history (profiles/r02_smallm.json + rollout_bench):
# * standalone microbenches (back-to-back calls, weights L2-resident) say
#   the MFMA tile kernel wins at M<=256 by up to 4x and loses at M>=512;
# * IN CONTEXT the library wins everywhere we dispatch: captured rollout
#   policy path 0.399 ms/step with the kernel off vs 0.532 ms on
#   (weights stream from HBM each step — the microbench advantage was a
#   cache artifact), and the update at M~12k measures faster on
#   hipBLASLt too (update 0.123 s vs 0.146 s per 512 steps).
# Default is therefore library GEMMs everywhere (FUSED_MAX_M=0); the
# hand-written kernel stays built, numerics-tested and dispatchable via
# GCBF_AMD_FUSED_MAX_M for future shapes.  The always-on hand-written
# HIP surface is the graph builder / env step / mask / segment-attention
# kernel family, where no library covers the fusion.
FUSED_MAX_M = int(os.environ.get("GCBF_AMD_FUSED_MAX_M", "0"))


def _ext():
    from gcbf_amd import _C
    return _C


def _pad_k_x(x: Tensor, K64: int) -> Tensor:
    return x if x.shape[1] == K64 else F.pad(x, (0, K64 - x.shape[1]))


def shapes_ok(m_padded: int, n: int) -> bool:
    return (m_padded % 128 == 0 and n % 128 == 0 and n >= 128
            and m_padded <= FUSED_MAX_M)


# ---------------------------------------------------------------- mirrors

def _mirror(lin, force: bool = False):
    """(bf16 K-padded weight mirror, bf16 bias mirror, fp32 bias) for a
    plain nn.Linear, refreshed when the master weight version changes.

    ``force`` refreshes unconditionally: fused Adam
    (torch._fused_adam_) updates parameters WITHOUT bumping their
    version counters (measured on ROCm 7 / torch 2.10 — caught by the
    engine-vs-eager gradient parity test), so version-keyed staleness
    detection misses optimizer steps.  Every post-optimizer refresh path
    must pass force=True."""
    ver = lin.weight._version + (0 if lin.bias is None
                                 else lin.bias._version)
    cache = getattr(lin, "_bf16_mirror", None)
    if not force and cache is not None and cache[0] == ver:
        return cache[1], cache[2], cache[3]
    K = lin.weight.shape[1]
    K64 = (K + 63) // 64 * 64
    with torch.no_grad():
        wb_new = F.pad(lin.weight.detach().bfloat16(),
                       (0, K64 - K)).contiguous()
        bb_new = None if lin.bias is None else \
            lin.bias.detach().bfloat16().contiguous()
        # copy=True: .float() on an fp32 bias would alias the master and
        # the refresh copy_ would mutate it (version-bump cascade)
        b32_new = None if lin.bias is None else \
            lin.bias.detach().to(torch.float32, copy=True).contiguous()
        if cache is not None:
            # reuse storage so captured graphs pick up the refresh
            cache[1].copy_(wb_new)
            if bb_new is not None:
                cache[2].copy_(bb_new)
                cache[3].copy_(b32_new)
            lin._bf16_mirror = (ver, cache[1], cache[2], cache[3])
        else:
            lin._bf16_mirror = (ver, wb_new, bb_new, b32_new)
    return lin._bf16_mirror[1], lin._bf16_mirror[2], lin._bf16_mirror[3]


def sync_bf16_mirrors(module: torch.nn.Module):
    """Refresh every existing mirror after an optimizer step (captured
    replays skip Python, so the refresh must be explicit; force=True
    because fused Adam does not bump version counters)."""
    for m in module.modules():
        if getattr(m, "_bf16_mirror", None) is not None:
            _mirror(m, force=True)


# ----------------------------------------------------------- autograd path

class _FusedLinearAct(torch.autograd.Function):
    """act(x @ w^T + b) on the MFMA kernel; wb is the bf16 (K-padded)
    compute weight, w the fp32 master that receives the gradient."""

    @staticmethod
    def forward(ctx, x: Tensor, w: Tensor, bias: Optional[Tensor],
                act: int, out_fp32: bool, wb: Optional[Tensor],
                b32: Optional[Tensor]):
        if wb is None:
            K64 = (w.shape[1] + 63) // 64 * 64
            wb = F.pad(w.detach().bfloat16(),
                       (0, K64 - w.shape[1])).contiguous()
        xb = x if x.dtype == torch.bfloat16 else x.bfloat16()
        xb = _pad_k_x(xb, wb.shape[1]).contiguous()
        if bias is not None and b32 is None:
            b32 = bias.detach().to(torch.float32, copy=True).contiguous()
        out = _ext().fused_linear(xb, wb, b32, act, out_fp32)
        ctx.save_for_backward(xb, wb, out)
        ctx.act = act
        ctx.x_dtype = x.dtype
        ctx.orig_k = x.shape[1]
        ctx.has_bias = bias is not None
        return out

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        xb, wb, out = ctx.saved_tensors
        act = ctx.act
        g = grad_out
        if act == ACT_RELU:
            g = g * (out > 0)
        elif act == ACT_TANH:
            of = out.float()
            g = g.float() * (1.0 - of * of)
        g16 = g if g.dtype == torch.bfloat16 else g.bfloat16()
        dx = (g16 @ wb)[:, :ctx.orig_k].to(ctx.x_dtype)
        dw = (g16.t() @ xb).float()[:, :ctx.orig_k]
        db = g.float().sum(0) if ctx.has_bias else None
        return dx, dw, db, None, None, None, None


def fused_linear_act(x: Tensor, w: Tensor, bias: Optional[Tensor], act: int,
                     out_fp32: bool, wb: Optional[Tensor] = None,
                     b32: Optional[Tensor] = None) -> Tensor:
    return _FusedLinearAct.apply(x, w, bias, act, out_fp32, wb, b32)


# ------------------------------------------------------------ plan runner

def run_plan(plan, x: Tensor, bucket: int = 256) -> Tensor:
    """Execute an MLP's (linear, act) plan with the fused MFMA kernels where
    shapes allow, falling back to hipBLASLt per layer otherwise.  The row
    dim is padded to ``bucket`` once up front and sliced at the end."""
    from .mlp import SNLinear
    M = x.shape[0]
    if M == 0:
        for lin, act in plan:
            x = _apply_act(F.linear(x, _weight_of(lin), lin.bias), act)
        return x
    Mp = (M + bucket - 1) // bucket * bucket
    h = torch.cat([x, x.new_zeros(Mp - M, x.shape[1])]) if Mp != M else x
    n_layers = len(plan)
    grad_mode = torch.is_grad_enabled()
    for i, (lin, act) in enumerate(plan):
        last = i == n_layers - 1
        if isinstance(lin, SNLinear):
            # differentiable W/σ, cast per forward (power-iteration
            # semantics); σ is recomputed each call by design
            w = lin.effective_weight()
            if shapes_ok(Mp, w.shape[0]):
                h = fused_linear_act(h, w, lin.bias, act, last)
            else:
                h = _apply_act(F.linear(h.float(), w, lin.bias), act)
            continue
        wb, bb, b32 = _mirror(lin)
        if shapes_ok(Mp, wb.shape[0]):
            if grad_mode:
                h = fused_linear_act(h, lin.weight, lin.bias, act, last,
                                     wb=wb, b32=b32)
            else:
                hb = h if h.dtype == torch.bfloat16 else h.bfloat16()
                h = _ext().fused_linear(
                    _pad_k_x(hb, wb.shape[1]).contiguous(), wb, b32, act,
                    last)
        else:
            hb = h if h.dtype == torch.bfloat16 else h.bfloat16()
            if grad_mode:
                # small layer: hipBLASLt bf16 GEMM, grads via autograd
                # through the cast of the master weight
                h = _apply_act(F.linear(
                    hb, lin.weight.bfloat16(),
                    None if lin.bias is None else lin.bias.bfloat16()), act)
            else:
                h = _apply_act(F.linear(hb, wb[:, :lin.weight.shape[1]], bb),
                               act)
            if last:
                h = h.float()
    return h[:M] if Mp != M else h


def _weight_of(lin) -> Tensor:
    from .mlp import SNLinear
    if isinstance(lin, SNLinear):
        return lin.effective_weight()
    return lin.weight


def _apply_act(x: Tensor, act: int) -> Tensor:
    if act == ACT_RELU:
        return torch.relu(x)
    if act == ACT_TANH:
        return torch.tanh(x)
    return x
