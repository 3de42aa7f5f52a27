"""Fused MFMA linear layers: autograd wrapper and MLP execution plan.

Forward runs the hand-written CDNA4 MFMA kernel (ops/hip/fused_linear.hip):
bf16 GEMM with fp32 accumulation and the bias+activation epilogue fused in.
Backward uses bf16 rocBLAS matmuls on the saved bf16 activations (the same
precision autocast training uses).  Weights arrive as fp32 (master weights,
or SNLinear's differentiable W/σ) and are cast at the boundary, so gradient
flow through spectral norm's σ is preserved.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F
from torch import Tensor

ACT_NONE, ACT_RELU, ACT_TANH = 0, 1, 2


def _ext():
    from gcbf_amd import _C
    return _C


class _FusedLinearAct(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x: Tensor, w: Tensor, bias: Optional[Tensor], act: int,
                out_fp32: bool):
        xb = x if x.dtype == torch.bfloat16 else x.bfloat16()
        wb = w if w.dtype == torch.bfloat16 else w.bfloat16()
        b32 = None if bias is None else (
            bias if bias.dtype == torch.float32 else bias.float())
        out = _ext().fused_linear(xb.contiguous(), wb.contiguous(),
                                  None if b32 is None else b32.contiguous(),
                                  act, out_fp32)
        ctx.save_for_backward(xb, wb, out)
        ctx.act = act
        ctx.x_dtype = x.dtype
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
        dx = (g16 @ wb).to(ctx.x_dtype)
        dw = (g16.t() @ xb).float()
        db = g.float().sum(0) if ctx.has_bias else None
        return dx, dw, db, None, None


def fused_linear_act(x: Tensor, w: Tensor, bias: Optional[Tensor], act: int,
                     out_fp32: bool) -> Tensor:
    return _FusedLinearAct.apply(x, w, bias, act, out_fp32)


def _pad_k(x: Tensor, w: Tensor):
    """Zero-pad the contraction dim up to a multiple of 64 (zeros contribute
    nothing to the dot products)."""
    K = x.shape[1]
    if K % 64 == 0:
        return x, w
    pad = 64 - K % 64
    return F.pad(x, (0, pad)), F.pad(w, (0, pad))


import os

# below this many rows the 128x128-tile kernel cannot fill 256 CUs and
# hipBLASLt's small-M kernels win (measured: rollout at M=256 is ~35% slower
# on the tile kernel; update batches at M>=8K are faster on it)
FUSED_MIN_M = int(os.environ.get("GCBF_AMD_FUSED_MIN_M", "1024"))


def shapes_ok(m_padded: int, n: int) -> bool:
    return (m_padded % 128 == 0 and n % 128 == 0 and n >= 128
            and m_padded >= FUSED_MIN_M)


def run_plan(plan, x: Tensor, bucket: int = 256) -> Tensor:
    """Execute an MLP's (linear, act) plan with the fused MFMA kernels where
    shapes allow, falling back to F.linear per layer otherwise.  The row dim
    is padded to ``bucket`` once up front and sliced at the end."""
    M = x.shape[0]
    if M == 0:
        # preserve the eager path's empty-output semantics
        for lin, act in plan:
            x = F.linear(x, _weight_of(lin), lin.bias)
            x = _apply_act(x, act)
        return x
    Mp = (M + bucket - 1) // bucket * bucket
    h = torch.cat([x, x.new_zeros(Mp - M, x.shape[1])]) if Mp != M else x
    n_layers = len(plan)
    for i, (lin, act) in enumerate(plan):
        w = _weight_of(lin)
        last = i == n_layers - 1
        if shapes_ok(Mp, w.shape[0]):
            hp, wp = _pad_k(h, w)
            h = fused_linear_act(hp, wp, lin.bias, act, out_fp32=last)
        else:
            h = F.linear(h.float() if h.dtype != torch.float32 else h,
                         w, lin.bias)
            h = _apply_act(h, act)
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
