"""Op dispatch layer: HIP/CDNA4 kernels on GPU, eager PyTorch on CPU.

Every hot-path op has (a) an eager PyTorch implementation in ``eager.py``
(the numerics oracle, used on CPU and in tests) and (b) a HIP kernel for
gfx950 in ``gcbf_amd/ops/hip`` exposed through the in-tree extension
``gcbf_amd._C``.  On a GPU tensor the HIP path is mandatory: a missing
extension raises instead of silently falling back to eager (so GPU runs
always exercise the native kernels).  Set ``GCBF_AMD_ALLOW_EAGER_GPU=1``
to permit the eager path on GPU (bring-up/debug only).
"""
from __future__ import annotations

import os
from typing import Optional

import torch
from torch import Tensor

from . import eager

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from gcbf_amd import _C  # built in-tree by setup.py / __graft_entry__.build()
        _EXT = _C
    except ImportError as e:  # remember why, report on first GPU use
        _EXT_ERR = str(e)
    return _EXT


def hip_available() -> bool:
    return _load_ext() is not None


def _require_ext(opname: str):
    ext = _load_ext()
    if ext is None:
        if os.environ.get("GCBF_AMD_ALLOW_EAGER_GPU") == "1":
            return None
        raise RuntimeError(
            f"gcbf_amd HIP extension is required for {opname} on GPU but could "
            f"not be imported ({_EXT_ERR}). Build it with "
            f"`python setup.py build_ext --inplace` "
            f"(or set GCBF_AMD_ALLOW_EAGER_GPU=1 to allow the slow eager path).")
    return ext


def _csr_ptr(dst: Tensor, num_nodes: int) -> Tensor:
    """CSR segment pointer from a dst-sorted edge list (device op, no host
    sync)."""
    bounds = torch.arange(num_nodes + 1, device=dst.device, dtype=dst.dtype)
    return torch.searchsorted(dst, bounds).to(torch.int32)


# --------------------------------------------------------------------------
# segment attention aggregation (softmax over incoming edges + weighted sum)
# --------------------------------------------------------------------------

class _SegmentAttnAggregate(torch.autograd.Function):
    """Fused scatter-softmax + weighted scatter-sum with analytic VJP.

    forward: a_e = softmax over {e: dst[e]=n} of gate_e;  out_n = sum a_e m_e
    backward: dm_e = a_e * g_{n(e)};  s_e = <m_e, g_{n(e)}>;
              dgate_e = a_e * (s_e - sum_{e' in n} a_e' s_e')
    """

    @staticmethod
    def forward(ctx, msg: Tensor, gate: Tensor, dst: Tensor, num_nodes: int):
        ext = _require_ext("segment_attn_aggregate") if msg.is_cuda else None
        ctx.msg_dtype = msg.dtype
        ctx.gate_dtype = gate.dtype
        if ext is not None and msg.is_cuda:
            # kernel computes fp32 (softmax stability; the op is tiny next to
            # the bf16 GEMMs around it) — upcast bf16 inputs at the boundary
            msg = msg.float()
            ptr = _csr_ptr(dst, num_nodes)
            att, out = ext.segment_attn_fwd(
                msg.contiguous(), gate.reshape(-1).float().contiguous(), ptr)
            att = att.unsqueeze(-1)
            ctx.ptr = ptr
        else:
            att = eager.segment_softmax(gate, dst, num_nodes)
            out = eager.segment_sum(att * msg, dst, num_nodes)
            ctx.ptr = None
        ctx.save_for_backward(msg, att, dst)
        ctx.num_nodes = num_nodes
        return out

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        msg, att, dst = ctx.saved_tensors
        n = ctx.num_nodes
        ext = _EXT
        if ext is not None and msg.is_cuda and ctx.ptr is not None:
            dmsg, dgate = ext.segment_attn_bwd(
                grad_out.float().contiguous(), msg.contiguous(),
                att.reshape(-1).contiguous(), ctx.ptr)
            dgate = dgate.unsqueeze(-1)
            dmsg = dmsg.to(ctx.msg_dtype)
            dgate = dgate.to(ctx.gate_dtype)
        else:
            g = grad_out.index_select(0, dst)            # (E, D)
            dmsg = att * g
            s = (msg * g).sum(dim=1, keepdim=True)       # (E, 1)
            seg = torch.zeros(n, 1, dtype=s.dtype, device=s.device)
            seg = seg.index_add(0, dst, att * s)
            dgate = att * (s - seg.index_select(0, dst))
        return dmsg, dgate, None, None


def segment_attn_aggregate(msg: Tensor, gate: Tensor, dst: Tensor,
                           num_nodes: int) -> Tensor:
    if not msg.is_cuda:
        return eager.segment_attn_aggregate(msg, gate, dst, num_nodes)
    return _SegmentAttnAggregate.apply(msg, gate, dst, num_nodes)


def segment_softmax(gate: Tensor, dst: Tensor, num_nodes: int) -> Tensor:
    return eager.segment_softmax(gate, dst, num_nodes)


class _SegmentMaxHip(torch.autograd.Function):
    """CSR segment max on the HIP kernel (dst-sorted edges)."""

    @staticmethod
    def forward(ctx, values: Tensor, dst: Tensor, num_nodes: int):
        ext = _require_ext("segment_max")
        vdtype = values.dtype
        v32 = values if vdtype == torch.float32 else values.float()
        ptr = _csr_ptr(dst, num_nodes)
        out, argmax = ext.segment_max_fwd(v32.contiguous(), ptr)
        ctx.save_for_backward(argmax)
        ctx.E = values.shape[0]
        ctx.vdtype = vdtype
        return out.to(vdtype)

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        (argmax,) = ctx.saved_tensors
        from gcbf_amd import _C
        dval = _C.segment_max_bwd(grad_out.float().contiguous(), argmax,
                                  ctx.E)
        return dval.to(ctx.vdtype), None, None


class _SegmentMax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, values: Tensor, dst: Tensor, num_nodes: int):
        out = eager.segment_max(values.detach(), dst, num_nodes)
        # argmax for backward: first edge attaining the max in its segment
        sel = out.index_select(0, dst)
        is_max = (values.detach() == sel)
        # keep only the first maximal edge per (segment, feature)
        order = torch.arange(values.shape[0], device=values.device)
        big = values.shape[0] + 1
        cand = torch.where(is_max, order.unsqueeze(1), torch.full_like(
            is_max, big, dtype=torch.long))
        first = torch.full((num_nodes, values.shape[1]), big, dtype=torch.long,
                           device=values.device)
        first = first.scatter_reduce(0, dst.unsqueeze(1).expand_as(cand), cand,
                                     reduce="amin", include_self=True)
        ctx.save_for_backward(dst, first)
        ctx.E = values.shape[0]
        return out

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        dst, first = ctx.saved_tensors
        E = ctx.E
        dval = torch.zeros(E + 1, grad_out.shape[1], dtype=grad_out.dtype,
                           device=grad_out.device)
        idx = torch.clamp(first, max=E)
        dval.scatter_add_(0, idx, grad_out)
        return dval[:E], None, None


def segment_max(values: Tensor, dst: Tensor, num_nodes: int) -> Tensor:
    if values.is_cuda and _load_ext() is not None:
        return _SegmentMaxHip.apply(values, dst, num_nodes)
    return _SegmentMax.apply(values, dst, num_nodes)


# --------------------------------------------------------------------------
# graph construction
# --------------------------------------------------------------------------

def dense_radius_graph(pos: Tensor, agent_mask: Optional[Tensor],
                       comm_radius: float,
                       max_neighbors: Optional[int] = None,
                       batch: int = 1) -> Tensor:
    """Edge list only (eager helper; GPU paths use :func:`build_graph`)."""
    return eager.dense_radius_graph(pos, agent_mask, comm_radius,
                                    max_neighbors, batch)


# edge_attr kinds understood by the fused builder (mirrors graph_build.hip)
ATTR_DIFF = 0    # states[src] - states[dst]
ATTR_DUBINS = 1  # [x, y, theta, v cos, v sin] difference


def build_graph(pos: Tensor, states: Tensor, n_rec: Optional[int],
                comm_radius: float, max_neighbors: Optional[int],
                batch: int, attr_kind: int, attr_dim: int,
                eager_attr_fn) -> tuple:
    """Fused radius-graph + edge_attr construction.

    ``n_rec`` is the number of receiver (agent) nodes per graph — agents are
    the first ``n_rec`` rows of each graph; ``None`` means every node
    receives.  GPU: one count/scan/fill kernel pair writing edge_index and
    edge_attr in a single pass.  CPU: eager builder + the env's edge_attr
    function.  Returns (edge_index (2,E) long, edge_attr (E, attr_dim)).
    """
    N = pos.shape[0] // batch
    if n_rec is None:
        n_rec = N
    if pos.is_cuda:
        ext = _require_ext("build_graph")
        if ext is not None:
            topk = -1 if max_neighbors is None else int(max_neighbors)
            ei, ea = ext.build_graph(
                pos.contiguous(), states.contiguous(), batch, n_rec,
                float(comm_radius), topk, int(attr_kind), int(attr_dim))
            return ei, ea
    am = None
    if n_rec != N:
        am = torch.zeros(batch, N, dtype=torch.bool, device=pos.device)
        am[:, :n_rec] = True
        am = am.view(-1)
    ei = eager.dense_radius_graph(pos, am, comm_radius, max_neighbors, batch)
    return ei, eager_attr_fn(states, ei)


def pairwise_dist_masked(states: Tensor, agent_mask: Optional[Tensor],
                         batch: int, pos_dim: int, diag_offset: float) -> Tensor:
    return eager.pairwise_dist_masked(states, agent_mask, batch, pos_dim,
                                      diag_offset)


def env_step_fused(kind: str, *args, out=None):
    """Fused single-graph rollout step (env_step.hip); None on CPU/no-ext.

    kind: "dubins" | "car" | "drone"; args are forwarded to the binding.
    Returns (new_states, u_ref_next, reward, reach, collision).  ``out``
    optionally provides those five output buffers (ping-pong capture: the
    kernel writes the next phase's inputs directly).
    """
    states = args[0]
    if not states.is_cuda:
        return None
    ext = _require_ext(f"{kind}_step")
    if ext is None:
        return None
    fn = {"dubins": ext.dubins_step, "car": ext.car_step,
          "drone": ext.drone_step}[kind]
    return fn(*[a.contiguous() if torch.is_tensor(a) else a for a in args],
              out=out)


# env kinds understood by the fused mask kernel (mirrors masks.hip)
ENV_CAR = 0
ENV_DUBINS = 1
ENV_DRONE = 2

_WHICH = {"safe": 0, "unsafe": 1, "collision": 2}


def fused_masks(states: Tensor, batch: int, n_rec: int, radius: float,
                env_kind: int, which: str) -> Optional[Tensor]:
    """One-pass batched agent mask on GPU; returns None on CPU (callers fall
    back to the eager vectorized math)."""
    if not states.is_cuda:
        return None
    ext = _require_ext("fused_masks")
    if ext is None:
        return None
    idx = _WHICH[which]
    res = ext.fused_masks(states.contiguous(), batch, n_rec, float(radius),
                          env_kind, idx == 0, idx == 1, idx == 2)
    return res[idx]
