"""Minimal dependency shim that lets the UNMODIFIED reference
(MIT-REALM/gcbf-pytorch at /root/reference) execute on plain PyTorch.

Purpose: BASELINE.md's protocol requires *measured* reference numbers
(safety/reach rates + env-steps/s) on the BASELINE.json configs, but the
reference's dependencies (torch_geometric + torch_cluster/torch_scatter
CUDA wheels, cvxpy, pybullet, cv2, seaborn, tensorboard) are not
installable here (no network).  This module re-implements exactly the
surface the reference imports, with plain-torch semantics faithful to
PyG 2.3 (the pinned version), and `install()` injects it into
``sys.modules`` so `import torch_geometric` resolves here.

This is measurement harness code, NOT part of the framework: the
framework itself (gcbf_amd) never imports torch_geometric or this shim.
Faithfulness is cross-checked by tests/test_reference_shim.py, which runs
the reference's own CBFGNN through this shim against gcbf_amd's eager
implementations on identical weights and graphs.

Shimmed surfaces (reference call sites):
* torch_geometric.data.Data/Batch       (gcbf/algo/gcbf.py:7, env/*.py)
* torch_geometric.nn.Sequential         (gcbf/algo/gcbf.py:8)
* ...nn.conv.message_passing.MessagePassing   (gcbf/nn/gnn.py:4)
* ...nn.aggr.attention.AttentionalAggregation (gcbf/nn/gnn.py:7)
* ...utils.{softmax,mask_to_index,index_to_mask,to_networkx}
* ...transforms.radius_graph.RadiusGraph (gcbf/env/simple_car.py:8)
* torch_sparse.SparseTensor             (type annotation only)
* cvxpy.{Variable,Expression}           (env/base.py:8, overload types)
* pybullet / pybullet_data / cv2        (demo/video paths; stubs raise on use)
* seaborn.color_palette                 (trainer/utils.py:278)
* torch.utils.tensorboard.SummaryWriter (in-memory scalar collector)
"""
from __future__ import annotations

import inspect
import sys
import types
from typing import List, Optional

import torch
import torch.nn as nn
from torch import Tensor


# --------------------------------------------------------------------- Data
class Data:
    """PyG-like attribute container (torch_geometric.data.Data)."""

    def __init__(self, **kwargs):
        for k, v in kwargs.items():
            setattr(self, k, v)

    def update(self, other):
        items = other.__dict__.items() if isinstance(other, Data) \
            else dict(other).items()
        for k, v in items:
            setattr(self, k, v)
        return self

    @property
    def num_nodes(self) -> int:
        for key in ("x", "pos", "states"):
            v = getattr(self, key, None)
            if v is not None:
                return v.shape[0]
        return 0

    @property
    def num_edges(self) -> int:
        ei = getattr(self, "edge_index", None)
        return 0 if ei is None else ei.shape[1]

    def to(self, device):
        for k, v in self.__dict__.items():
            if torch.is_tensor(v):
                setattr(self, k, v.to(device))
        return self

    def clone(self):
        out = self.__class__()
        for k, v in self.__dict__.items():
            setattr(out, k, v.clone() if torch.is_tensor(v) else v)
        return out


# node-count-sliced attributes vs edge-sliced; anything else splits by its
# own first dimension proportionally (PyG slices each attr independently)
_EDGE_ATTRS = ("edge_attr",)


class Batch(Data):
    """Block-diagonal concatenation with to_data_list support."""

    @classmethod
    def from_data_list(cls, data_list: List[Data]) -> "Batch":
        assert len(data_list) > 0
        out = cls()
        keys = [k for k in data_list[0].__dict__
                if not k.startswith("_")]
        node_counts = [g.num_nodes for g in data_list]
        device = None
        for k in keys:
            vals = [getattr(g, k) for g in data_list]
            if torch.is_tensor(vals[0]):
                device = vals[0].device
                if k == "edge_index":
                    offs, total = [], 0
                    for g, nc in zip(data_list, node_counts):
                        offs.append(g.edge_index + total)
                        total += nc
                    setattr(out, k, torch.cat(offs, dim=1))
                else:
                    setattr(out, k, torch.cat(vals, dim=0))
            else:
                setattr(out, k, vals[0])
        out._slices = {
            k: [getattr(g, k).shape[1 if k == "edge_index" else 0]
                if torch.is_tensor(getattr(g, k)) else 0
                for g in data_list]
            for k in keys}
        out._node_counts = node_counts
        out._keys = keys
        out._num_graphs = len(data_list)
        # batch vector (node -> graph id), PyG's `batch` attribute
        out.batch = torch.repeat_interleave(
            torch.arange(len(data_list), device=device),
            torch.tensor(node_counts, device=device))
        return out

    def to_data_list(self) -> List[Data]:
        n = self._num_graphs
        out = [Data() for _ in range(n)]
        node_offsets = [0]
        for c in self._node_counts:
            node_offsets.append(node_offsets[-1] + c)
        for k in self._keys:
            v = getattr(self, k, None)
            if not torch.is_tensor(v):
                for g in out:
                    setattr(g, k, v)
                continue
            sizes = self._slices[k]
            if k == "edge_index":
                pos = 0
                for gi in range(n):
                    e = v[:, pos:pos + sizes[gi]] - node_offsets[gi]
                    setattr(out[gi], k, e)
                    pos += sizes[gi]
            else:
                pos = 0
                for gi in range(n):
                    setattr(out[gi], k, v[pos:pos + sizes[gi]])
                    pos += sizes[gi]
        return out


# ------------------------------------------------------------------- utils
def softmax(src: Tensor, index: Optional[Tensor] = None,
            ptr: Optional[Tensor] = None, num_nodes: Optional[int] = None,
            dim: int = 0) -> Tensor:
    """Segment softmax over `index` groups (torch_geometric.utils.softmax)."""
    d = dim if dim >= 0 else src.dim() + dim
    assert d == 0, f"shim softmax supports the edge dim only (got dim={dim})"
    N = num_nodes if num_nodes is not None else (
        int(index.max()) + 1 if index.numel() else 0)
    if N == 0 or src.shape[0] == 0:
        return src.exp()
    idx = index.view(-1, *([1] * (src.dim() - 1))).expand_as(src)
    smax = src.new_full((N,) + src.shape[1:], float("-inf"))
    smax.scatter_reduce_(0, idx, src.detach(), reduce="amax",
                         include_self=True)
    out = (src - smax.index_select(0, index)).exp()
    ssum = src.new_zeros((N,) + src.shape[1:])
    ssum.index_add_(0, index, out)
    return out / (ssum.index_select(0, index) + 1e-16)


def mask_to_index(mask: Tensor) -> Tensor:
    return mask.nonzero(as_tuple=False).view(-1)


def index_to_mask(index: Tensor, size: Optional[int] = None) -> Tensor:
    size = int(index.max()) + 1 if size is None else size
    mask = torch.zeros(size, dtype=torch.bool, device=index.device)
    mask[index] = True
    return mask


def to_networkx(data: Data, to_undirected: bool = False):
    import networkx as nx
    G = nx.Graph() if to_undirected else nx.DiGraph()
    G.add_nodes_from(range(data.num_nodes))
    ei = getattr(data, "edge_index", None)
    if ei is not None and ei.numel():
        G.add_edges_from(ei.t().cpu().numpy().tolist())
    return G


# ----------------------------------------------------------- radius graph
def _radius(pos: Tensor, r: float, max_num_neighbors: int) -> Tensor:
    """torch_cluster.radius(x=pos, y=pos): for each query (center) find up
    to max_num_neighbors points within r, first-in-index-order (the CPU
    scan order of torch_cluster).  Returns [center, neighbor] rows."""
    dist = torch.cdist(pos, pos)
    within = dist <= r
    if max_num_neighbors < pos.shape[0]:
        rank = within.long().cumsum(dim=1)
        within = within & (rank <= max_num_neighbors)
    center, neighbor = within.nonzero(as_tuple=True)
    return torch.stack([center, neighbor], dim=0)


def radius_graph(pos: Tensor, r: float, batch=None, loop: bool = False,
                 max_num_neighbors: int = 32,
                 flow: str = "source_to_target") -> Tensor:
    assert batch is None, "shim radius_graph: single graph only"
    edge_index = _radius(pos, r,
                         max_num_neighbors if loop else max_num_neighbors + 1)
    if flow == "source_to_target":
        row, col = edge_index[1], edge_index[0]   # (source j, target i)
    else:
        row, col = edge_index[0], edge_index[1]
    if not loop:
        keep = row != col
        row, col = row[keep], col[keep]
    return torch.stack([row, col], dim=0)


class RadiusGraph:
    """torch_geometric.transforms.RadiusGraph over data.pos."""

    def __init__(self, r: float, loop: bool = False,
                 max_num_neighbors: int = 32,
                 flow: str = "source_to_target"):
        self.r = r
        self.loop = loop
        self.max_num_neighbors = max_num_neighbors
        self.flow = flow

    def __call__(self, data: Data) -> Data:
        data.edge_index = radius_graph(
            data.pos, self.r, None, self.loop, self.max_num_neighbors,
            self.flow)
        return data


# --------------------------------------------------------- message passing
class _Inspector:
    def __init__(self, owner):
        self.owner = owner

    def distribute(self, name: str, coll: dict) -> dict:
        fn = getattr(self.owner, name)
        return {k: coll[k] for k in inspect.signature(fn).parameters
                if k in coll}


class MessagePassing(nn.Module):
    """Gather → message → aggregate → update, PyG flow source_to_target:
    edge_index[0]=source j, edge_index[1]=target i; aggregation at i."""

    def __init__(self, aggr="add", **kwargs):
        super().__init__()
        if isinstance(aggr, nn.Module):
            self.aggr_module = aggr
            self.aggr = None
        else:
            self.aggr_module = None
            self.aggr = aggr
        self.inspector = _Inspector(self)
        self._user_args = list(
            inspect.signature(self.message).parameters.keys())

    def _check_input(self, edge_index, size):
        return [None, None]

    def _collect(self, user_args, edge_index, size, kwargs) -> dict:
        src, dst = edge_index[0], edge_index[1]
        coll = {}
        for arg in user_args:
            if arg.endswith("_i") or arg.endswith("_j"):
                val = kwargs.get(arg[:-2])
                coll[arg] = None if val is None else val.index_select(
                    0, dst if arg.endswith("_i") else src)
            elif arg in kwargs:
                coll[arg] = kwargs[arg]
        for k, v in kwargs.items():
            coll.setdefault(k, v)
        x = kwargs.get("x")
        dim_size = None
        if isinstance(size, (list, tuple)) and len(size) > 1:
            dim_size = size[1]
        if dim_size is None:
            dim_size = x.shape[0] if x is not None else (
                int(dst.max()) + 1 if dst.numel() else 0)
        coll["index"] = dst
        coll["ptr"] = None
        coll["dim_size"] = dim_size
        return coll

    def propagate(self, edge_index, size=None, **kwargs):
        size = self._check_input(edge_index, size)
        coll = self._collect(self._user_args, edge_index, size, kwargs)
        msg = self.message(**self.inspector.distribute("message", coll))
        out = self.aggregate(msg, index=coll["index"], ptr=None,
                             dim_size=coll["dim_size"])
        upd = self.inspector.distribute("update", coll)
        upd.pop("aggr_out", None)
        return self.update(out, **upd)

    def aggregate(self, inputs, index, ptr=None, dim_size=None):
        if self.aggr_module is not None:
            return self.aggr_module(inputs, index=index, ptr=ptr,
                                    dim_size=dim_size)
        out = inputs.new_zeros((dim_size,) + inputs.shape[1:])
        if self.aggr == "add":
            return out.index_add_(0, index, inputs)
        if self.aggr == "max":
            idx = index.view(-1, *([1] * (inputs.dim() - 1))).expand_as(
                inputs)
            # empty segments stay 0 (torch_scatter.scatter_max fill value)
            out.scatter_reduce_(0, idx, inputs, reduce="amax",
                                include_self=False)
            return out
        raise NotImplementedError(f"aggr={self.aggr!r}")

    def message(self, x_j):  # overridden by subclasses
        return x_j

    def update(self, aggr_out):
        return aggr_out


class AttentionalAggregation(nn.Module):
    """softmax(gate_nn(x)) per target segment, then weighted scatter-sum."""

    def __init__(self, gate_nn: nn.Module, nn_mod: Optional[nn.Module] = None):
        super().__init__()
        self.gate_nn = gate_nn
        self.nn = nn_mod

    def forward(self, x, index=None, ptr=None, dim_size=None, dim=-2):
        gate = self.gate_nn(x)
        if self.nn is not None:
            x = self.nn(x)
        alpha = softmax(gate, index, ptr, dim_size, dim=0)
        out = x.new_zeros((dim_size,) + x.shape[1:])
        return out.index_add_(0, index, alpha * x)


class Sequential(nn.Module):
    """torch_geometric.nn.Sequential('x, edge_attr, edge_index', [...])
    with the module_0.. child naming the reference state dicts rely on."""

    def __init__(self, input_args: str, modules: list):
        super().__init__()
        self._input_args = [a.strip() for a in input_args.split(",")]
        self._descs = []
        for i, entry in enumerate(modules):
            mod, desc = entry if isinstance(entry, (tuple, list)) \
                else (entry, None)
            setattr(self, f"module_{i}", mod)
            self._descs.append(desc)

    def forward(self, *args):
        scope = dict(zip(self._input_args, args))
        out = None
        for i, desc in enumerate(self._descs):
            mod = getattr(self, f"module_{i}")
            if desc is None:
                out = mod(out)
                continue
            ins, outs = desc.split("->")
            res = mod(*[scope[a.strip()] for a in ins.split(",")])
            out_names = [a.strip() for a in outs.split(",")]
            if len(out_names) == 1:
                scope[out_names[0]] = res
            else:
                for nm, v in zip(out_names, res):
                    scope[nm] = v
            out = res
        return out


# ------------------------------------------------------------ small stubs
class Expression:   # cvxpy.Expression (isinstance checks only)
    pass


class Variable(Expression):  # cvxpy.Variable
    def __init__(self, *a, **k):
        pass


class SummaryWriter:
    """In-memory stand-in for torch.utils.tensorboard.SummaryWriter."""

    def __init__(self, log_dir=None, *a, **k):
        self.log_dir = log_dir
        self.scalars = []

    def add_scalar(self, tag, value, step=None):
        self.scalars.append((tag, float(value), step))

    def flush(self):
        pass

    def close(self):
        pass


def _raising_module(name: str, reason: str) -> types.ModuleType:
    mod = types.ModuleType(name)

    def _getattr(attr, _name=name, _reason=reason):
        if attr.startswith("__") and attr.endswith("__"):
            raise AttributeError(attr)   # introspection (inspect, pickling)
        raise RuntimeError(
            f"{_name} stub: {_reason} (attribute {attr!r} requested)")

    mod.__getattr__ = _getattr
    return mod


# ----------------------------------------------------------------- install
def install():
    """Register the shim under the reference's import names."""
    tg = types.ModuleType("torch_geometric")
    tg_data = types.ModuleType("torch_geometric.data")
    tg_data.Data = Data
    tg_data.Batch = Batch
    tg_nn = types.ModuleType("torch_geometric.nn")
    tg_nn.Sequential = Sequential
    tg_nn_conv = types.ModuleType("torch_geometric.nn.conv")
    tg_nn_conv_mp = types.ModuleType("torch_geometric.nn.conv.message_passing")
    tg_nn_conv_mp.MessagePassing = tg_nn_conv.MessagePassing = \
        tg_nn.MessagePassing = MessagePassing
    tg_nn_aggr = types.ModuleType("torch_geometric.nn.aggr")
    tg_nn_aggr_att = types.ModuleType("torch_geometric.nn.aggr.attention")
    tg_nn_aggr_att.AttentionalAggregation = tg_nn_aggr.AttentionalAggregation \
        = AttentionalAggregation
    tg_utils = types.ModuleType("torch_geometric.utils")
    tg_utils.softmax = softmax
    tg_utils.mask_to_index = mask_to_index
    tg_utils.index_to_mask = index_to_mask
    tg_utils.to_networkx = to_networkx
    tg_tr = types.ModuleType("torch_geometric.transforms")
    tg_tr_rg = types.ModuleType("torch_geometric.transforms.radius_graph")
    tg_tr_rg.RadiusGraph = tg_tr.RadiusGraph = RadiusGraph
    tg.data = tg_data
    tg.nn = tg_nn
    tg.utils = tg_utils
    tg.transforms = tg_tr

    ts = types.ModuleType("torch_sparse")

    class SparseTensor:  # annotation-only in the reference
        pass

    ts.SparseTensor = SparseTensor

    cp = types.ModuleType("cvxpy")
    cp.Expression = Expression
    cp.Variable = Variable

    sns = types.ModuleType("seaborn")

    def color_palette(name="rocket", as_cmap=False, *a, **k):
        import matplotlib.pyplot as plt
        return plt.get_cmap("magma") if as_cmap else None

    sns.color_palette = color_palette

    tb = types.ModuleType("torch.utils.tensorboard")
    tb.SummaryWriter = SummaryWriter

    mods = {
        "torch_geometric": tg,
        "torch_geometric.data": tg_data,
        "torch_geometric.nn": tg_nn,
        "torch_geometric.nn.conv": tg_nn_conv,
        "torch_geometric.nn.conv.message_passing": tg_nn_conv_mp,
        "torch_geometric.nn.aggr": tg_nn_aggr,
        "torch_geometric.nn.aggr.attention": tg_nn_aggr_att,
        "torch_geometric.utils": tg_utils,
        "torch_geometric.transforms": tg_tr,
        "torch_geometric.transforms.radius_graph": tg_tr_rg,
        "torch_sparse": ts,
        "cvxpy": cp,
        "seaborn": sns,
        "torch.utils.tensorboard": tb,
        "pybullet": _raising_module(
            "pybullet", "demo modes unavailable in the baseline harness"),
        "pybullet_data": _raising_module(
            "pybullet_data", "demo modes unavailable"),
    }
    try:
        import cv2  # noqa: F401
    except ImportError:
        mods["cv2"] = _raising_module(
            "cv2", "video export unavailable in the baseline harness")
    sys.modules.update(mods)
