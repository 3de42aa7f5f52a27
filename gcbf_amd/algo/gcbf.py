"""GCBF: jointly learned graph CBF and GNN policy.

Behavioral equivalent of the reference GCBF algorithm (gcbf/algo/gcbf.py):
CBF GNN + actor GNN, Adam optimizers, replay buffers with balanced
safe/unsafe segment sampling, CBF-condition losses with the ḣ re-link
residue trick, and test-time per-agent action refinement.

MI355X redesign points:
* the ḣ re-link step rebuilds communication links for the whole batch in one
  batched kernel (``env.add_communication_links_batched``) instead of the
  reference's Python loop over ~300 graphs (gcbf/algo/gcbf.py:196-200);
* safety masks run batched (no ``to_data_list`` loops);
* scalar logging is deferred and synced once per update, not per item;
* a ``grad_sync`` hook lets the DP layer all-reduce gradients over
  RCCL/xGMI between backward and the optimizer steps.
"""
from __future__ import annotations

import os
from typing import Callable, Optional

import numpy as np
import torch
import torch.nn as nn
from torch import Tensor
from torch.optim import Adam

from ..controller import GNNController
from ..env import MultiAgentEnv
from ..graph import GraphBatch
from ..nn import MLP, CBFGNNLayer
from ..utils.trace import trace_range
from .base import Algorithm
from .buffer import Buffer


class _Seq(nn.Module):
    """Name-compat container mirroring PyG ``Sequential``'s ``module_0``."""

    def __init__(self, layer: nn.Module):
        super().__init__()
        self.module_0 = layer


class CBFGNN(nn.Module):
    """CBF value network h(x) per agent (reference gcbf/algo/gcbf.py:21-61)."""

    def __init__(self, num_agents: int, node_dim: int, edge_dim: int,
                 phi_dim: int):
        super().__init__()
        self.num_agents = num_agents
        self.feat_transformer = _Seq(CBFGNNLayer(
            node_dim=node_dim, edge_dim=edge_dim, output_dim=1024,
            phi_dim=phi_dim))
        self.feat_2_CBF = MLP(in_channels=1024, out_channels=1,
                              hidden_layers=(512, 128, 32),
                              output_activation=nn.Tanh())

    def forward(self, data: GraphBatch) -> Tensor:
        nm = data.agent_mask
        if data.agent_index is not None:
            nm = data.agent_index      # static LONG indices (capture-safe)
        elif data.agents_first_n is not None:
            nm = data.agents_first_n
        x = self.feat_transformer.module_0(
            data.x, data.edge_attr, data.edge_index,
            node_mask=nm, seg_dst=data.seg_dst)
        return self.feat_2_CBF(x)

    def attention(self, data: GraphBatch) -> Tensor:
        return self.feat_transformer.module_0.attention(data)


class GCBF(Algorithm):

    def __init__(self, env: MultiAgentEnv, num_agents: int, node_dim: int,
                 edge_dim: int, action_dim: int, device: torch.device,
                 batch_size: int = 500, params: Optional[dict] = None):
        super().__init__(env=env, num_agents=num_agents, node_dim=node_dim,
                         edge_dim=edge_dim, action_dim=action_dim,
                         device=device)
        self.cbf = CBFGNN(num_agents=num_agents, node_dim=node_dim,
                          edge_dim=edge_dim, phi_dim=256).to(device)
        self.actor = GNNController(num_agents=num_agents, node_dim=node_dim,
                                   edge_dim=edge_dim, phi_dim=256,
                                   action_dim=action_dim).to(device)

        # fused multi-tensor Adam on GPU: one kernel per step instead of a
        # foreach chain (~10 launches/step measured in rocprof r02);
        # numerics identical (fp32 master weights)
        fused = (device.type == "cuda"
                 and os.environ.get("GCBF_AMD_FUSED_ADAM", "1") == "1")
        try:
            self.optim_cbf = Adam(self.cbf.parameters(), lr=3e-4,
                                  fused=fused)
            self.optim_actor = Adam(self.actor.parameters(), lr=1e-3,
                                    fused=fused)
        except (RuntimeError, ValueError):
            self.optim_cbf = Adam(self.cbf.parameters(), lr=3e-4)
            self.optim_actor = Adam(self.actor.parameters(), lr=1e-3)

        self.buffer = Buffer()   # current-episode buffer
        self.memory = Buffer()   # replay memory
        self.batch_size = batch_size

        if params is None:
            params = {  # defaults (reference gcbf/algo/gcbf.py:112-120)
                "alpha": 1.0,
                "eps": 0.02,
                "inner_iter": 10,
                "loss_action_coef": 0.001,
                "loss_unsafe_coef": 1.0,
                "loss_safe_coef": 1.0,
                "loss_h_dot_coef": 0.1,
            }
        self.params = params

        # DP hook: called after backward, before the optimizer steps
        self.grad_sync: Optional[Callable[[], None]] = None

        # device ring of replayed states (fast batched re-batching for the
        # eager update path; also backs the captured update engine)
        self._ring = None
        self._ring_tried = False
        # hipGraph-captured update engine (created lazily at first update
        # when supported); None -> eager inner iterations
        self._upd_engine = None
        self._upd_engine_tried = False

    # ---------------------------------------------------------------- acting
    @torch.no_grad()
    def act(self, data: GraphBatch) -> Tensor:
        return self.actor(data)

    @torch.no_grad()
    def step(self, data: GraphBatch, prob: float) -> Tensor:
        action = self.actor(data)
        if np.random.rand() < prob:
            action = torch.zeros_like(action)
        # is_safe stays a device tensor; the buffer classifies lazily in one
        # batched transfer (no per-step host sync)
        is_safe = torch.logical_not(torch.any(self._env.unsafe_mask(data)))
        self.buffer.append(data, is_safe)
        return action

    def is_update(self, step: int) -> bool:
        return step % self.batch_size == 0

    # -------------------------------------------------------------- training
    def _make_ring(self):
        """Device ring + batched exact rebuild for sampled batches (replaces
        per-graph Python concatenation in the update; see gcbf_amd/ring.py).
        """
        self._ring_tried = True
        if os.environ.get("GCBF_AMD_RING", "1") == "0":
            return
        env = self._env
        if (env.data is None
                or getattr(env, "_max_neighbors", None) is not None):
            return
        try:
            from ..ring import RingStore
            ring = RingStore(env, Buffer.MAX_SIZE + 2 * self.batch_size)
            for g in list(self.buffer.data) + list(self.memory.data):
                ring.push(g)
            self.buffer.on_append = ring.push
            self._ring = ring
        except Exception as e:
            import warnings
            warnings.warn(f"ring batcher unavailable ({e})")
            self._ring = None

    def _make_update_engine(self):
        """Try to build the hipGraph-captured update engine (GPU only).

        Opt-in via GCBF_AMD_UPDATE_CAPTURE=1.  r02 status: gradient parity
        vs eager is VALIDATED on hardware (tests/test_gpu_kernels.py
        update-engine tests), but the engine measures SLOWER than the
        improved eager path (0.17-0.18 s vs 0.13-0.14 s per update) —
        its fixed-capacity padding and duplicated actor forward cost more
        GPU time than the launch gaps it saves (docs/ARCHITECTURE.md,
        "Captured update engine: measured decision").  Default stays
        eager on measurement, not correctness."""
        self._upd_engine_tried = True
        if os.environ.get("GCBF_AMD_UPDATE_CAPTURE", "0") != "1":
            return
        if type(self) is not GCBF:
            return
        if self.device.type != "cuda":
            return
        env = self._env
        data = env.data
        # obstacle nodes are supported via the static agent-index layout
        # (agents are the first n rows of every graph); the MACBF
        # neighbor-cap path stays eager
        if (data is None
                or getattr(env, "_max_neighbors", None) is not None):
            return
        try:
            from ..update_engine import UpdateEngine
            self._upd_engine = UpdateEngine(self, env)
        except Exception as e:  # fall back to eager iterations
            import traceback
            import warnings
            if os.environ.get("GCBF_AMD_UPDATE_CAPTURE_DEBUG") == "1":
                traceback.print_exc()
            warnings.warn(f"update capture unavailable ({e}); eager updates")
            self._upd_engine = None

    def update(self, step: int, writer=None) -> dict:
        seg_len = 3
        inner_iter = self.params["inner_iter"]
        logs = []  # deferred scalars, synced once at the end

        prof = None
        if os.environ.get("GCBF_AMD_UPDATE_PROF") == "1":
            import time as _time
            prof = {"sample": 0.0, "batch": 0.0, "fwd": 0.0, "mask": 0.0,
                    "hdot": 0.0, "bwd": 0.0, "opt": 0.0, "engine": 0.0}

            def _tick():
                if self.device.type == "cuda":
                    torch.cuda.synchronize()
                return _time.perf_counter()
        else:
            _tick = None

        if self._ring is None and not self._ring_tried:
            self._make_ring()
        if self._upd_engine is None and not self._upd_engine_tried:
            self._make_update_engine()

        for i_inner in range(inner_iter):
            t0 = _tick() if prof else 0
            # sample segments from the current buffer and the replay memory
            if self.memory.size == 0:
                graph_list = self.buffer.sample(self.batch_size // 5, seg_len)
            else:
                curr = self.buffer.sample(self.batch_size // 10, seg_len, True)
                prev = self.memory.sample(
                    self.batch_size // 5 - self.batch_size // 10, seg_len, True)
                graph_list = curr + prev
            if not graph_list:
                # degenerate batch_size (< 10): the reference's balanced
                # quotas (n//2 draws) collapse to zero and it crashes at
                # Batch.from_data_list([]); sample unbalanced instead
                graph_list = self.buffer.sample(
                    max(1, self.batch_size // 5), seg_len)
            if prof:
                t1 = _tick(); prof["sample"] += t1 - t0; t0 = t1

            log7 = None
            if self._upd_engine is not None:
                log7 = self._upd_engine.try_iter(graph_list)
                if prof and log7 is not None:
                    t1 = _tick(); prof["engine"] += t1 - t0; t0 = t1
            if log7 is None:
                try:
                    log7 = self._iter_eager(graph_list, prof, _tick)
                except Exception:
                    if getattr(self, "_sn_ctx", None) is not None:
                        self._sn_ctx.__exit__()
                        self._sn_ctx = None
                    raise
            logs.append(log7)

        if prof:
            import sys
            print("# update prof: " + " ".join(
                f"{k}={v * 1000:.1f}ms" for k, v in prof.items()),
                file=sys.stderr, flush=True)

        return self._update_tail(step, writer, logs, inner_iter)

    def _iter_eager(self, graph_list, prof=None, _tick=None) -> Tensor:
        """One eager inner iteration (sampled batch -> losses -> backward ->
        optimizer).  Returns the 7-scalar log stack (device tensor)."""
        eps = self.params["eps"]
        alpha = self.params["alpha"]
        t0 = _tick() if prof else 0

        graphs = None
        if self._ring is not None and self._ring.usable(graph_list):
            try:
                graphs = self._ring.batch(graph_list)
            except Exception:
                # metadata-only snapshots (ring-backed rollout) cannot fall
                # back to from_list — there are no stored tensors to
                # concatenate; fail loudly rather than mix gradients
                if any(g.states is None for g in graph_list):
                    raise
                import warnings
                warnings.warn("ring batch failed; falling back to from_list")
                self._ring = None
        if graphs is None:
            graphs = GraphBatch.from_list(graph_list)
        graphs.edge_attr.requires_grad_(True)
        if prof:
            t1 = _tick(); prof["batch"] += t1 - t0; t0 = t1
        from ..nn.mlp import sn_weight_reuse
        self._sn_ctx = sn_weight_reuse().__enter__()
        with trace_range("gcbf/forward"):
            actions = self.actor(graphs)
            # h and h_next in ONE doubled-batch CBF forward: halves the
            # CBF forward/backward chains vs. the reference's separate
            # calls (gcbf/algo/gcbf.py:161,194) and evaluates both sides
            # of the finite difference under the SAME spectral-norm σ
            graphs_next = self._env.forward_graph(graphs, actions)
            both = GraphBatch.from_list([graphs, graphs_next])
            h_both = self.cbf(both)
            n_ag = h_both.shape[0] // 2
            h, h_next = h_both[:n_ag], h_both[n_ag:]
        if prof:
            t1 = _tick(); prof["fwd"] += t1 - t0; t0 = t1

        # unsafe region: h < 0 (reference gcbf/algo/gcbf.py:167-177).
        # Masked means are computed as weighted sums: boolean-mask
        # indexing (h[mask]) calls nonzero and forces a device→host
        # sync per mask per inner iteration; the weighted form is
        # mathematically identical (incl. the empty-mask fallbacks of
        # loss 0 / acc 1) and stays on device.
        unsafe_mask = self._env.unsafe_mask(graphs)
        hv = h[:, 0]
        wu = unsafe_mask.to(hv.dtype)
        cu = wu.sum()
        cu1 = cu.clamp(min=1)
        any_u = (cu > 0).to(hv.dtype)
        loss_unsafe = any_u * (torch.relu(hv + eps) * wu).sum() / cu1
        acc_unsafe = (any_u * ((hv < 0).to(hv.dtype) * wu).sum() / cu1
                      + (1 - any_u))

        # safe region: h > 0
        safe_mask = self._env.safe_mask(graphs)
        ws = safe_mask.to(hv.dtype)
        cs = ws.sum()
        cs1 = cs.clamp(min=1)
        any_s = (cs > 0).to(hv.dtype)
        loss_safe = any_s * (torch.relu(-hv + eps) * ws).sum() / cs1
        acc_safe = (any_s * ((hv >= 0).to(hv.dtype) * ws).sum() / cs1
                    + (1 - any_s))

        if prof:
            t1 = _tick(); prof["mask"] += t1 - t0; t0 = t1
        # ḣ condition with the re-link residue trick
        # (reference gcbf/algo/gcbf.py:191-209): the VALUE reflects the
        # re-linked next graph, the GRADIENT flows through the
        # fixed-topology path.
        with trace_range("gcbf/h_dot"):
            with torch.no_grad():
                relinked = self._env.add_communication_links_batched(
                    graphs_next.detach())
                h_next_new_link = self.cbf(relinked)
        # σ reuse stops here: the optimizer step below changes the weights
        self._sn_ctx.__exit__()
        self._sn_ctx = None
        h_dot = (h_next - h) / self._env.dt
        h_dot_new_link = (h_next_new_link - h) / self._env.dt
        residue = (h_dot_new_link - h_dot).detach()
        h_dot = residue + h_dot

        if prof:
            t1 = _tick(); prof["hdot"] += t1 - t0; t0 = t1
        loss_h_dot = torch.mean(torch.relu(-h_dot - alpha * h + eps))
        acc_h_dot = torch.mean(
            torch.greater_equal(h_dot + alpha * h, 0).type_as(h_dot))

        loss_action = torch.mean(torch.square(actions).sum(dim=1))

        loss = (self.params["loss_unsafe_coef"] * loss_unsafe +
                self.params["loss_safe_coef"] * loss_safe +
                self.params["loss_h_dot_coef"] * loss_h_dot +
                self.params["loss_action_coef"] * loss_action)

        self.optim_cbf.zero_grad(set_to_none=True)
        self.optim_actor.zero_grad(set_to_none=True)
        with trace_range("gcbf/backward"):
            loss.backward()
        if self.grad_sync is not None:
            with trace_range("gcbf/grad_allreduce"):
                self.grad_sync()
        if prof:
            t1 = _tick(); prof["bwd"] += t1 - t0; t0 = t1
        with trace_range("gcbf/optim"):
            torch.nn.utils.clip_grad_norm_(self.cbf.parameters(), 1e-3)
            torch.nn.utils.clip_grad_norm_(self.actor.parameters(), 1e-3)
            self.optim_cbf.step()
            self.optim_actor.step()
            # bf16 mirror refresh must follow EVERY step: fused Adam does
            # not bump version counters, so lazy staleness checks miss it
            from ..nn.fused import sync_bf16_mirrors
            sync_bf16_mirrors(self.cbf)
            sync_bf16_mirrors(self.actor)

        if prof:
            t1 = _tick(); prof["opt"] += t1 - t0
        return torch.stack([
            loss_unsafe.detach(), loss_safe.detach(),
            loss_h_dot.detach(), loss_action.detach(),
            acc_unsafe.detach(), acc_safe.detach(), acc_h_dot.detach()])

    def _update_tail(self, step, writer, logs, inner_iter) -> dict:
        # one host sync for the whole update's scalars; under DP the stack
        # is mean-reduced across ranks first, so rank-0 TensorBoard curves
        # reflect the GLOBAL batch (one small collective per update, not
        # one per scalar; reference semantics gcbf/algo/gcbf.py:229-237)
        log_stack = torch.stack(logs).detach()
        import torch.distributed as dist
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(log_stack, op=dist.ReduceOp.SUM)
            log_stack /= dist.get_world_size()
        log_vals = log_stack.cpu()
        if writer is not None:
            names = ("loss/unsafe", "loss/safe", "loss/derivative",
                     "loss/action", "acc/unsafe", "acc/safe", "acc/derivative")
            for i_inner in range(inner_iter):
                t = step * inner_iter + i_inner
                for k, name in enumerate(names):
                    writer.add_scalar(name, float(log_vals[i_inner, k]), t)

        # refresh bf16 weight mirrors (captured rollout graphs read them
        # by address; Python is skipped during replay)
        from ..nn.fused import sync_bf16_mirrors
        sync_bf16_mirrors(self.actor)
        sync_bf16_mirrors(self.cbf)

        self.memory.merge(self.buffer)
        self.buffer.clear()

        return {
            "acc/safe": float(log_vals[-1, 5]),
            "acc/unsafe": float(log_vals[-1, 4]),
            "acc/derivative": float(log_vals[-1, 6]),
        }

    # ----------------------------------------------------------- checkpoints
    def save(self, save_dir: str):
        os.makedirs(save_dir, exist_ok=True)
        torch.save(self.cbf.state_dict(), os.path.join(save_dir, "cbf.pkl"))
        torch.save(self.actor.state_dict(),
                   os.path.join(save_dir, "actor.pkl"))

    def load(self, load_dir: str):
        assert os.path.exists(load_dir)
        self.cbf.load_state_dict(torch.load(
            os.path.join(load_dir, "cbf.pkl"), map_location=self.device,
            weights_only=True))
        self.actor.load_state_dict(torch.load(
            os.path.join(load_dir, "actor.pkl"), map_location=self.device,
            weights_only=True))

    def extra_state(self) -> dict:
        """Optimizer state for training resume (an addition over the
        reference, stored in a separate file so the checkpoint layout stays
        compatible)."""
        return {
            "optim_cbf": self.optim_cbf.state_dict(),
            "optim_actor": self.optim_actor.state_dict(),
        }

    def load_extra_state(self, state: dict):
        self.optim_cbf.load_state_dict(state["optim_cbf"])
        self.optim_actor.load_state_dict(state["optim_actor"])

    # ------------------------------------------------------------- test time
    def apply(self, data: GraphBatch, rand: Optional[float] = 30) -> Tensor:
        """Test-time refinement (reference gcbf/algo/gcbf.py:260-309):
        agents already satisfying the ḣ condition under the nominal (zero)
        action fall back to it; the rest run up to 30 rounds of per-agent
        Adam descent on the ḣ violation through the frozen CBF, plus
        exploration noise."""
        lr = 0.1
        h = self.cbf(data).detach()
        action = self.actor(data).detach()
        nominal = torch.zeros_like(action)

        data_next = self._env.forward_graph(data, nominal)
        h_next = self.cbf(data_next)
        h_dot = (h_next - h) / self._env.dt
        max_val_h_dot = torch.relu(-h_dot - self.params["alpha"] * h)

        ok = (max_val_h_dot[:, 0] <= 0)
        action = torch.where(ok.unsqueeze(1), nominal, action)

        # Vectorized per-agent Adam (equivalent to the reference's list of
        # per-agent optimizers, gcbf/algo/gcbf.py:275-307: Adam state and
        # step counts advance only for violating agents; gradients keep
        # accumulating for agents that are not stepped).
        action = action.clone().requires_grad_(True)
        exp_avg = torch.zeros_like(action)
        exp_avg_sq = torch.zeros_like(action)
        step_cnt = torch.zeros(action.shape[0], 1, device=action.device)
        grad_buf = torch.zeros_like(action)
        b1, b2, adam_eps = 0.9, 0.999, 1e-8

        i_iter = 0
        max_iter = 30
        while True:
            data_next = self._env.forward_graph(data, action)
            h_next = self.cbf(data_next)
            h_dot = (h_next - h) / self._env.dt
            max_val_h_dot = torch.relu(-h_dot - self.params["alpha"] * h)
            loss_h_dot = torch.mean(max_val_h_dot)
            if loss_h_dot <= 0 or i_iter > max_iter:
                break
            val = (max_val_h_dot[:, 0] > 0).unsqueeze(1)
            g, = torch.autograd.grad(loss_h_dot, action)
            with torch.no_grad():
                grad_buf = torch.where(val, g, grad_buf + g)
                step_cnt = step_cnt + val
                exp_avg = torch.where(
                    val, b1 * exp_avg + (1 - b1) * grad_buf, exp_avg)
                exp_avg_sq = torch.where(
                    val, b2 * exp_avg_sq + (1 - b2) * grad_buf ** 2,
                    exp_avg_sq)
                bc1 = 1 - b1 ** step_cnt.clamp(min=1)
                bc2 = 1 - b2 ** step_cnt.clamp(min=1)
                upd = (lr / bc1) * exp_avg / (
                    (exp_avg_sq / bc2).sqrt() + adam_eps)
                noise = rand * lr * torch.randn_like(grad_buf) * grad_buf
                action -= torch.where(val, upd + noise,
                                      torch.zeros_like(upd))
            i_iter += 1

        return action.detach()
