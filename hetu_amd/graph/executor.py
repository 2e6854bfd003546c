"""Executable-plan executor.

Replaces the reference's ExecutableGraph run loop
(/root/reference/hetu/graph/executable_graph.cc:883,1756 ComputeFunc/Run) with
an MI355X-first design: torch-ROCm tensors, the torch caching allocator, HIP
streams via torch.cuda.Stream, and (optionally, from the engine) hipGraph
capture of the steady-state step. Plans (topo order + free schedule) are
cached per fetch-set, mirroring the reference's exec-graph plan pool.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import torch

from .op import Op
from .tensor import Tensor

import os

# gradient-allreduce / backward overlap on the comm stream (disable with
# HETU_AMD_COMM_OVERLAP=0 if a RCCL/capture combination misbehaves)
_COMM_OVERLAP = os.environ.get("HETU_AMD_COMM_OVERLAP", "1") == "1"
# per-op allocated-memory trace (debug): records the op at the peak
_MEM_TRACE = os.environ.get("HETU_AMD_MEM_TRACE", "0") == "1"
# per-op NaN/Inf guard (reference CheckNumeric/CheckFinite kernels,
# hetu/impl/kernel + gradscaler): raises at the FIRST op whose output goes
# non-finite, naming the op — turns a step-level NaN into a one-op bisect.
# Synchronizes per op, so debug-only; skipped inside hipGraph capture.
_CHECK_NUMERIC = os.environ.get("HETU_AMD_CHECK_NUMERIC", "0") == "1"


class NonFiniteError(RuntimeError):
    """Raised by HETU_AMD_CHECK_NUMERIC=1 when an op output contains
    NaN/Inf. .op_type/.op_name/.out_index locate the producing op."""

    def __init__(self, op_type: str, op_name: str, out_index: int,
                 n_nan: int, n_inf: int, shape):
        self.op_type, self.op_name = op_type, op_name
        self.out_index = out_index
        super().__init__(
            f"non-finite output {out_index} of {op_type} '{op_name}' "
            f"shape={tuple(shape)}: {n_nan} NaN, {n_inf} Inf")


def _check_numeric(op, outs):
    for j, v in enumerate(outs):
        if not isinstance(v, torch.Tensor) or not v.is_floating_point():
            continue
        if v.is_cuda and torch.cuda.is_current_stream_capturing():
            return
        fin = torch.isfinite(v)
        if not bool(fin.all()):
            bad = ~fin
            n_nan = int(torch.isnan(v).sum())
            n_inf = int(bad.sum()) - n_nan
            raise NonFiniteError(op.type, op.name, j, n_nan, n_inf, v.shape)


class ExecContext:
    """Per-run execution context handed to every op's compute()."""

    def __init__(self, device: Optional[torch.device] = None, comm=None,
                 training: bool = True):
        self.device = device or torch.device("cpu")
        self.comm = comm                  # parallel.comm.CommBackend or None
        self.training = training
        self.symbols = {}
        self.profiler = None              # utils.profiler.OpProfiler
        # stream roles (MI355X: overlap collectives with compute on separate
        # HIP streams; reference used a fixed 16-stream convention
        # hetu/core/stream.h:8-20 — we keep {compute, comm, p2p, h2d} roles)
        self.streams = {}
        if self.device.type == "cuda":
            self.streams = {
                "compute": torch.cuda.current_stream(self.device),
                "comm": torch.cuda.Stream(self.device),
                "p2p": torch.cuda.Stream(self.device),
                "h2d": torch.cuda.Stream(self.device),
            }

    def stream(self, role: str):
        return self.streams.get(role)


class _Plan:
    __slots__ = ("topo", "last_use", "fetch_ids", "dead_outputs")

    def __init__(self, topo: List[Op], last_use: Dict[int, int],
                 fetch_ids: List[int], dead_outputs: Dict[int, list]):
        self.topo = topo
        self.last_use = last_use   # tensor_id -> index of last consuming op
        self.fetch_ids = fetch_ids
        # op index -> output tensor ids with NO consumer in this plan
        # (e.g. recompute rewires backward-saved tensors to clones): freed
        # right after the producing op instead of living the whole run
        self.dead_outputs = dead_outputs


class Executor:
    def __init__(self, graph):
        self.graph = graph
        self._plan_pool: Dict = {}
        self.ctx: Optional[ExecContext] = None
        self._inflight: Dict[int, "torch.cuda.Event"] = {}
        self._mem_peak = (0, "", "", -1)

    def bind_context(self, ctx: ExecContext):
        self.ctx = ctx

    def _get_plan(self, fetches: Sequence[Tensor],
                  seed_ids: frozenset = frozenset()) -> _Plan:
        key = (tuple(t.id for t in fetches), len(self.graph.ops), seed_ids)
        plan = self._plan_pool.get(key)
        if plan is None:
            topo = self.graph.topo_sort(fetches)
            if seed_ids:
                # cut the graph at seeded tensors: ops only reachable
                # through them are dropped from the plan (the pipeline
                # backward and the split-capture optimizer graph seed the
                # cached forward/grad tensors)
                stack = [t.producer for t in fetches
                         if t.producer is not None and t.id not in seed_ids]
                seen_ops = set()
                while stack:
                    op = stack.pop()
                    if op.id in seen_ops:
                        continue
                    seen_ops.add(op.id)
                    for t in op.inputs:
                        if t.id not in seed_ids and t.producer is not None:
                            stack.append(t.producer)
                    stack.extend(op.in_deps)
                topo = [op for op in topo if op.id in seen_ops]
            fetch_ids = [t.id for t in fetches]
            last_use: Dict[int, int] = {}
            for i, op in enumerate(topo):
                for t in op.inputs:
                    last_use[t.id] = i
            # fetched tensors are never freed
            for tid in fetch_ids:
                last_use.pop(tid, None)
            dead: Dict[int, list] = {}
            fset = set(fetch_ids)
            for i, op in enumerate(topo):
                dd = [t.id for t in op.outputs
                      if t.id not in last_use and t.id not in fset]
                if dd:
                    dead[i] = dd
            plan = _Plan(topo, last_use, fetch_ids, dead)
            self._plan_pool[key] = plan
        return plan

    def run(self, fetches: Sequence[Tensor], feed_dict: Dict,
            ctx: Optional[ExecContext] = None,
            seed_values: Optional[Dict[int, torch.Tensor]] = None,
            keep_values: Optional[Dict[int, torch.Tensor]] = None
            ) -> List[torch.Tensor]:
        """seed_values: {tensor_id: value} of already-computed tensors (their
        producing ops are skipped) — the pipeline engine seeds the backward
        pass with the forward pass's cached activations.  keep_values: a
        dict the caller provides to capture every computed value (disables
        the degree-based free)."""
        ctx = ctx or self.ctx or ExecContext()
        plan = self._get_plan(
            fetches, frozenset(seed_values) if seed_values else frozenset())
        values: Dict[int, torch.Tensor] = dict(seed_values or {})

        # feed_dict keys may be Tensors or names
        feeds: Dict[int, torch.Tensor] = {}
        for k, v in feed_dict.items():
            t = k if isinstance(k, Tensor) else None
            if t is None:
                raise TypeError("feed_dict keys must be Tensors")
            if not isinstance(v, torch.Tensor):
                v = torch.as_tensor(v)
            feeds[t.id] = v

        for i, op in enumerate(plan.topo):
            if op.outputs and all(t.id in values for t in op.outputs):
                continue                     # seeded (already computed)
            ins = []
            for t in op.inputs:
                if t.id in values:
                    ins.append(values[t.id])
                elif t.id in feeds:
                    ins.append(feeds[t.id])
                elif t.get_data() is not None:   # variable / persistent
                    ins.append(t.get_data())
                else:
                    raise RuntimeError(
                        f"no value for input {t.name} of op {op.name}")
            if op.type == "Placeholder":
                tid = op.outputs[0].id
                if tid in feeds:
                    values[tid] = feeds[tid]
                    continue
                raise RuntimeError(f"placeholder {op.name} not fed")
            # gradient all-reduces ride the comm stream so they overlap
            # with the rest of backward (reference: grad-buffer bucket
            # reduction on the comm stream, executable_graph.cc:1756);
            # consumers wait via recorded events
            comm_async = (_COMM_OVERLAP and ctx.stream("comm") is not None
                          and op.name.startswith("grad_allreduce"))
            # any input still in flight on the comm stream: make the
            # compute stream wait before using it
            for t in op.inputs:
                ev = self._inflight.pop(t.id, None) if not comm_async \
                    else None
                if ev is not None:
                    cur = torch.cuda.current_stream(ctx.device)
                    cur.wait_event(ev)
                    v = values.get(t.id)
                    if isinstance(v, torch.Tensor) and v.is_cuda and \
                            not torch.cuda.is_current_stream_capturing():
                        v.record_stream(cur)
            prof = ctx.profiler
            if comm_async:
                cs = ctx.stream("comm")
                ev_in = torch.cuda.Event()
                ev_in.record(torch.cuda.current_stream(ctx.device))
                cs.wait_event(ev_in)
                with torch.cuda.stream(cs):
                    outs = op.interface.compute(op, ins, ctx)
                    if not torch.cuda.is_current_stream_capturing():
                        for v in ins:
                            if isinstance(v, torch.Tensor) and v.is_cuda:
                                v.record_stream(cs)
                    ev_out = torch.cuda.Event()
                    ev_out.record(cs)
                for t in op.outputs:
                    self._inflight[t.id] = ev_out
            elif prof is not None:
                tok = prof.begin(op)
                outs = op.interface.compute(op, ins, ctx)
                prof.end(tok)
            else:
                outs = op.interface.compute(op, ins, ctx)
            if _CHECK_NUMERIC:
                _check_numeric(op, outs)
            if _MEM_TRACE and ctx.device.type == "cuda":
                a = torch.cuda.memory_allocated(ctx.device)
                if a > self._mem_peak[0]:
                    census = {}
                    id2n = {t.id: t.name for o2 in plan.topo
                            for t in o2.outputs}
                    for tid, v in values.items():
                        if isinstance(v, torch.Tensor) and v.is_cuda:
                            census[id2n.get(tid, tid)] =                                 v.numel() * v.element_size()
                    top = sorted(census.items(), key=lambda kv: -kv[1])[:12]
                    self._mem_peak = (a, op.type, op.name, i, top,
                                      sum(census.values()))
            for t, v in zip(op.outputs, outs):
                values[t.id] = v
            if keep_values is not None:
                for t, v in zip(op.outputs, outs):
                    keep_values[t.id] = v
                continue                     # caller owns lifetimes
            # free dead intermediates (degree-based free, as in ComputeFunc)
            for t in op.inputs:
                if plan.last_use.get(t.id) == i and t.id in values:
                    del values[t.id]
            for tid in plan.dead_outputs.get(i, ()):
                values.pop(tid, None)

        # join any comm-stream work not consumed by an op (e.g. fetched
        # tensors) back into the compute stream
        if self._inflight:
            cur = torch.cuda.current_stream(ctx.device)
            for ev in self._inflight.values():
                cur.wait_event(ev)
            self._inflight.clear()

        out: List[torch.Tensor] = []
        for t in fetches:
            if t.id in values:
                out.append(values[t.id])
            elif t.id in feeds:
                out.append(feeds[t.id])
            elif t.get_data() is not None:
                out.append(t.get_data())
            else:
                raise RuntimeError(f"fetch {t.name} produced no value")
        return out
