"""Graph zoo: Eager + DefineAndRun graphs, context stack, autodiff.

MI355X-native re-design of the reference graph layer
(/root/reference/hetu/graph/graph.h:25-880, graph.cc:117 Gradients,
define_and_run_graph.cc): graphs are Python objects holding the op store;
execution dispatches to torch-ROCm tensors + hand-written HIP kernels.
Instead of the reference's compiled ExecutableGraph instances, a
DefineAndRunGraph caches per-(fetches, strategy) execution plans (topo order +
refcount schedule) and the steady-state hot loop can be captured into a
hipGraph via torch.cuda.graphs by the engine.
"""
from __future__ import annotations

from typing import Dict, Iterable, List, Optional, Sequence

import torch

from .op import Op, OpInterface
from .tensor import Tensor, TensorMeta

_GRAPH_STACK: List["Graph"] = []


def current_graph() -> "Graph":
    if not _GRAPH_STACK:
        _GRAPH_STACK.append(EagerGraph("default_eager"))
    return _GRAPH_STACK[-1]


def push_graph(g: "Graph"):
    _GRAPH_STACK.append(g)


def pop_graph() -> "Graph":
    return _GRAPH_STACK.pop()


class _GraphCtx:
    def __init__(self, g):
        self.g = g

    def __enter__(self):
        push_graph(self.g)
        return self.g

    def __exit__(self, *a):
        pop_graph()


def graph(kind: str = "define_and_run", name: str = "", **kwargs):
    """Context manager mirroring hetu.graph(...) (reference context.py:115)."""
    if kind in ("define_and_run", "dar"):
        g = DefineAndRunGraph(name or "dar", **kwargs)
    elif kind == "eager":
        g = EagerGraph(name or "eager", **kwargs)
    else:
        raise ValueError(f"unknown graph kind {kind}")
    return _GraphCtx(g)


class Graph:
    def __init__(self, name: str):
        self.name = name
        self.ops: List[Op] = []
        self.parameters: List[Tensor] = []
        self._tensor_by_id: Dict[int, Tensor] = {}
        self._rc_scope_stack: List[int] = []   # active recompute scopes

    # ---- op construction -------------------------------------------------
    def make_op(self, interface: OpInterface, inputs: Sequence[Tensor],
                attrs: Dict, name: str = "",
                ds_list=None, device_group=None) -> Op:
        op = Op(interface, list(inputs), attrs, name=name, graph=self)
        if self._rc_scope_stack and interface.type not in (
                "Placeholder", "Variable", "Constant"):
            op.attrs["_rc_scope"] = self._rc_scope_stack[-1]
        metas = interface.infer_meta(op.attrs, op.inputs)
        for i, meta in enumerate(metas):
            t = Tensor(op, i, meta, graph=self)
            op.outputs.append(t)
            self._tensor_by_id[t.id] = t
        if ds_list is not None:
            for t, ds in zip(op.outputs, ds_list):
                t.ds = ds
        if device_group is not None:
            for t in op.outputs:
                t.device_group = device_group
        interface.deduce_states(op)
        self.ops.append(op)
        self._post_make_op(op)
        return op

    def _post_make_op(self, op: Op):
        pass

    # ---- activation recompute (reference recompute.cc:23-318: per-op
    # recompute flags duplicate the forward subgraph into the backward) ---
    class _RecomputeScope:
        def __init__(self, graph, idx):
            self.graph, self.idx = graph, idx

        def __enter__(self):
            self.graph._rc_scope_stack.append(self.idx)
            return self

        def __exit__(self, *a):
            self.graph._rc_scope_stack.pop()

    def recompute_scope(self, idx: int):
        """Ops built inside `with g.recompute_scope(i)` form recompute
        scope i: after autodiff, apply_recompute() clones the scope's ops
        that backward needs and rewires backward to the clones, so the
        scope's internal activations free right after forward."""
        return Graph._RecomputeScope(self, idx)

    def apply_recompute(self, watermark: int) -> int:
        """watermark = len(self.ops) when autodiff started (ops at index
        >= watermark are backward/optimizer ops).  Returns the number of
        cloned ops.  Boundary tensors (consumed by another scope or
        unscoped forward ops) are kept, exactly like torch checkpointing
        keeps block inputs."""
        fwd_ops = self.ops[:watermark]
        bwd_ops = self.ops[watermark:]
        scope_of = {op.id: op.attrs.get("_rc_scope") for op in fwd_ops}
        # forward consumers per tensor
        fwd_consumers: Dict[int, List[Op]] = {}
        for op in fwd_ops:
            for t in op.inputs:
                fwd_consumers.setdefault(t.id, []).append(op)
        # internal tensors of each scope: produced in scope s and only
        # consumed (in forward) within scope s
        internal: Dict[int, int] = {}   # tensor id -> scope
        for op in fwd_ops:
            s_id = scope_of.get(op.id)
            if s_id is None:
                continue
            for t in op.outputs:
                cons = fwd_consumers.get(t.id, [])
                if all(scope_of.get(c.id) == s_id for c in cons):
                    internal[t.id] = s_id
        # which internal tensors does backward reference?
        needed: Dict[int, set] = {}     # scope -> set of tensor ids
        for op in bwd_ops:
            for t in op.inputs:
                s_id = internal.get(t.id)
                if s_id is not None:
                    needed.setdefault(s_id, set()).add(t.id)
        total = 0
        for s_id, tids in sorted(needed.items()):
            # ops of the scope whose outputs are transitively needed
            scope_ops = [op for op in fwd_ops
                         if scope_of.get(op.id) == s_id]
            need_ops: List[Op] = []
            need_t = set(tids)
            for op in reversed(scope_ops):
                if any(t.id in need_t for t in op.outputs):
                    need_ops.append(op)
                    for t in op.inputs:
                        if t.id in internal and internal[t.id] == s_id:
                            need_t.add(t.id)
            need_ops.reverse()
            # clone in topo (= creation) order
            mapping: Dict[int, Tensor] = {}
            need_ops_cloned: List[Op] = []
            for op in need_ops:
                new_in = [mapping.get(t.id, t) for t in op.inputs]
                attrs = {k: v for k, v in op.attrs.items()
                         if k != "_rc_scope"}
                new_op = self.make_op(type(op.interface)(), new_in, attrs,
                                      name=op.name + "_rc")
                for old_t, new_t in zip(op.outputs, new_op.outputs):
                    new_t.ds = old_t.ds
                    new_t.device_group = old_t.device_group
                    mapping[old_t.id] = new_t
                need_ops_cloned.append(new_op)
                total += 1
            # rewire backward references to the clones
            for op in bwd_ops:
                for i, t in enumerate(op.inputs):
                    if t.id in mapping:
                        op.inputs[i] = mapping[t.id]
            # scheduling: the clones' inputs (block inputs) are ready right
            # after forward, so Kahn's min-id order would run EVERY scope's
            # clones at the fwd->bwd boundary and re-materialize all blocks
            # at once (worse than no recompute).  Key each scope's clones
            # just before their earliest backward consumer so they run
            # lazily, one block at a time.
            clone_out_ids = {t.id for op_ in need_ops_cloned
                             for t in op_.outputs}
            first_consumer = None
            for op in bwd_ops:
                if any(t.id in clone_out_ids for t in op.inputs):
                    if first_consumer is None or op.id < first_consumer:
                        first_consumer = op.id
            if first_consumer is not None:
                for j, op_ in enumerate(need_ops_cloned):
                    op_.attrs["_sched_key"] = \
                        first_consumer - 0.5 + j * 1e-6
        return total

    # ---- topology --------------------------------------------------------
    def topo_sort(self, fetches: Iterable[Tensor]) -> List[Op]:
        """Reverse-reachable subgraph from fetches, in topological order.

        Kahn's algorithm with a min-heap keyed by each op's scheduling key
        (creation id by default).  An op may carry attrs["_sched_key"] to
        be emitted as soon as possible after its dependencies — the grad
        bucket all-reduce ops use this so the collective is issued right
        after the last gradient of the bucket, overlapping with the rest
        of backward (reference AllReduceCoalesce placement).
        HETU_AMD_TOPO=dfs restores the round-1 DFS post-order (debug)."""
        import os as _os
        if _os.environ.get("HETU_AMD_TOPO", "kahn") == "dfs":
            return self._topo_sort_dfs(fetches)
        import heapq
        reach: Dict[int, Op] = {}
        stack = [t.producer for t in fetches if t.producer is not None]
        while stack:
            op = stack.pop()
            if op.id in reach:
                continue
            reach[op.id] = op
            for t in op.inputs:
                if t.producer is not None:
                    stack.append(t.producer)
            stack.extend(op.in_deps)

        indeg: Dict[int, int] = {}
        consumers: Dict[int, List[int]] = {}
        for op in reach.values():
            deps = {t.producer.id for t in op.inputs
                    if t.producer is not None}
            deps.update(d.id for d in op.in_deps)
            indeg[op.id] = len(deps)
            for d in deps:
                consumers.setdefault(d, []).append(op.id)

        def key(op: Op) -> float:
            return op.attrs.get("_sched_key", float(op.id))

        heap = [(key(reach[oid]), oid) for oid, n in indeg.items() if n == 0]
        heapq.heapify(heap)
        order: List[Op] = []
        while heap:
            _, oid = heapq.heappop(heap)
            op = reach[oid]
            order.append(op)
            for cid in consumers.get(oid, ()):
                indeg[cid] -= 1
                if indeg[cid] == 0:
                    heapq.heappush(heap, (key(reach[cid]), cid))
        if len(order) != len(reach):
            bad = [reach[oid].name for oid, n in indeg.items() if n > 0]
            raise RuntimeError(f"cycle detected among ops {bad[:5]}")
        return order

    def _topo_sort_dfs(self, fetches: Iterable[Tensor]) -> List[Op]:
        visited: Dict[int, bool] = {}
        order: List[Op] = []

        def visit(op: Op):
            state = visited.get(op.id)
            if state is True:
                return
            if state is False:
                raise RuntimeError(f"cycle detected at op {op.name}")
            visited[op.id] = False
            for t in op.inputs:
                if t.producer is not None:
                    visit(t.producer)
            for dep in op.in_deps:
                visit(dep)
            visited[op.id] = True
            order.append(op)

        for t in fetches:
            if t.producer is not None:
                visit(t.producer)
        return order

    # ---- autodiff (reference graph.cc:117 Gradients) ---------------------
    def gradients(self, ys: Sequence[Tensor], xs: Sequence[Tensor],
                  grad_ys: Optional[Sequence[Tensor]] = None
                  ) -> List[Optional[Tensor]]:
        from ..graph.ops.basics import make_ones_like, make_add_n
        ys = list(ys)
        xs = list(xs)
        if grad_ys is None:
            grad_ys = [make_ones_like(self, y) for y in ys]

        # accumulate grads per tensor id
        grad_map: Dict[int, List[Tensor]] = {}
        for y, gy in zip(ys, grad_ys):
            grad_map.setdefault(y.id, []).append(gy)

        topo = self.topo_sort(ys)
        needed = self._backward_reachable(topo, xs)

        def finalize(t: Tensor, g: Optional[Tensor]) -> Optional[Tensor]:
            """Partial-grad sum->reduce rewrite (reference graph.cc:161-260):
            a non-parameter tensor whose gradient is partial gets a comm op
            to t's own layout (allreduce / reduce-scatter); parameter grads
            stay partial for the optimizer's bucketed reduction."""
            if g is None or t.is_parameter:
                return g
            if (g.ds is not None and g.ds.partial > 1
                    and t.ds is not None and t.ds.partial <= 1):
                from .ops.comm import make_comm
                return make_comm(self, g, t.ds, name=f"grad_reduce_{t.name}")
            return g

        def reduce_grads(t: Tensor) -> Optional[Tensor]:
            gs = grad_map.get(t.id)
            if not gs:
                return None
            if len(gs) == 1:
                return finalize(t, gs[0])
            ds0 = gs[0].ds
            if all((g.ds is None and ds0 is None)
                   or (g.ds is not None and ds0 is not None
                       and g.ds.check_equal(ds0)) for g in gs[1:]):
                return finalize(t, make_add_n(self, gs))
            gs = [finalize(t, g) for g in gs]
            return make_add_n(self, gs)

        for op in reversed(topo):
            if op.id not in needed:
                continue
            gouts = [reduce_grads(t) for t in op.outputs]
            if all(g is None for g in gouts):
                continue
            gins = op.interface.gradient(op, gouts)
            assert len(gins) == len(op.inputs), (
                f"{op.type}.gradient returned {len(gins)} grads for "
                f"{len(op.inputs)} inputs")
            for t, g in zip(op.inputs, gins):
                if g is not None:
                    grad_map.setdefault(t.id, []).append(g)

        return [reduce_grads(x) for x in xs]

    def _backward_reachable(self, topo: List[Op], xs: Sequence[Tensor]):
        """Ops on a path from xs to ys: only these need gradient calls."""
        xs_ids = {x.id for x in xs}
        needed = set()
        for op in topo:  # forward order: op needed if any input is x or from needed op
            if any(t.id in xs_ids or
                   (t.producer is not None and t.producer.id in needed)
                   for t in op.inputs):
                needed.add(op.id)
        return needed


class EagerGraph(Graph):
    """Imperative mode: compute() runs at op creation, values stored on
    tensors (reference eager_graph.cc)."""

    def __init__(self, name: str, device: Optional[torch.device] = None):
        super().__init__(name)
        self.device = device or torch.device("cpu")

    def _post_make_op(self, op: Op):
        if op.type in ("Variable", "Placeholder"):
            return  # data attached right after construction
        from .executor import ExecContext
        ctx = ExecContext(device=self.device)
        inputs = [t.get_data() for t in op.inputs]
        outs = op.interface.compute(op, inputs, ctx)
        for t, v in zip(op.outputs, outs):
            t.set_data(v)


class DefineAndRunGraph(Graph):
    """Symbolic graph; run(fetches, feed_dict) executes via a cached plan.

    The reference keys plans by (strategy_id, fetches) and re-instantiates an
    ExecutableGraph per strategy (define_and_run_graph.cc:1174); here a plan
    is the topo order + free schedule for a fetch set, and parallel execution
    context (rank, groups) lives in the Executor.
    """

    def __init__(self, name: str, num_strategy: int = 1):
        super().__init__(name)
        self.num_strategy = num_strategy
        self.cur_strategy_id = 0
        self._plans: Dict = {}
        self._executor = None

    def executor(self):
        from .executor import Executor
        if self._executor is None:
            self._executor = Executor(self)
        return self._executor

    def run(self, fetches, feed_dict=None, **kwargs):
        single = isinstance(fetches, Tensor)
        if single:
            fetches = [fetches]
        results = self.executor().run(fetches, feed_dict or {}, **kwargs)
        return results[0] if single else results
