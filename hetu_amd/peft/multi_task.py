"""Multi-tenant LoRA fine-tuning (LobRA).

Reference parity: examples/lobra — many fine-tuning TASKS share one frozen
base model, each task owning its LoRA adapter pair per wrapped layer; a
static planner assigns per-task adapter ranks under a budget
(trainer/planner.py:211 GroupedStaticPlanner, :418 BalancedStaticPlanner,
:629 PrunedStaticPlanner) and a batch scheduler interleaves task
micro-batches proportional to their data sizes.

MI355X-native shape: adapters are ordinary graph variables (one A/B pair
per task per layer) so every task trains through the same frozen-base
subgraph with its own parameter set; planners/scheduler are pure functions
usable before building the graph.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch

from ..graph.ops import api as ht
from ..nn import init
from ..nn.module import Module
from ..nn.parallel import ColumnParallelLinear, ParallelSpec, \
    RowParallelLinear, _shard


class MultiLoRALinear(Module):
    """One frozen parallel linear + one LoRA adapter pair PER TASK.

    forward(x, task) routes through that task's adapters; tasks never mix
    gradients because each owns distinct variables (named
    "<name>.<task>.A/B").  Per-task ranks may differ (planner output)."""

    def __init__(self, base: Module, ranks: Dict[str, int],
                 alpha: float = 16.0, name: str = "mlora"):
        super().__init__()
        self.base = base
        self.ranks = dict(ranks)
        self.alpha = alpha
        self.name = name
        spec: ParallelSpec = base.spec
        w = base.weight
        g = w.graph
        for p in (base.weight, getattr(base, "bias", None)):
            if p is not None and p in g.parameters:
                g.parameters.remove(p)
                p.is_parameter = False
        out_f_local, in_f_local = tuple(w.shape)
        tp, ti = spec.tp, spec.my_tp_index()
        self.A: Dict[str, object] = {}
        self.B: Dict[str, object] = {}
        for task, r in ranks.items():
            nm = f"{name}.{task}"
            if isinstance(base, ColumnParallelLinear):
                a = init.normal((r, in_f_local), std=0.01,
                                name=f"{nm}.A").to(w.dtype)
                b = torch.zeros(out_f_local * tp, r)
                self.A[task] = ht.variable(
                    a, name=f"{nm}.A", ds=spec.ds_weight_dup(),
                    device_group=spec.device_group)
                self.B[task] = ht.variable(
                    _shard(b, 0, tp, ti).to(w.dtype), name=f"{nm}.B",
                    ds=spec.ds_weight_col(0),
                    device_group=spec.device_group)
            elif isinstance(base, RowParallelLinear):
                a = init.normal((r, in_f_local * tp), std=0.01,
                                name=f"{nm}.A").to(w.dtype)
                self.A[task] = ht.variable(
                    _shard(a, 1, tp, ti), name=f"{nm}.A",
                    ds=spec.ds_weight_row(1),
                    device_group=spec.device_group)
                self.B[task] = ht.variable(
                    torch.zeros(out_f_local, r, dtype=w.dtype),
                    name=f"{nm}.B", ds=spec.ds_weight_dup(),
                    device_group=spec.device_group)
            else:
                raise TypeError(
                    "MultiLoRALinear wraps Column/RowParallelLinear")

    def forward(self, x, task: str):
        y = self.base(x)
        r = self.ranks[task]
        h = ht.linear(x, self.A[task])
        d = ht.linear(h, self.B[task])
        return ht.add(y, ht.mul(d, self.alpha / r))

    def task_parameters(self, task: str):
        return [self.A[task], self.B[task]]


# --------------------------------------------------------------------------
# static planners (reference trainer/planner.py Group/Balance/Prune)
# --------------------------------------------------------------------------
def balance_plan(demands: Dict[str, float], rank_budget: int,
                 r_min: int = 2, r_max: int = 64) -> Dict[str, int]:
    """Allocate adapter ranks proportional to per-task demand weights
    under a total-rank budget (largest remainder), clamped to
    [r_min, r_max]."""
    total = sum(demands.values())
    assert total > 0
    raw = {t: rank_budget * w / total for t, w in demands.items()}
    out = {t: max(r_min, min(r_max, int(v))) for t, v in raw.items()}
    left = rank_budget - sum(out.values())
    order = sorted(demands, key=lambda t: out[t] - raw[t])
    i = 0
    while left > 0 and i < len(order):
        t = order[i]
        if out[t] < r_max:
            out[t] += 1
            left -= 1
        i += 1
    return out


def group_plan(ranks: Dict[str, int], n_groups: int
               ) -> List[Tuple[int, List[str]]]:
    """Partition tasks into n_groups adapter groups; a group shares one
    adapter of rank max(member ranks) (memory-saving at some quality
    cost, reference GroupedStaticPlanner).  Greedy first-fit-decreasing
    balancing the per-group wasted rank."""
    items = sorted(ranks.items(), key=lambda kv: -kv[1])
    groups: List[List[str]] = [[] for _ in range(n_groups)]
    gmax = [0] * n_groups
    gload = [0] * n_groups
    for t, r in items:
        # placing into group i wastes (max(gmax[i], r)*(len+1) - load - r)
        best = min(range(n_groups), key=lambda i: (
            max(gmax[i], r) * (len(groups[i]) + 1) - gload[i] - r,
            len(groups[i])))
        groups[best].append(t)
        gmax[best] = max(gmax[best], r)
        gload[best] += r
    return [(gmax[i], groups[i]) for i in range(n_groups) if groups[i]]


def prune_plan(ranks: Dict[str, int], utilities: Dict[str, float],
               rank_budget: int) -> Dict[str, int]:
    """Halve the rank of the lowest-utility-per-rank tasks until the
    total fits the budget (reference PrunedStaticPlanner)."""
    out = dict(ranks)
    while sum(out.values()) > rank_budget:
        cands = [t for t in out if out[t] > 1]
        if not cands:
            break
        t = min(cands, key=lambda t: utilities.get(t, 0.0) / out[t])
        out[t] = max(1, out[t] // 2)
    return out


class TaskBatchScheduler:
    """Deterministic stride scheduler: next() yields task ids with
    long-run frequency proportional to data sizes (reference lobra batch
    scheduler semantics — every task progresses every epoch, large tasks
    more often)."""

    def __init__(self, sizes: Dict[str, int]):
        assert sizes and all(v > 0 for v in sizes.values())
        self.sizes = dict(sizes)
        self._deficit = {t: 0.0 for t in sizes}
        total = float(sum(sizes.values()))
        self._share = {t: v / total for t, v in sizes.items()}

    def next(self) -> str:
        for t in self._deficit:
            self._deficit[t] += self._share[t]
        t = max(sorted(self._deficit), key=lambda t: self._deficit[t])
        self._deficit[t] -= 1.0
        return t

    def schedule(self, n: int) -> List[str]:
        return [self.next() for _ in range(n)]
