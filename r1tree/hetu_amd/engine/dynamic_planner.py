"""Hydraulis-style per-batch strategy dispatch for variable sequence
lengths.

Reference parity: examples/hydraulis/strategy/{cost_model.py:14 (the
a*s^2 + b*s + c per-strategy time fit), dynamic_pulp.py, new_planning.py}:
per batch, choose which strategy each sequence bucket trains under and how
to pack, coordinated via the kv store.  The MI355X fit uses the Galvatron
cost model to generate the (a, b, c) coefficients per candidate strategy;
the ILP is replaced by an equivalent small exhaustive/greedy assignment
(bucket count is tiny), which is exact for the bucket-to-strategy case.
"""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

from ..galvatron.cost_model import CostModel, ModelShape, Strategy


def fit_cost_curve(cm: CostModel, st: Strategy, seqs=(512, 1024, 2048, 4096)
                   ) -> Tuple[float, float, float]:
    """Least-squares fit t(s) = a*s^2 + b*s + c from the cost model."""
    import numpy as np
    xs, ys = [], []
    for s in seqs:
        cm2 = CostModel(cm.m, s, cm.hw)
        try:
            r = cm2.evaluate(st, st.dp * st.micro_batch)
        except AssertionError:
            continue
        xs.append(s)
        ys.append(r["time"])
    A = np.stack([np.array(xs, float) ** 2, np.array(xs, float),
                  np.ones(len(xs))], 1)
    coef, *_ = np.linalg.lstsq(A, np.array(ys), rcond=None)
    return tuple(float(c) for c in coef)


class DynamicPlanner:
    """Per-batch: given the batch's seq-len histogram (bucketed), assign
    each bucket to a strategy minimizing the serialized makespan
    (buckets run one after another under hot switching)."""

    def __init__(self, model: ModelShape, n_gpus: int,
                 candidates: Sequence[Strategy], buckets: Sequence[int]):
        self.buckets = sorted(buckets)
        self.cands = list(candidates)
        self.curves: Dict[Tuple[int, int], Tuple[float, float, float]] = {}
        cm = CostModel(model, self.buckets[-1])
        for ci, st in enumerate(self.cands):
            self.curves[ci] = fit_cost_curve(cm, st)

    def time_for(self, ci: int, seq: int, n_seqs: int) -> float:
        a, b, c = self.curves[ci]
        per = a * seq * seq + b * seq + c
        st = self.cands[ci]
        per_step = st.dp * st.micro_batch
        steps = max(1, (n_seqs + per_step - 1) // per_step)
        return per * steps

    def plan(self, seq_lens: Sequence[int]) -> Dict[int, int]:
        """bucket -> candidate index; exact enumeration per bucket (each
        bucket independently picks its fastest strategy — the serialized
        objective decomposes)."""
        counts: Dict[int, int] = {b: 0 for b in self.buckets}
        for s in seq_lens:
            for b in self.buckets:
                if s <= b:
                    counts[b] += 1
                    break
            else:
                counts[self.buckets[-1]] += 1
        out: Dict[int, int] = {}
        for b, n in counts.items():
            if n == 0:
                continue
            best = min(range(len(self.cands)),
                       key=lambda ci: self.time_for(ci, b, n))
            out[b] = best
        return out
