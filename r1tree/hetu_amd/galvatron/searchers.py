"""Additional strategy searchers: GPipe / PipeDream partitioners +
FlexFlow-style MCMC.

Reference parity: hetu/v1/python/hetu/distributed_strategies/ (gpipe,
pipedream, flexflow, optcnn, pipeopt searchers over the v1 graph).  Here
they operate on the same measured cost model the Galvatron search uses
(hetu_amd/galvatron/cost_model.py), so all searchers price work with the
MI355X constants (MFMA GEMM rate, RCCL/xGMI bandwidths).
"""
from __future__ import annotations

import math
import random
from typing import Dict, List, Sequence, Tuple

from .cost_model import CostModel, ModelShape, Strategy


# ---------------------------------------------------------------------------
# Pipeline partitioners
# ---------------------------------------------------------------------------
def gpipe_partition(layer_costs: Sequence[float], pp: int) -> List[int]:
    """Balanced contiguous partition: layers per stage minimizing the max
    stage cost (classic GPipe planner).  Returns layer counts per stage."""
    n = len(layer_costs)
    pp = min(pp, n)
    prefix = [0.0]
    for c in layer_costs:
        prefix.append(prefix[-1] + c)

    def stage_cost(i, j):                     # layers [i, j)
        return prefix[j] - prefix[i]

    # DP over (stage, first layer): bottleneck partition
    INF = float("inf")
    best = [[INF] * (n + 1) for _ in range(pp + 1)]
    cut = [[0] * (n + 1) for _ in range(pp + 1)]
    best[0][0] = 0.0
    for s in range(1, pp + 1):
        for j in range(1, n + 1):
            for i in range(s - 1, j):
                v = max(best[s - 1][i], stage_cost(i, j))
                if v < best[s][j]:
                    best[s][j] = v
                    cut[s][j] = i
    counts = []
    j = n
    for s in range(pp, 0, -1):
        i = cut[s][j]
        counts.append(j - i)
        j = i
    return counts[::-1]


def pipedream_partition(layer_costs: Sequence[float], pp: int,
                        act_comm_cost: float = 0.0
                        ) -> Tuple[List[int], float]:
    """PipeDream-style planner: bottleneck partition where each stage
    boundary also pays the activation p2p cost.  Returns (counts,
    bottleneck stage time)."""
    n = len(layer_costs)
    pp = min(pp, n)
    prefix = [0.0]
    for c in layer_costs:
        prefix.append(prefix[-1] + c)
    INF = float("inf")
    best = [[INF] * (n + 1) for _ in range(pp + 1)]
    cut = [[0] * (n + 1) for _ in range(pp + 1)]
    best[0][0] = 0.0
    for s in range(1, pp + 1):
        for j in range(1, n + 1):
            for i in range(s - 1, j):
                stage = prefix[j] - prefix[i]
                if s > 1 or j < n:
                    stage += act_comm_cost      # boundary send/recv
                v = max(best[s - 1][i], stage)
                if v < best[s][j]:
                    best[s][j] = v
                    cut[s][j] = i
    counts = []
    j = n
    for s in range(pp, 0, -1):
        i = cut[s][j]
        counts.append(j - i)
        j = i
    return counts[::-1], best[pp][n]


# ---------------------------------------------------------------------------
# FlexFlow-style MCMC over (dp, tp, pp, micro-batch)
# ---------------------------------------------------------------------------
def mcmc_search(shape: ModelShape, seq_len: int, world: int,
                global_batch: int, iters: int = 400,
                temperature: float = 0.05, seed: int = 0
                ) -> Tuple[Strategy, float]:
    """Metropolis search over the strategy space using the measured cost
    model as the simulator (FlexFlow's MCMC with delta-evaluation, applied
    to the SPMD strategy axes instead of per-op placement)."""
    rng = random.Random(seed)
    cm = CostModel(shape, seq_len)

    def factor_pairs(w):
        out = []
        for pp in [1, 2, 4, 8]:
            if w % pp:
                continue
            rest = w // pp
            for tp in [1, 2, 4, 8]:
                if rest % tp:
                    continue
                out.append((pp, tp, rest // tp))
        return out

    space = factor_pairs(world)

    def mk(pp, tp, dp, mb):
        st = Strategy()
        st.pp, st.tp, st.dp, st.cp = pp, tp, dp, 1
        st.micro_batch = mb
        return st

    def cost(st):
        try:
            r = cm.evaluate(st, global_batch)
        except Exception:
            return float("inf")
        if not r.get("fits", True):
            return float("inf")
        return r["time"]

    pp, tp, dp = space[0]
    mb = next(m for m in (1, 2, 4, 8)
              if global_batch % (dp * m) == 0)
    cur = mk(pp, tp, dp, mb)
    cur_c = cost(cur)
    best, best_c = cur, cur_c
    for _ in range(iters):
        pp, tp, dp = space[rng.randrange(len(space))]
        mb_cands = [m for m in (1, 2, 4, 8, 16)
                    if global_batch % (dp * m) == 0]
        if not mb_cands:
            continue
        cand = mk(pp, tp, dp, rng.choice(mb_cands))
        c = cost(cand)
        if c < cur_c or (c < float("inf") and cur_c < float("inf") and
                         rng.random() < math.exp((cur_c - c)
                                                 / max(temperature * cur_c,
                                                       1e-9))):
            cur, cur_c = cand, c
        if c < best_c:
            best, best_c = cand, c
    return best, best_c


def hetero_pipeline_partition(layer_costs: Sequence[float], pp: int,
                              stage_speeds: Sequence[float]
                              ) -> Tuple[List[int], float]:
    """Malleus-style heterogeneous pipeline partition: stage i's wall time
    is its layer-cost sum divided by its (relative) speed; the bottleneck
    partition assigns fewer layers to slower stages.  Returns (layer
    counts per stage, bottleneck time)."""
    n = len(layer_costs)
    pp = min(pp, n)
    assert len(stage_speeds) == pp
    prefix = [0.0]
    for c in layer_costs:
        prefix.append(prefix[-1] + c)
    INF = float("inf")
    best = [[INF] * (n + 1) for _ in range(pp + 1)]
    cut = [[0] * (n + 1) for _ in range(pp + 1)]
    best[0][0] = 0.0
    for s in range(1, pp + 1):
        spd = max(stage_speeds[s - 1], 1e-9)
        for j in range(1, n + 1):
            for i in range(s - 1, j):
                stage = (prefix[j] - prefix[i]) / spd
                v = max(best[s - 1][i], stage)
                if v < best[s][j]:
                    best[s][j] = v
                    cut[s][j] = i
    counts = []
    j = n
    for s in range(pp, 0, -1):
        i = cut[s][j]
        counts.append(j - i)
        j = i
    return counts[::-1], best[pp][n]
