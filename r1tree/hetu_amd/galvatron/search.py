"""Galvatron-style automatic parallel-strategy search for MI355X nodes.

Reference parity: tools/Galvatron (csrc/dp_core.cpp DP knapsack;
galvatron/core/hybrid_parallel_config.py strategy enumeration).  The search
enumerates (pp, tp, dp, cp, micro-batch, zero) over the node, scores each
with the MI355X cost model, uses the native DP core (galvatron_dp in the
_hetu_hip extension; Python fallback below) to pick per-layer recompute
under the 288 GB budget, and emits the best Strategy.
"""
from __future__ import annotations

import math
from typing import List, Optional, Tuple

from .cost_model import CostModel, HardwareModel, ModelShape, Strategy


def _dp_knapsack_py(times, mems, L, S, cap, buckets):
    """Python mirror of csrc galvatron_dp (dp_core.cpp semantics)."""
    INF = float("inf")
    unit = cap / buckets
    f = [[INF] * (buckets + 1) for _ in range(L + 1)]
    choice = [[-1] * (buckets + 1) for _ in range(L)]
    for b in range(buckets + 1):
        f[0][b] = 0.0
    for l in range(L):
        for b in range(buckets + 1):
            best, arg = INF, -1
            for s in range(S):
                mu = int(math.ceil(mems[l][s] / unit))
                if mu > b or f[l][b - mu] == INF:
                    continue
                t = f[l][b - mu] + times[l][s]
                if t < best:
                    best, arg = t, s
            f[l + 1][b] = best
            choice[l][b] = arg
    best = f[L][buckets]
    out = [-1] * L
    if best < INF:
        b = buckets
        for l in range(L - 1, -1, -1):
            s = choice[l][b]
            out[l] = s
            b -= int(math.ceil(mems[l][s] / (cap / buckets)))
    return best, out


def dp_knapsack(times, mems, cap, buckets=256):
    """times/mems: [L][S] python lists. Returns (best_time, choices)."""
    L = len(times)
    S = len(times[0]) if L else 0
    try:
        from ..ops import functional as F
        ext = F._load_ext()
        if ext is not None and hasattr(ext, "galvatron_dp"):
            flat_t = [t for row in times for t in row]
            flat_m = [m for row in mems for m in row]
            best, out = ext.galvatron_dp(flat_t, flat_m, L, S, float(cap),
                                         buckets)
            return best, list(out)
    except Exception:  # noqa: BLE001
        pass
    return _dp_knapsack_py(times, mems, L, S, cap, buckets)


def _recompute_plan(cm: CostModel, st: Strategy, global_batch: int
                    ) -> Tuple[Strategy, dict]:
    """Pick per-layer recompute via the DP knapsack when the plain strategy
    does not fit (dp_core.cpp usage)."""
    base = cm.evaluate(st, global_batch)
    if base["fits"]:
        return st, base
    m = cm.m
    layers_per_stage = (m.n_layer + st.pp - 1) // st.pp
    num_mb = global_batch // (st.dp * st.micro_batch)
    tokens_mb = st.micro_batch * (cm.s // st.cp)
    inflight = min(st.pp, num_mb)
    # per-layer (time, act-mem) for {plain, recompute}
    times, mems = [], []
    for _ in range(layers_per_stage):
        t0 = cm.layer_time(tokens_mb, st.tp, cm.s // st.cp, False)
        t1 = cm.layer_time(tokens_mb, st.tp, cm.s // st.cp, True)
        m0 = inflight * cm.layer_act_bytes(tokens_mb, st.tp, False)
        m1 = inflight * cm.layer_act_bytes(tokens_mb, st.tp, True)
        times.append([t0, t1])
        mems.append([m0, m1])
    # activation budget = capacity - (params + optimizer + grads + head)
    full = cm.evaluate(
        Strategy(**{**st.__dict__, "recompute_layers": layers_per_stage}),
        global_batch)
    static_mem = full["mem"] - sum(r[1] for r in mems)
    act_cap = cm.hw.hbm_capacity - static_mem
    if act_cap <= 0:
        return st, base      # hopeless; caller discards (not fits)
    best, choices = dp_knapsack(times, mems, act_cap)
    if not choices or -1 in choices:
        st2 = Strategy(**{**st.__dict__,
                          "recompute_layers": layers_per_stage})
        return st2, cm.evaluate(st2, global_batch)
    n_rec = sum(1 for c in choices if c == 1)
    st2 = Strategy(**{**st.__dict__, "recompute_layers": n_rec})
    return st2, cm.evaluate(st2, global_batch)


def search(model: ModelShape, seq_len: int, n_gpus: int, global_batch: int,
           hw: Optional[HardwareModel] = None, allow_cp: bool = False,
           verbose: bool = False) -> Tuple[Strategy, dict]:
    """Best strategy for one MI355X node (xGMI all-to-all mesh)."""
    cm = CostModel(model, seq_len, hw)
    best: Optional[Tuple[Strategy, dict]] = None
    cands: List[Strategy] = []
    for pp in [p for p in (1, 2, 4, 8) if p <= n_gpus]:
        if model.n_layer % pp != 0 and pp > 1:
            continue
        for tp in [t for t in (1, 2, 4, 8) if pp * t <= n_gpus]:
            if model.n_head % tp != 0:
                continue
            rest = n_gpus // (pp * tp)
            if pp * tp * rest != n_gpus:
                continue
            cps = [1]
            if allow_cp:
                cps += [c for c in (2, 4, 8) if c <= rest]
            for cp in cps:
                dp = rest // cp
                if dp * cp != rest or dp < 1:
                    continue
                if global_batch % dp != 0:
                    continue
                per_dp = global_batch // dp
                for mb in (1, 2, 4, 8, 16):
                    if per_dp % mb != 0:
                        continue
                    num_mb = per_dp // mb
                    if pp > 1 and num_mb < pp:
                        continue     # bubble-dominated
                    for zero in ((False, True) if dp * cp > 1
                                 else (False,)):
                        cands.append(Strategy(dp=dp, tp=tp, pp=pp, cp=cp,
                                              micro_batch=mb,
                                              num_micro_batches=num_mb,
                                              zero=zero))
    for st in cands:
        st2, res = _recompute_plan(cm, st, global_batch)
        if not res["fits"] and not st2.zero and st2.dp * st2.cp > 1:
            # OSDP: shard just enough layers' optimizer states before
            # falling back to full ZeRO / rejecting the candidate
            plan, res2 = osdp_plan(cm, st2, global_batch)
            if res2["fits"]:
                res = res2
        if not res["fits"]:
            continue
        if verbose:
            print(f"  {st2.name():28s} t={res['time']*1e3:8.1f} ms "
                  f"mem={res['mem']/1e9:6.1f} GB "
                  f"tok/s={res['tokens_per_sec']:.0f}")
        if best is None or res["time"] < best[1]["time"]:
            best = (st2, res)
    if best is None:
        raise RuntimeError("no feasible strategy found")
    return best


# ---------------------------------------------------------------------------
# OSDP: per-layer optimizer-state sharding plan (reference: Galvatron-family
# OSDP — choose, layer by layer, whether optimizer states are replicated
# (fast update, 12 bytes/param memory on every rank) or ZeRO-sharded
# (memory / dp, but a reduce-scatter + all-gather per step).  The same
# layers x choices DP knapsack picks the cheapest plan that fits.)
# ---------------------------------------------------------------------------
def osdp_plan(cm: CostModel, st: "Strategy", global_batch: int):
    """Returns (plan, est): plan[i] = 1 if layer i keeps ZeRO-sharded
    optimizer states, 0 if replicated.  Shards exactly as many layers as
    the memory overshoot requires (identical layers -> greedy is optimal;
    each sharded layer pays ~one fp32 param all-gather per step)."""
    import math as _math
    base = cm.evaluate(st, global_batch)
    n = st.dp * st.cp
    L = (cm.m.n_layer + st.pp - 1) // st.pp
    if base["fits"] or n <= 1:
        return [0] * L, base

    p_layer = cm.m.layer_params / st.tp
    saving = 12.0 * p_layer * (1.0 - 1.0 / n)     # bytes freed per layer
    t_shard = cm.hw.allreduce_time(4.0 * p_layer, n) / 2.0  # ~all-gather
    over = base["mem"] - cm.hw.hbm_capacity
    k = min(L, int(_math.ceil(over / max(saving, 1.0))))
    plan = [1] * k + [0] * (L - k)
    est = dict(base)
    est["time"] = base["time"] + k * t_shard
    est["mem"] = base["mem"] - k * saving
    est["fits"] = est["mem"] <= cm.hw.hbm_capacity
    est["osdp_plan"] = plan
    return plan, est
