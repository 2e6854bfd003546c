"""Galvatron-style strategy search: cost model sanity + native DP core
parity with the Python fallback (reference tools/Galvatron csrc/dp_core)."""
import pytest

from hetu_amd.galvatron.cost_model import (CostModel, HardwareModel,
                                           ModelShape, Strategy)
from hetu_amd.galvatron.search import _dp_knapsack_py, dp_knapsack, search

GPT7B = ModelShape(n_layer=32, hidden=4096, ffn_hidden=16384, vocab=50304,
                   n_head=32, kind="gpt")


def test_dp_core_native_matches_python():
    times = [[1.0, 1.5], [2.0, 2.2], [0.5, 0.9], [3.0, 3.1]]
    mems = [[10.0, 4.0], [8.0, 3.0], [6.0, 2.0], [12.0, 5.0]]
    for cap in (36.0, 20.0, 15.0, 13.0, 5.0):
        bn, cn = dp_knapsack(times, mems, cap, buckets=512)
        bp, cp_ = _dp_knapsack_py(times, mems, 4, 2, cap, 512)
        if bp == float("inf"):
            assert bn == float("inf"), (cap, bn)
        else:
            assert abs(bn - bp) < 1e-9, (cap, bn, bp)
        assert cn == cp_, (cap, cn, cp_)


def test_search_single_gpu():
    st, res = search(GPT7B, 2048, 1, 8)
    assert st.world == 1
    assert res["fits"]
    assert res["tokens_per_sec"] > 1000


def test_search_8gpu_prefers_parallelism():
    st, res = search(GPT7B, 2048, 8, 64)
    assert st.world == 8
    assert res["fits"]
    st1, res1 = search(GPT7B, 2048, 1, 8)
    # whole-node throughput must beat a single GPU
    assert res["tokens_per_sec"] > 2 * res1["tokens_per_sec"]


def test_memory_forces_recompute_or_sharding():
    """Tiny memory cap must push the search to recompute/zero/tp."""
    hw = HardwareModel(hbm_capacity=40e9)
    st, res = search(GPT7B, 2048, 8, 64, hw=hw)
    assert res["fits"]
    assert st.zero or st.tp > 1 or st.pp > 1 or st.recompute_layers > 0


def test_cost_model_monotonic_in_batch():
    cm = CostModel(GPT7B, 2048)
    a = cm.evaluate(Strategy(dp=1, micro_batch=2, num_micro_batches=1), 2)
    b = cm.evaluate(Strategy(dp=1, micro_batch=4, num_micro_batches=1), 4)
    assert b["time"] > a["time"]
    assert b["tokens_per_sec"] > a["tokens_per_sec"] * 0.8


def test_osdp_per_layer_sharding_plan():
    """OSDP: shard exactly enough layers' optimizer states to fit
    (replicate the rest to avoid the per-step all-gather)."""
    from hetu_amd.galvatron.cost_model import CostModel, ModelShape, Strategy
    from hetu_amd.galvatron.search import osdp_plan
    # ~22B at dp8: replicated states overshoot HBM; partial shard fits
    shape = ModelShape(n_layer=40, hidden=6144, ffn_hidden=24576,
                       vocab=50304, n_head=48, kind="gpt")
    cm = CostModel(shape, 2048)
    st = Strategy()
    st.dp, st.tp, st.pp, st.micro_batch = 8, 1, 1, 1
    base = cm.evaluate(st, 8)
    plan, est = osdp_plan(cm, st, 8)
    if base["fits"]:
        assert sum(plan) == 0
    else:
        assert 0 < sum(plan) <= len(plan)
        assert est["mem"] < base["mem"]
        assert est["time"] >= base["time"]
        assert est["fits"]
    # small model: nothing sharded
    small = ModelShape(n_layer=12, hidden=768, ffn_hidden=3072, vocab=50304,
                       n_head=12, kind="gpt")
    p2, e2 = osdp_plan(CostModel(small, 1024), st, 8)
    assert sum(p2) == 0 and e2["fits"]
