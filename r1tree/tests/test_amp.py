"""AMP grad scaler: overflow skip + backoff + growth; CheckFiniteOp."""
import torch

from hetu_amd.engine.amp import GradScaler


def test_scaler_overflow_and_growth():
    sc = GradScaler(init_scale=1024.0, growth_interval=3)
    g_ok = [torch.randn(8)]
    g_bad = [torch.tensor([1.0, float("inf")])]
    assert sc.check_and_update(g_ok)
    assert not sc.check_and_update(g_bad)
    assert sc.scale == 512.0 and sc.skipped == 1
    for _ in range(3):
        assert sc.check_and_update(g_ok)
    assert sc.scale == 1024.0     # grew back after interval


def test_scaler_unscale():
    sc = GradScaler(init_scale=8.0)
    g = [torch.full((4,), 8.0)]
    sc.unscale_(g)
    assert torch.allclose(g[0], torch.ones(4))


def test_check_finite_op():
    from hetu_amd.graph.graph import DefineAndRunGraph, push_graph, pop_graph
    from hetu_amd.graph.ops import api as ht
    from hetu_amd.engine.runner import prepare_run_context
    g = DefineAndRunGraph("cf")
    push_graph(g)
    try:
        a = ht.placeholder((4,), name="a")
        b = ht.placeholder((4,), name="b")
        f = ht.check_finite([a, b])
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    ok = g.run([f], {a: torch.ones(4), b: torch.ones(4)}, ctx=ctx)[0]
    assert float(ok) == 1.0
    bad = g.run([f], {a: torch.ones(4),
                      b: torch.tensor([1.0, float("nan"), 1, 1])},
                ctx=ctx)[0]
    assert float(bad) == 0.0


def test_pipeline_runner_scaler_skips(tmp_path):
    """A runner with a scaler must skip the update on inf grads."""
    from hetu_amd.models.llama import LlamaConfig, build_llama_pipeline_stage
    from hetu_amd.parallel.pipeline import PipelineRunner, PipelineSpec
    cfg = LlamaConfig(n_layer=1, n_head=2, n_kv_head=2, hidden=32,
                      ffn_hidden=64, vocab=64, max_seq=8)
    pspec = PipelineSpec(pp=1)
    stage = build_llama_pipeline_stage(cfg, pspec, 1, 8,
                                       dtype=torch.float32, lr=10.0)
    sc = GradScaler(init_scale=4.0)
    runner = PipelineRunner(pspec, stage, torch.device("cpu"), scaler=sc)
    h = stage.h
    ids = torch.randint(0, 64, (1, 8))
    labels = torch.randint(0, 64, (8,))
    before = {p.name: p.get_data().clone() for p in stage.graph.parameters}
    # poison the grads: force an inf by patching a buffer post-accumulate
    orig_acc = runner._bwd

    def bad_bwd(saved, gin):
        r = orig_acc(saved, gin)
        runner.grad_bufs[0][0, 0] = float("inf")
        return r
    runner._bwd = bad_bwd
    runner.step([{h["input_ids"]: ids, h["labels"]: labels}])
    assert sc.skipped == 1
    for p in stage.graph.parameters:
        assert torch.equal(p.get_data(), before[p.name]), p.name


def test_op_profiler():
    from hetu_amd.graph.graph import DefineAndRunGraph, push_graph, pop_graph
    from hetu_amd.graph.ops import api as ht
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.utils.profiler import OpProfiler, MemorySnapshots
    g = DefineAndRunGraph("prof")
    push_graph(g)
    try:
        a = ht.placeholder((64, 64), name="a")
        y = ht.gelu(ht.matmul(a, a))
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    prof = OpProfiler(use_events=False)
    ctx.profiler = prof
    g.run([y], {a: torch.randn(64, 64)}, ctx=ctx)
    s = prof.summary()
    assert "MatMul" in s and "Gelu" in s
    ms = MemorySnapshots()
    ms.mark("step0")
    assert "step0" in ms.report()


def test_autocast_context():
    """ht.autocast inserts bf16 casts on compute ops; grads return fp32
    (reference autocast.cc DataTransferOp insertion)."""
    import torch
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,
                                      push_graph)
    from hetu_amd.graph.ops import api as ht
    g = DefineAndRunGraph("ac")
    push_graph(g)
    try:
        x = ht.placeholder((4, 8), name="x")
        w = ht.variable(torch.randn(6, 8) * 0.1, name="w")
        with ht.autocast(torch.bfloat16):
            y = ht.linear(x, w)
        loss = ht.reduce_sum(ht.mul(y, y))
        gs = ht.gradients([loss], [w])
    finally:
        pop_graph()
    assert y.dtype == torch.bfloat16
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    yl, gw = g.run([y, gs[0]], {x: torch.randn(4, 8)}, ctx=ctx)
    assert yl.dtype == torch.bfloat16
    assert gw.dtype == torch.float32
