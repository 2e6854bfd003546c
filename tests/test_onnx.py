"""ONNX export/import roundtrip (reference v1/python/hetu/onnx parity)."""
import torch

from hetu_amd.graph.graph import DefineAndRunGraph, pop_graph, push_graph
from hetu_amd.graph.ops import api as ht
from hetu_amd.onnx import export_onnx, import_onnx


def _build_mlp():
    g = DefineAndRunGraph("mlp")
    push_graph(g)
    try:
        torch.manual_seed(0)
        x = ht.placeholder((4, 16), name="x")
        w1 = ht.variable(torch.randn(32, 16) * 0.1, name="w1")
        b1 = ht.variable(torch.zeros(32), name="b1")
        w2 = ht.variable(torch.randn(8, 32) * 0.1, name="w2")
        lw = ht.variable(torch.ones(16), name="ln_w")
        lb = ht.variable(torch.zeros(16), name="ln_b")
        h0 = ht.layer_norm(x, lw, lb)
        h = ht.gelu(ht.linear(h0, w1, b1))
        y = ht.softmax(ht.linear(h, w2), dim=-1)
    finally:
        pop_graph()
    return g, x, y


def test_onnx_roundtrip_mlp(tmp_path):
    from hetu_amd.engine.runner import prepare_run_context
    g, x, y = _build_mlp()
    path = str(tmp_path / "mlp.onnx")
    blob = export_onnx(g, [y], path)
    assert blob[:1]  # non-empty
    g2, inputs, outputs = import_onnx(path)
    assert list(inputs) == ["x"] and len(outputs) == 1
    xd = torch.randn(4, 16)
    ctx1 = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    ctx2 = prepare_run_context(g2, torch.device("cpu"), use_comm=False)
    y1, = g.run([y], {x: xd}, ctx=ctx1)
    y2, = g2.run([outputs[0]], {inputs["x"]: xd}, ctx=ctx2)
    assert torch.allclose(y1, y2, atol=1e-5), (y1 - y2).abs().max()


def test_onnx_embedding_and_matmul(tmp_path):
    from hetu_amd.engine.runner import prepare_run_context
    g = DefineAndRunGraph("emb")
    push_graph(g)
    try:
        torch.manual_seed(1)
        ids = ht.placeholder((6,), dtype=torch.int64, name="ids")
        table = ht.variable(torch.randn(50, 8), name="table")
        wa = ht.variable(torch.randn(8, 8) * 0.3, name="wa")
        e = ht.embedding(table, ids)
        y = ht.relu(ht.matmul(e, wa, trans_b=True))
    finally:
        pop_graph()
    blob = export_onnx(g, [y])
    g2, inputs, outputs = import_onnx(blob)
    idv = torch.randint(0, 50, (6,))
    ctx1 = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    ctx2 = prepare_run_context(g2, torch.device("cpu"), use_comm=False)
    y1, = g.run([y], {ids: idv}, ctx=ctx1)
    y2, = g2.run([outputs[0]], {inputs["ids"]: idv}, ctx=ctx2)
    assert torch.allclose(y1, y2, atol=1e-5)


def test_onnx_roundtrip_fuzz():
    """Random MLP-ish graphs roundtrip through ONNX bytes with identical
    outputs (40 cases)."""
    from hetu_amd.engine.runner import prepare_run_context
    from hypothesis import given, settings
    from hypothesis import strategies as st

    ACTS = ["gelu", "relu", "sigmoid", "tanh", "exp_c",
            "abs_n", "red_mean"]

    @settings(max_examples=40, deadline=None)
    @given(st.lists(st.tuples(st.integers(2, 24),
                              st.integers(0, len(ACTS) - 1)),
                    min_size=1, max_size=4),
           st.integers(0, 9999))
    def run(widths, seed):
        torch.manual_seed(seed)
        g = DefineAndRunGraph("f")
        push_graph(g)
        try:
            d = 8
            x = ht.placeholder((3, d), name="x")
            cur = x
            for li, (w_out, ai) in enumerate(widths):
                w = ht.variable(torch.randn(w_out, d) * 0.3,
                                name=f"w{li}")
                bvar = ht.variable(torch.randn(w_out) * 0.1,
                                   name=f"b{li}")
                lin = ht.linear(cur, w, bvar)
                a = ACTS[ai]
                if a == "exp_c":          # bounded exp + scalar op
                    cur = ht.exp(ht.mul(ht.tanh(lin), 0.5))
                elif a == "abs_n":
                    cur = ht.abs_(ht.neg(lin))
                elif a == "red_mean":     # reduce + broadcast back
                    cur = ht.sub(lin, ht.reduce_mean(lin, dim=1,
                                                     keepdim=True))
                else:
                    cur = getattr(ht, a)(lin)
                d = w_out
            y = ht.softmax(cur, dim=-1)
        finally:
            pop_graph()
        blob = export_onnx(g, [y])
        g2, inputs, outputs = import_onnx(blob)
        xd = torch.randn(3, 8)
        ctx1 = prepare_run_context(g, torch.device("cpu"), use_comm=False)
        ctx2 = prepare_run_context(g2, torch.device("cpu"),
                                   use_comm=False)
        y1, = g.run([y], {x: xd}, ctx=ctx1)
        y2, = g2.run([outputs[0]], {list(inputs.values())[0]: xd},
                     ctx=ctx2)
        assert torch.allclose(y1, y2, atol=1e-5)

    run()


def test_reduce_roundtrip():
    """Reduce sum/mean/max/min/prod export-import roundtrip."""
    import torch
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import DefineAndRunGraph, pop_graph, push_graph
    from hetu_amd.graph.ops import api as ht
    from hetu_amd.onnx.convert import export_onnx, import_onnx
    g = DefineAndRunGraph("r")
    push_graph(g)
    try:
        x = ht.placeholder((3, 5), name="x")
        y = ht.add(ht.reduce_mean(x, dim=1, keepdim=True),
                   ht.reduce_max(x, dim=1, keepdim=True))
    finally:
        pop_graph()
    blob = export_onnx(g, [y])
    g2, inputs, outputs = import_onnx(blob)
    xd = torch.randn(3, 5)
    ctx = prepare_run_context(g2, torch.device("cpu"), use_comm=False)
    (out,) = g2.run(outputs, {inputs["x"]: xd}, ctx=ctx)
    ref = xd.mean(1, keepdim=True) + xd.max(1, keepdim=True).values
    assert torch.allclose(out, ref, atol=1e-6)


def test_unary_concat_roundtrip():
    import torch
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import DefineAndRunGraph, pop_graph, push_graph
    from hetu_amd.graph.ops import api as ht
    from hetu_amd.onnx.convert import export_onnx, import_onnx
    g = DefineAndRunGraph("u")
    push_graph(g)
    try:
        x = ht.placeholder((2, 3), name="x")
        y = ht.concat([ht.exp(x), ht.abs_(ht.neg(x)),
                       ht.log(ht.add(ht.abs_(x), 1.0))], dim=1)
    finally:
        pop_graph()
    g2, inputs, outputs = import_onnx(export_onnx(g, [y]))
    xd = torch.randn(2, 3)
    ctx = prepare_run_context(g2, torch.device("cpu"), use_comm=False)
    (out,) = g2.run(outputs, {inputs["x"]: xd}, ctx=ctx)
    ref = torch.cat([xd.exp(), xd.neg().abs(), (xd.abs() + 1).log()], 1)
    assert torch.allclose(out, ref, atol=1e-6)


def test_slice_where_roundtrip():
    import torch
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import DefineAndRunGraph, pop_graph, push_graph
    from hetu_amd.graph.ops import api as ht
    from hetu_amd.onnx.convert import export_onnx, import_onnx
    g = DefineAndRunGraph("sw")
    push_graph(g)
    try:
        x = ht.placeholder((4, 6), name="x")
        sl = ht.slice_(x, 1, 2, 3)
        y = ht.where(ht.bool_(sl), sl, ht.neg(sl))
    finally:
        pop_graph()
    g2, inputs, outputs = import_onnx(export_onnx(g, [y]))
    xd = torch.randn(4, 6)
    ctx = prepare_run_context(g2, torch.device("cpu"), use_comm=False)
    (out,) = g2.run(outputs, {inputs["x"]: xd}, ctx=ctx)
    sref = xd[:, 2:5]
    ref = torch.where(sref.bool(), sref, -sref)
    assert torch.allclose(out, ref, atol=1e-6)
