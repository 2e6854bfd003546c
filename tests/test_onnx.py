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
