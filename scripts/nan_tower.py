#!/usr/bin/env python3
"""Strip-down: which op combination makes capture-replay corrupt Adam
states?  Builds an N-layer tower with selectable pieces."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,  # noqa: E402
                                  push_graph)
from hetu_amd.graph.ops import api as ht  # noqa: E402
from hetu_amd.graph.ops.optim import Adam, SGD  # noqa: E402
from hetu_amd.engine.trainer import Trainer  # noqa: E402

dev = torch.device("cuda", 0)


def probe(tag, layers=12, h=256, rows=512, ln=True, gelu=True, resid=True,
          bias=True, opt="adam", dtype=torch.bfloat16, attn=False,
          steps=6):
    torch.manual_seed(0)
    g = DefineAndRunGraph(f"tower_{tag}")
    push_graph(g)
    try:
        x = ht.placeholder((rows, h), dtype=dtype, name="x")
        cur = x
        for i in range(layers):
            y = cur
            if ln:
                w = ht.variable(torch.ones(h), name=f"ln{i}.w")
                b = ht.variable(torch.zeros(h), name=f"ln{i}.b")
                y = ht.layer_norm(y, w, b, 1e-5)
            if attn:
                qkv_w = ht.variable(
                    torch.randn(3 * h, h, dtype=dtype) * 0.02,
                    name=f"at{i}.wqkv")
                qkv_b = ht.variable(torch.zeros(3 * h, dtype=dtype),
                                    name=f"at{i}.bqkv")
                qkv = ht.linear(ht.reshape(y, (1, rows, h)), qkv_w, qkv_b)
                nh = h // 128
                o = ht.fused_qkv_attention(qkv, nh, nh, 128, causal=True)
                y = ht.reshape(o, (rows, h))
            w1 = ht.variable(torch.randn(2 * h, h, dtype=dtype) * 0.02,
                             name=f"l{i}.w1")
            b1 = ht.variable(torch.zeros(2 * h, dtype=dtype),
                             name=f"l{i}.b1") if bias else None
            y = ht.linear(y, w1, b1)
            if gelu:
                y = ht.gelu(y)
            w2 = ht.variable(torch.randn(h, 2 * h, dtype=dtype) * 0.02,
                             name=f"l{i}.w2")
            b2 = ht.variable(torch.zeros(h, dtype=dtype),
                             name=f"l{i}.b2") if bias else None
            y = ht.linear(y, w2, b2)
            cur = ht.add(cur, y) if resid else y
        loss = ht.reduce_mean(ht.mul(cur, cur))
        o = Adam(lr=1e-4) if opt == "adam" else SGD(lr=1e-4)
        train_op = o.minimize(loss)
    finally:
        pop_graph()
    h_ = {"loss": loss, "train_op": train_op}
    tr = Trainer(g, h_, dev)
    pool = [torch.randn(rows, h, dtype=dtype, device=dev)
            for _ in range(4)]
    adams = [op for op in g.ops if op.type in ("AdamStep", "SGDStep")]
    verdict = "clean"
    for i in range(steps):
        lv = tr.step({x: pool[i % 4]})
        torch.cuda.synchronize()
        nbad = 0
        for op in adams:
            st = op.interface.state
            for k in ("m", "v", "master", "momentum_buffer"):
                if k in st and not torch.isfinite(st[k]).all():
                    nbad += 1
                    break
        params_bad = sum(
            1 for p in g.parameters
            if not torch.isfinite(p.get_data().float()).all())
        if nbad or params_bad or not torch.isfinite(lv.float()):
            verdict = (f"BAD@step{i} nbad={nbad} pbad={params_bad} "
                       f"loss={float(lv.float()):.4f}")
            break
    print(f"{tag}: {verdict}", flush=True)
    del tr, g
    torch.cuda.empty_cache()


if __name__ == "__main__":
    probe("mlp-full")                       # no attention at all
    probe("mlp-noln", ln=False)
    probe("mlp-nogelu", gelu=False)
    probe("mlp-noresid", resid=False)
    probe("mlp-nobias", bias=False)
    probe("mlp-sgd", opt="sgd")
    probe("mlp-fp32", dtype=torch.float32)
    probe("attn-full", attn=True, h=256)
