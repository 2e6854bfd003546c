#!/usr/bin/env python3
"""Which captured subgraph corrupts: forward-only / +grads / +adam?"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,  # noqa: E402
                                  push_graph)
from hetu_amd.graph.ops import api as ht  # noqa: E402
from hetu_amd.graph.ops.optim import Adam  # noqa: E402
from hetu_amd.engine.runner import prepare_run_context  # noqa: E402

dev = torch.device("cuda", 0)


def build(with_adam):
    torch.manual_seed(0)
    layers, h, rows = 12, 256, 512
    dtype = torch.bfloat16
    g = DefineAndRunGraph("t")
    push_graph(g)
    try:
        x = ht.placeholder((rows, h), dtype=dtype, name="x")
        cur = x
        for i in range(layers):
            w = ht.variable(torch.ones(h), name=f"ln{i}.w")
            b = ht.variable(torch.zeros(h), name=f"ln{i}.b")
            y = ht.layer_norm(cur, w, b, 1e-5)
            w1 = ht.variable(torch.randn(2 * h, h, dtype=dtype) * 0.02,
                             name=f"l{i}.w1")
            y = ht.gelu(ht.linear(y, w1))
            w2 = ht.variable(torch.randn(h, 2 * h, dtype=dtype) * 0.02,
                             name=f"l{i}.w2")
            cur = ht.add(cur, ht.linear(y, w2))
        loss = ht.reduce_mean(ht.mul(cur, cur))
        fetches = [loss]
        if with_adam == "adam":
            fetches.append(Adam(lr=1e-4).minimize(loss))
        elif with_adam == "grads":
            gs = g.gradients([loss], list(g.parameters))
            fetches += [t for t in gs if t is not None]
    finally:
        pop_graph()
    return g, x, fetches


def probe(mode):
    g, x, fetches = build(mode)
    ctx = prepare_run_context(g, dev)
    pool = [torch.randn(512, 256, dtype=torch.bfloat16, device=dev)
            for _ in range(2)]
    outs = g.run(fetches, {x: pool[0]}, ctx=ctx)
    torch.cuda.synchronize()
    torch.cuda.synchronize()
    cg = torch.cuda.CUDAGraph()
    with torch.cuda.graph(cg):
        outs = g.run(fetches, {x: pool[1].clone()}, ctx=ctx)
    res = []
    for r in range(3):
        cg.replay()
        torch.cuda.synchronize()
        nbad = sum(1 for o in outs
                   if o.is_floating_point()
                   and not torch.isfinite(o.float()).all())
        res.append((float(outs[0].float()), nbad))
    print(f"{mode}: " + "  ".join(f"loss={l:.4f} badfetch={n}"
                                  for l, n in res), flush=True)
    del g
    torch.cuda.empty_cache()


probe("fwd")
probe("grads")
probe("adam")
