#!/usr/bin/env python3
"""Split-capture workaround validation: graph A = fwd+bwd (fetch loss +
pre-update grads), graph B = optimizer step seeded with A's grad buffers.
Replay A;B per step — does the tower stay finite?"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
if os.environ.get("NAN_ROCBLAS") == "1":
    torch.backends.cuda.preferred_blas_library("cublas")
from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,  # noqa: E402
                                  push_graph)
from hetu_amd.graph.ops import api as ht  # noqa: E402
from hetu_amd.graph.ops.optim import Adam, AdamStepOp  # noqa: E402
from hetu_amd.engine.runner import prepare_run_context  # noqa: E402

dev = torch.device("cuda", 0)
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
    opt = Adam(lr=1e-4)
    train_op = opt.minimize(loss)
finally:
    pop_graph()

ctx = prepare_run_context(g, dev)
upd_ops = [op for op in g.ops
           if op.type in ("AdamStep", "ZeroAdamStep", "SGDStep")]
grad_ts = []
seen = set()
for op in upd_ops:
    t = op.inputs[1]
    if t.id not in seen:
        seen.add(t.id)
        grad_ts.append(t)

pool = [torch.randn(rows, h, dtype=dtype, device=dev) for _ in range(2)]
g.run([loss, train_op], {x: pool[0]}, ctx=ctx)
torch.cuda.synchronize()

static_x = pool[1].clone()
ga = torch.cuda.CUDAGraph()
with torch.cuda.graph(ga):
    outs_a = g.run([loss] + grad_ts, {x: static_x}, ctx=ctx)
loss_out, grad_outs = outs_a[0], outs_a[1:]
seed = {t.id: v for t, v in zip(grad_ts, grad_outs)}

adams = [op for op in g.ops if op.type == "AdamStep"]
for r in range(4):
    AdamStepOp.set_replay_step(2 + r)
    ga.replay()
    # optimizer runs EAGERLY on the replayed grad buffers
    g.run([train_op], {x: static_x}, ctx=ctx, seed_values=seed)
    torch.cuda.synchronize()
    nb = sum(1 for op in adams
             if not torch.isfinite(op.interface.state["m"]).all())
    pb = sum(1 for p in g.parameters
             if not torch.isfinite(p.get_data().float()).all())
    print(f"replay {r}: loss={float(loss_out.float()):.5f} bad_adam={nb} "
          f"bad_params={pb}", flush=True)
