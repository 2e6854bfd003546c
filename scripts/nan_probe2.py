#!/usr/bin/env python3
"""Tower repro, finely instrumented: when exactly does state corrupt
(capture? first replay?) and what does the garbage look like?"""
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
torch.manual_seed(0)
layers, h, rows = 12, 256, 512
dtype = torch.bfloat16

g = DefineAndRunGraph("tower")
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
        y = ht.linear(y, w1)
        y = ht.gelu(y)
        w2 = ht.variable(torch.randn(h, 2 * h, dtype=dtype) * 0.02,
                         name=f"l{i}.w2")
        y = ht.linear(y, w2)
        cur = ht.add(cur, y)
    loss = ht.reduce_mean(ht.mul(cur, cur))
    opt = Adam(lr=1e-4)
    train_op = opt.minimize(loss)
finally:
    pop_graph()

ctx = prepare_run_context(g, dev)
adams = [op for op in g.ops if op.type == "AdamStep"]
id2name = {}
for op in g.ops:
    for t in op.outputs:
        id2name[t.id] = f"{op.type}:{op.name}"


def stat():
    torch.cuda.synchronize()
    bad = []
    for op in adams:
        st = op.interface.state
        if not st:
            continue
        for k in ("m", "v", "master"):
            if not torch.isfinite(st[k]).all():
                bad.append((op.name, k))
                break
    pb = [p.name for p in g.parameters
          if not torch.isfinite(p.get_data().float()).all()]
    return bad, pb


pool = [torch.randn(rows, h, dtype=dtype, device=dev) for _ in range(4)]
# step 0 eager
lv = g.run([loss, train_op], {x: pool[0]}, ctx=ctx)[0]
torch.cuda.synchronize()
print(f"eager step0 loss={float(lv.float()):.5f} bad={stat()[0][:3]}",
      flush=True)

# capture (records, does not execute)
static_x = pool[1].clone()
kept = {}
torch.cuda.synchronize()
cg = torch.cuda.CUDAGraph()
with torch.cuda.graph(cg):
    lv_out = g.run([loss, train_op], {x: static_x}, ctx=ctx,
                   keep_values=kept)[0]
bad, pb = stat()
print(f"after capture (no replay): bad_adam={bad[:3]} ({len(bad)}) "
      f"bad_params={pb[:3]} ({len(pb)})", flush=True)

for r in range(3):
    from hetu_amd.graph.ops.optim import AdamStepOp
    AdamStepOp.set_replay_step(2 + r)
    cg.replay()
    bad, pb = stat()
    print(f"after replay {r}: loss={float(lv_out.float()):.5f} "
          f"bad_adam={bad[:3]} ({len(bad)}) bad_params={pb[:3]} "
          f"({len(pb)})", flush=True)
    if bad or pb:
        # inspect the first bad adam: grad (kept), states
        name, k = bad[0] if bad else (None, None)
        for op in adams:
            if op.name == name:
                st = op.interface.state
                gt = kept.get(op.inputs[1].id)
                gf = gt.float() if gt is not None else None
                print(f"  {name}: master[min,max]="
                      f"[{st['master'].min():.3e},{st['master'].max():.3e}]"
                      f" m=[{st['m'].min():.3e},{st['m'].max():.3e}]"
                      f" v=[{st['v'].min():.3e},{st['v'].max():.3e}]",
                      flush=True)
                if gf is not None:
                    print(f"  grad({id2name.get(op.inputs[1].id)}): "
                          f"finite={bool(torch.isfinite(gf).all())} "
                          f"[{gf.min():.3e},{gf.max():.3e}]", flush=True)
        # first op in EXECUTION order whose inputs are finite but whose
        # output is not — the true NaN producer
        plan_topo = g.topo_sort([loss, train_op])

        def fin(t):
            v = kept.get(t.id)
            if v is None:
                v = t.get_data()
            if v is None or not v.is_floating_point():
                return None
            return bool(torch.isfinite(v.float()).all())

        shown = 0
        for op2 in plan_topo:
            if op2.type in ("Variable", "AdamStep", "Placeholder"):
                continue
            outs = [fin(t) for t in op2.outputs]
            if any(o is False for o in outs):
                ins = [(t.name, fin(t)) for t in op2.inputs]
                print(f"  bad op: {op2.type}:{op2.name} ins={ins} "
                      f"outs={outs}", flush=True)
                shown += 1
                if shown >= 6:
                    break
        break
