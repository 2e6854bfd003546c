#!/usr/bin/env python3
"""Does the replayed hipGraph run Adam BEFORE backward?  Sentinel test:
poison the kept grad buffers before replay; if Adam's m reflects the
sentinel, node dependencies were lost.  Also dumps the graph topology via
CUDAGraph.debug_dump when available."""
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
adams = [op for op in g.ops if op.type == "AdamStep"]
pool = [torch.randn(rows, h, dtype=dtype, device=dev) for _ in range(2)]
g.run([loss, train_op], {x: pool[0]}, ctx=ctx)
torch.cuda.synchronize()

kept = {}
cg = torch.cuda.CUDAGraph()
try:
    cg.enable_debug_mode()
    dbg = True
except Exception as e:  # noqa: BLE001
    print(f"debug mode unavailable: {e}")
    dbg = False
with torch.cuda.graph(cg):
    outs = g.run([loss, train_op], {x: pool[1].clone()}, ctx=ctx,
                 keep_values=kept)
if dbg:
    try:
        cg.debug_dump("gpurun_out/graph.dot")
        print("dumped gpurun_out/graph.dot")
    except Exception as e:  # noqa: BLE001
        print(f"debug_dump failed: {e}")

# sentinel: poison ln5.w1's kept grad buffer
target = None
for op in adams:
    if op.inputs[0].name.startswith("l5.w1"):
        target = op
        break
gt = kept[target.inputs[1].id]
st = target.interface.state
print(f"grad buffer shape {tuple(gt.shape)} dtype {gt.dtype}")
gt.fill_(777.0)
m_before = st["m"].clone()
torch.cuda.synchronize()
cg.replay()
torch.cuda.synchronize()
# real grads are O(0.01); if adam consumed 777s, m jumps by ~0.1*777
delta = (st["m"] - m_before).abs().max().item()
print(f"replayed: max|m delta| = {delta:.4f} "
      f"({'ADAM SAW SENTINEL (ran before backward)' if delta > 1.0 else 'adam saw real grads'})")
print(f"loss after replay: {float(outs[0].float()):.4f}")
nb = sum(1 for op in adams if not torch.isfinite(op.interface.state['m']).all())
print(f"bad adams: {nb}")
