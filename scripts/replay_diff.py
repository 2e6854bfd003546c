#!/usr/bin/env python3
"""From an identical state snapshot: one EAGER step vs one REPLAY of the
captured step, same feed.  Diff every param + adam state; the first
divergence names the guilty kernel."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from hetu_amd.engine.trainer import Trainer  # noqa: E402
from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph  # noqa: E402
from hetu_amd.graph.ops.optim import AdamStepOp  # noqa: E402

dev = torch.device("cuda", 0)
L, H, B, S, V = 12, 256, 2, 256, 50304
torch.manual_seed(1234)
cfg = GPTConfig(n_layer=L, n_head=2, n_kv_head=2, hidden=H,
                ffn_hidden=4 * H, vocab=V, max_seq=S)
g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                             dtype=torch.bfloat16, lr=1e-4)
tr = Trainer(g, h, dev, capture=False)
feed = {h["input_ids"]: torch.randint(0, V, (B, S), device=dev),
        h["labels"]: torch.randint(0, V, (B * S,), device=dev)}
tr.run_step(dict(feed))
torch.cuda.synchronize()
for t, v in feed.items():
    tr._static_feeds[t] = v.to(dev).clone()
kept_r = {}
cg = torch.cuda.CUDAGraph()
with torch.cuda.graph(cg):
    loss_out = g.run([h["loss"], h["train_op"]],
                     dict(tr._static_feeds), ctx=tr.ctx,
                     keep_values=kept_r)[0]
tr._cuda_graph = cg
tr._loss_out = loss_out
# one replay to advance to the state where things break
tr.replay()
torch.cuda.synchronize()

adams = [op for op in g.ops if op.type == "AdamStep"]
params = list(g.parameters)


def snapshot():
    torch.cuda.synchronize()
    snap = {"params": [p.get_data().clone() for p in params],
            "adam": [{k: (v.clone() if isinstance(v, torch.Tensor) else v)
                      for k, v in op.interface.state.items()
                      if k in ("master", "m", "v", "step")}
                     for op in adams],
            "shared": {k: (sh["host"].clone(), sh["dev"].clone(),
                           sh["step"])
                       for k, sh in AdamStepOp._shared_bc.items()},
            "tstep": tr._step}
    return snap


def restore(snap):
    torch.cuda.synchronize()
    for p, v in zip(params, snap["params"]):
        p.get_data().copy_(v)
    for op, st in zip(adams, snap["adam"]):
        for k, v in st.items():
            if isinstance(v, torch.Tensor):
                op.interface.state[k].copy_(v)
            else:
                op.interface.state[k] = v
    for k, (hst, dvc, stp) in snap["shared"].items():
        AdamStepOp._shared_bc[k]["host"].copy_(hst)
        AdamStepOp._shared_bc[k]["dev"].copy_(dvc)
        AdamStepOp._shared_bc[k]["step"] = stp
    tr._step = snap["tstep"]
    torch.cuda.synchronize()


base = snapshot()

# EAGER step from base (keeping every intermediate)
restore(base)
kept_e = {}
le = g.run([h["loss"], h["train_op"]],
           {t: v.clone() for t, v in tr._static_feeds.items()},
           ctx=tr.ctx, keep_values=kept_e)[0]
tr._step += 1
torch.cuda.synchronize()
kept_e = {k: (v.clone() if isinstance(v, torch.Tensor) else v)
          for k, v in kept_e.items()}
eager = snapshot()
le = float(le.float())

# REPLAY from base
restore(base)
tr.replay()
torch.cuda.synchronize()
rep = snapshot()
lr_ = float(loss_out.float())

print(f"eager loss={le:.6f}  replay loss={lr_:.6f}", flush=True)
bad = []
for i, (p, pe, pr) in enumerate(zip(params, eager["params"],
                                    rep["params"])):
    d = (pe.float() - pr.float()).abs().max().item()
    if d > 1e-6:
        bad.append((d, p.name, "param"))
for op, ae, ar in zip(adams, eager["adam"], rep["adam"]):
    for k in ("master", "m", "v"):
        d = (ae[k] - ar[k]).abs().max().item()
        if d > 1e-6 or not torch.isfinite(ar[k]).all():
            bad.append((d, op.name, k))
bad.sort(reverse=True)
print(f"divergent state tensors: {len(bad)}")
for d, name, k in bad[:10]:
    print(f"  {name}.{k}: max|diff|={d:.4e}")

# intermediates, in execution order: first divergence = guilty kernel
plan_topo = g.topo_sort([h["loss"], h["train_op"]])
shown = 0
for op2 in plan_topo:
    if op2.type in ("Variable", "Placeholder"):
        continue
    for t in op2.outputs:
        ve, vr = kept_e.get(t.id), kept_r.get(t.id)
        if not (isinstance(ve, torch.Tensor) and isinstance(vr,
                                                            torch.Tensor)):
            continue
        if not ve.is_floating_point():
            continue
        d = (ve.float() - vr.float()).abs().max().item()
        if d > 1e-5 or d != d:
            print(f"  DIVERGES {op2.type}:{op2.name} out={t.name} "
                  f"max|diff|={d:.4e}", flush=True)
            shown += 1
            break
    if shown >= 12:
        break
if shown == 0:
    print("  no divergent intermediates (?)")
