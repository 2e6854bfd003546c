#!/usr/bin/env python3
"""Localize the capture-replay NaN: which tensor goes non-finite first,
and under which variants (fused attn off, later capture, model size)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

variant = sys.argv[1] if len(sys.argv) > 1 else "base"
model = sys.argv[2] if len(sys.argv) > 2 else "gpt3-7b"
cap_at = int(sys.argv[3]) if len(sys.argv) > 3 else 1

if variant == "nofuse":
    os.environ["HETU_AMD_FUSED_ATTN"] = "0"

import torch  # noqa: E402
from hetu_amd.engine.trainer import Trainer  # noqa: E402
from hetu_amd.models.gpt import GPT_CONFIGS, build_gpt_train_graph  # noqa: E402
from hetu_amd.graph.ops.optim import AdamStepOp  # noqa: E402

cfg = GPT_CONFIGS[model]
B, S = (16, 2048) if model == "gpt3-7b" else (8, 1024)
torch.manual_seed(1234)
dev = torch.device("cuda", 0)
g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                             dtype=torch.bfloat16, lr=1e-4)
tr = Trainer(g, h, dev, capture=False)
pool = [torch.randint(0, cfg.vocab, (B, S), device=dev) for _ in range(4)]
lpool = [torch.randint(0, cfg.vocab, (B * S,), device=dev)
         for _ in range(4)]
params = list(g.parameters)
adams = [op for op in g.ops if op.type == "AdamStep"]


def check(tag):
    torch.cuda.synchronize()
    bad_p = [p.name for p in params
             if not torch.isfinite(p.get_data()).all()]
    bad_m = []
    for op in adams[:400]:
        st = op.interface.state
        if "m" in st and (not torch.isfinite(st["m"]).all()
                          or not torch.isfinite(st["v"]).all()
                          or not torch.isfinite(st["master"]).all()):
            bad_m.append(op.name)
    print(f"  [{tag}] bad_params={bad_p[:4]} ({len(bad_p)}) "
          f"bad_adam={bad_m[:4]} ({len(bad_m)})", flush=True)
    return bad_p or bad_m


kept = {}
id2name = {}
for op in g.ops:
    for t in op.outputs:
        id2name[t.id] = f"{op.name}/{t.name}"

for i in range(8):
    feed = {h["input_ids"]: pool[i % 4], h["labels"]: lpool[i % 4]}
    if i < cap_at:
        lv = tr.run_step({t: v.to(dev) for t, v in feed.items()})
    elif i == cap_at:
        if variant == "keep":
            # capture while KEEPING every intermediate: maps the first
            # non-finite tensor to its op; if the NaN vanishes, the bug is
            # capture-pool tensor lifetime/aliasing
            for t, v in feed.items():
                tr._static_feeds[t] = v.to(dev).clone()
            torch.cuda.synchronize()
            cg = torch.cuda.CUDAGraph()
            with torch.cuda.graph(cg):
                lv_out = g.run([h["loss"], h["train_op"]],
                               dict(tr._static_feeds), ctx=tr.ctx,
                               keep_values=kept)[0]
            tr._cuda_graph = cg
            tr._loss_out = lv_out
            lv = tr.replay()
        else:
            tr.capture(feed)
            lv = tr.replay()
    else:
        for t, v in feed.items():
            tr._static_feeds[t].copy_(v, non_blocking=True)
        lv = tr.replay()
    torch.cuda.synchronize()
    print(f"[{variant} {model} cap@{cap_at}] step {i}: "
          f"loss={float(lv.float()):.4f}", flush=True)
    if i >= cap_at:
        bad = check(f"after step {i}")
        if kept:
            first_bad = None
            for op in g.ops:           # creation order ~ execution order
                for t in op.outputs:
                    v = kept.get(t.id)
                    if v is not None and v.is_floating_point() and \
                            not torch.isfinite(v.float()).all():
                        first_bad = id2name[t.id]
                        break
                if first_bad:
                    break
            print(f"  first non-finite kept tensor: {first_bad}",
                  flush=True)
        if bad:
            break
