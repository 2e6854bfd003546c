#!/usr/bin/env python3
"""Standalone repro probe: does a captured 3-D bf16 .sum(0) corrupt
neighboring allocations on replay?"""
import torch
dev = torch.device("cuda", 0)
torch.manual_seed(0)
# surround the suspect's working set with canaries
pre = [torch.full((4096,), 7.0, device=dev) for _ in range(64)]
x = torch.randn(2, 256, 256, dtype=torch.bfloat16, device=dev)
mid = [torch.full((4096,), 7.0, device=dev) for _ in range(64)]
ref = x.float().sum(0).sum(0)
torch.cuda.synchronize()
g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    outs = []
    for _ in range(24):                      # 24x like the GPT block count
        outs.append(x.sum(0).sum(0))
post = [torch.full((4096,), 7.0, device=dev) for _ in range(64)]
for r in range(4):
    g.replay()
    torch.cuda.synchronize()
    badc = sum(1 for c in pre + mid + post if not (c == 7.0).all())
    err = max((o.float() - ref).abs().max().item() for o in outs)
    fin = all(torch.isfinite(o.float()).all() for o in outs)
    print(f"replay {r}: canaries_bad={badc} max_err={err:.3e} finite={fin}",
          flush=True)
