#!/usr/bin/env python3
"""Shape-specific at::native column-sum under hipGraph replay."""
import torch
dev = torch.device("cuda", 0)
torch.manual_seed(0)
for R, C in [(512, 768), (512, 1024), (512, 4096), (32768, 1024),
             (512, 256), (131072, 16384)]:
    x = torch.randn(R, C, dtype=torch.bfloat16, device=dev)
    eager = x.sum(0)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        out = x.sum(0)
    diffs = []
    for r in range(3):
        g.replay()
        torch.cuda.synchronize()
        diffs.append((out.float() - eager.float()).abs().max().item())
    print(f"R{R} C{C}: replay-vs-eager max|diff| = "
          + " ".join(f"{d:.3e}" for d in diffs), flush=True)
