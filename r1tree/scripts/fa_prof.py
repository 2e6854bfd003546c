#!/usr/bin/env python3
"""Run flash-attention fwd/bwd in a loop for rocprof profiling."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hetu_amd.ops.functional as F

dev = torch.device("cuda", 0)
B, H, S, D = 4, 32, 4096, 128
q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
k = torch.randn_like(q); v = torch.randn_like(q)
scale = D ** -0.5
o, lse = F.ext().flash_attn_fwd(q, k, v, True, scale)
do = torch.randn_like(o)
for _ in range(int(os.environ.get("FA_ITERS", "10"))):
    o, lse = F.ext().flash_attn_fwd(q, k, v, True, scale)
    F.ext().flash_attn_bwd(do, q, k, v, o, lse, True, scale)
torch.cuda.synchronize()
print("done")
