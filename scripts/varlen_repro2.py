import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hetu_amd.ops.functional as F
dev = torch.device("cuda", 0)
e = F.ext()
for S in (32, 64, 128, 256):
    q = torch.randn(1, 2, S, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn_like(q); v = torch.randn_like(q)
    o, l = e.flash_attn_fwd(q, k, v, True, 0.088)
    torch.cuda.synchronize()
    print(f"dense S={S} ok {o.float().abs().mean().item():.4f}", flush=True)
