import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hetu_amd.ops.functional as F
dev = torch.device("cuda", 0)
e = F.ext()
def tryit(tag, T, H, cu_list, causal):
    q = torch.randn(T, H, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn_like(q); v = torch.randn_like(q)
    cu = torch.tensor(cu_list, dtype=torch.int32, device=dev)
    o, l = e.flash_attn_varlen_fwd(q, k, v, cu, T, causal, 0.088)
    torch.cuda.synchronize()
    print(f"{tag}: ok mean={o.float().abs().mean().item():.4f}", flush=True)

tryit("one-seg H1 nc", 64, 1, [0, 64], False)
tryit("one-seg H1 c", 64, 1, [0, 64], True)
tryit("one-seg H2 c", 64, 2, [0, 64], True)
tryit("two-seg H1 c", 64, 1, [0, 32, 64], True)
tryit("two-seg H2 c", 64, 2, [0, 32, 64], True)
