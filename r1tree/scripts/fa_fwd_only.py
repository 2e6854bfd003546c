import sys, torch
sys.path.insert(0, "/root/repo")
import hetu_amd.ops.functional as F
dev = torch.device("cuda", 0)
ext = F.ext()
import os
causal = os.environ.get("CAUSAL", "1") == "1"
B,H,S,D = 4,32,2048,128
q = torch.randn(B,H,S,D, dtype=torch.bfloat16, device=dev)
k, v = torch.randn_like(q), torch.randn_like(q)
for _ in range(20):
    ext.flash_attn_fwd(q,k,v,causal,0.0883)
torch.cuda.synchronize()
print("done")
