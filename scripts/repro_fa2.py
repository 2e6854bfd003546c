import sys, os, torch, math
sys.path.insert(0, "/root/repo")
import hetu_amd.ops.functional as F
dev = torch.device("cuda", 0)
for S in (300, 256, 320):
    torch.manual_seed(0)
    B,H,D = 1,2,128
    q = torch.randn(B,H,S,D, dtype=torch.bfloat16, device=dev)
    k = torch.randn_like(q); v = torch.randn_like(q)
    scale = 1/math.sqrt(D)
    o, lse = F.flash_attn_fwd(q, k, v, True, scale)
    orf, lser = F._attn_ref_fwd(q.cpu().float(), k.cpu().float(), v.cpu().float(), True, scale)
    do = torch.randn_like(o)
    dq, dk, dv = F.flash_attn_bwd(do, q, k, v, o, lse, True, scale)
    dqr, dkr, dvr = F._attn_ref_bwd(do.cpu().float(), q.cpu().float(), k.cpu().float(), v.cpu().float(), lser, True, scale)
    for name, a, r in (("dq",dq,dqr),("dk",dk,dkr),("dv",dv,dvr)):
        err = (a.cpu().float()-r).abs()
        mx = err.max()
        loc = (err==mx).nonzero()[0].tolist()
        print(f"S={S} {name} max_err={mx:.4f} at {loc} refmax={r.abs().max():.3f}")
