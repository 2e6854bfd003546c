import sys, torch, math
sys.path.insert(0, "/root/repo")
import hetu_amd.ops.functional as F
dev = torch.device("cuda", 0)
for S, causal in ((256, True), (256, False), (512, True)):
    torch.manual_seed(0)
    B,H,D = 2,4,128
    q = torch.randn(B,H,S,D, dtype=torch.bfloat16, device=dev)
    k = torch.randn_like(q); v = torch.randn_like(q)
    scale = 1/math.sqrt(D)
    o, lse = F.flash_attn_fwd(q, k, v, causal, scale)
    orf, lser = F._attn_ref_fwd(q.cpu().float(), k.cpu().float(), v.cpu().float(), causal, scale)
    do = torch.randn_like(o)
    dq, dk, dv = F.flash_attn_bwd(do, q, k, v, o, lse, causal, scale)
    dqr, dkr, dvr = F._attn_ref_bwd(do.cpu().float(), q.cpu().float(), k.cpu().float(), v.cpu().float(), lser, causal, scale)
    for name, a, r in (("dq",dq,dqr),("dk",dk,dkr),("dv",dv,dvr)):
        err = (a.cpu().float()-r).abs()
        mx = err.max()
        locs = (err > max(0.05, 0.05*r.abs().max())).nonzero()
        print(f"S={S} c={causal} {name} max={mx:.4f} bad={len(locs)}", locs[:4].tolist() if len(locs) else "")
