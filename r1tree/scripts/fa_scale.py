import sys, time, torch
sys.path.insert(0, "/root/repo")
import hetu_amd.ops.functional as F
dev = torch.device("cuda", 0)
ext = F.ext()
def bench(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters
D = 128
for (B, H, S) in ((4,32,512),(4,32,1024),(4,32,2048),(16,32,2048),(1,8,2048)):
    q = torch.randn(B,H,S,D, dtype=torch.bfloat16, device=dev)
    k, v = torch.randn_like(q), torch.randn_like(q)
    tc = bench(lambda: ext.flash_attn_fwd(q,k,v,True,0.088))
    tn = bench(lambda: ext.flash_attn_fwd(q,k,v,False,0.088))
    fl = 4*B*H*S*S*D
    print(f"B{B} H{H} S{S}: causal {tc*1e6:7.1f}us ({fl*0.5/tc/1e12:5.0f} TF) "
          f"non {tn*1e6:7.1f}us ({fl/tn/1e12:5.0f} TF) ratio {tc/tn:.2f} "
          f"blocks={B*H*((S+127)//128)}")
