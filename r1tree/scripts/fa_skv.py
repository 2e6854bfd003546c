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
B,H,S,D = 4,32,2048,128
q = torch.randn(B,H,S,D, dtype=torch.bfloat16, device=dev)
for skv in (2048, 1024, 512, 256):
    k = torch.randn(B,H,skv,D, dtype=torch.bfloat16, device=dev)
    v = torch.randn_like(k)
    t = bench(lambda: ext.flash_attn_fwd(q,k,v,False,0.0883))
    print(f"Skv={skv}: {t*1e6:7.1f}us")
