#!/usr/bin/env python3
"""A/B microbench: flash-attention fwd v1 vs v2 (+ bwd), TF/s on random data."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hetu_amd.ops.functional as F

dev = torch.device("cuda", 0)
B, H, Hkv, S, D = 4, 32, 32, 2048, 128
for a in sys.argv[1:]:
    k, v = a.split("=")
    if k == "S": S = int(v)
    if k == "B": B = int(v)
torch.manual_seed(0)
q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
kk = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
vv = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev)
scale = D ** -0.5
ext = F.ext()

def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

for causal in (True, False):
    flops = 4 * B * H * S * S * D * (0.5 if causal else 1.0)
    o, lse = ext.flash_attn_fwd(q, kk, vv, causal, scale)
    t = bench(lambda: ext.flash_attn_fwd(q, kk, vv, causal, scale))
    print(f"fwd causal={causal}: {t*1e3:.3f} ms  {flops/t/1e12:.0f} TF/s")
    if hasattr(ext, "flash_attn_fwd_v3"):
        o3, lse3 = ext.flash_attn_fwd_v3(q, kk, vv, causal, scale)
        err = (o3.float() - o.float()).abs().max().item()
        el = (lse3 - lse).abs().max().item()
        t3 = bench(lambda: ext.flash_attn_fwd_v3(q, kk, vv, causal, scale))
        print(f"fwd v3 causal={causal}: {t3*1e3:.3f} ms  "
              f"{flops/t3/1e12:.0f} TF/s  (err {err:.3e} lse {el:.3e})")
    do = torch.randn_like(o)
    tb = bench(lambda: ext.flash_attn_bwd(do, q, kk, vv, o, lse, causal,
                                          scale), iters=10)
    print(f"bwd causal={causal}: {tb*1e3:.3f} ms  {2.5*flops/tb/1e12:.0f} TF/s")
    if hasattr(ext, "flash_attn_bwd_v3") and H == Hkv:
        ref = ext.flash_attn_bwd(do, q, kk, vv, o, lse, causal, scale)
        g3 = ext.flash_attn_bwd_v3(do, q, kk, vv, o, lse, causal, scale)
        errs = [(a.float() - b.float()).abs().max().item()
                for a, b in zip(g3, ref)]
        t3 = bench(lambda: ext.flash_attn_bwd_v3(do, q, kk, vv, o, lse,
                                                 causal, scale), iters=10)
        print(f"bwd v3 causal={causal}: {t3*1e3:.3f} ms  "
              f"{2.5*flops/t3/1e12:.0f} TF/s  (err dq/dk/dv "
              f"{errs[0]:.2e}/{errs[1]:.2e}/{errs[2]:.2e})")
