"""Micro-timing for norm kernels at the 7B bench shape."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hetu_amd.ops.functional as F

def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us

R, D = 32768, 4096
x = torch.randn(R, D, dtype=torch.bfloat16, device="cuda")
w = torch.randn(D, dtype=torch.bfloat16, device="cuda")
b = torch.randn(D, dtype=torch.bfloat16, device="cuda")
dy = torch.randn(R, D, dtype=torch.bfloat16, device="cuda")
y, mean, rstd = F.layernorm_fwd(x, w, b, 1e-5)
print(f"ln fwd  {bench(lambda: F.layernorm_fwd(x, w, b, 1e-5)):8.1f} us")
print(f"ln bwd  {bench(lambda: F.layernorm_bwd(dy, x, w, mean, rstd)):8.1f} us")
yr, rs = F.rmsnorm_fwd(x, w, 1e-6)
print(f"rms fwd {bench(lambda: F.rmsnorm_fwd(x, w, 1e-6)):8.1f} us")
print(f"rms bwd {bench(lambda: F.rmsnorm_bwd(dy, x, w, rs)):8.1f} us")
E = F.ext()
print(f"ln bwd v2 {bench(lambda: E.layernorm_bwd2(dy, x, w, mean, rstd)):8.1f} us")
print(f"rms bwd v2 {bench(lambda: E.rmsnorm_bwd2(dy, x, w, rs)):8.1f} us")
# v2 numerics vs v1
a = E.layernorm_bwd(dy, x, w, mean, rstd)
b = E.layernorm_bwd2(dy, x, w, mean, rstd)
for n, (t1, t2) in zip(("dx","dw","db"), zip(a, b)):
    print("ln v2", n, (t1.float()-t2.float()).abs().max().item())
a = E.rmsnorm_bwd(dy, x, w, rs); b = E.rmsnorm_bwd2(dy, x, w, rs)
for n, (t1, t2) in zip(("dx","dw"), zip(a, b)):
    print("rms v2", n, (t1.float()-t2.float()).abs().max().item())
# numerics vs fp32 torch
ln = torch.nn.functional.layer_norm(x.float(), (D,), w.float().float(), b.float(), 1e-5)
print("ln fwd err", (y.float() - ln).abs().max().item())
