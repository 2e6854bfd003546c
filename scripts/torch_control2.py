#!/usr/bin/env python3
"""Pure-torch control: mixed fp32-LN/bf16-linear tower, full train step
captured in one hipGraph (fwd+bwd+adam), manual backward via saved ctx to
avoid autograd-capture issues."""
import torch
dev = torch.device("cuda", 0)
torch.manual_seed(0)
L, H, R = 12, 256, 512

lnw = [torch.ones(H, device=dev, requires_grad=False) for _ in range(L)]
w1 = [(torch.randn(2*H, H, device=dev) * 0.02).bfloat16() for _ in range(L)]
w2 = [(torch.randn(H, 2*H, device=dev) * 0.02).bfloat16() for _ in range(L)]
params = lnw + w1 + w2
masters = [p.float().clone() for p in params]
ms = [torch.zeros_like(p, dtype=torch.float32) for p in params]
vs = [torch.zeros_like(p, dtype=torch.float32) for p in params]
x = torch.randn(R, H, dtype=torch.bfloat16, device=dev)

def step():
    # forward (save minimal ctx)
    saves = []
    cur = x
    for i in range(L):
        xf = cur.float()
        mu = xf.mean(-1, keepdim=True); var = xf.var(-1, unbiased=False, keepdim=True)
        rstd = torch.rsqrt(var + 1e-5)
        xh = (xf - mu) * rstd
        y = (xh * lnw[i]).bfloat16()
        h1 = y @ w1[i].t()
        a = torch.nn.functional.gelu(h1, approximate="tanh")
        o = a @ w2[i].t()
        saves.append((cur, xh, rstd, y, h1, a))
        cur = cur + o
    loss = (cur.float() ** 2).mean()
    # backward
    dcur = (2.0 / cur.numel()) * cur.float()
    g_lnw = [None]*L; g_w1 = [None]*L; g_w2 = [None]*L
    for i in reversed(range(L)):
        cin, xh, rstd, y, h1, a = saves[i]
        do = dcur.bfloat16()
        g_w2[i] = (do.t() @ a).float()
        da = do @ w2[i]
        h1f = h1.float()
        c = 0.7978845608; aa = 0.044715
        t = torch.tanh(c * (h1f + aa * h1f**3))
        dgelu = 0.5*(1+t) + 0.5*h1f*(1-t*t)*c*(1+3*aa*h1f*h1f)
        dh1 = (da.float() * dgelu).bfloat16()
        g_w1[i] = (dh1.t() @ y).float()
        dy = dh1 @ w1[i]
        g_lnw[i] = (dy.float() * xh).sum(0)
        dxh = dy.float() * lnw[i]
        n = H
        dxf = rstd * (dxh - dxh.mean(-1, keepdim=True) - xh * (dxh * xh).mean(-1, keepdim=True))
        dcur = dcur + dxf
    grads = g_lnw + g_w1 + g_w2
    with torch.no_grad():
        for p, g, mm, vv, ma in zip(params, grads, ms, vs, masters):
            mm.mul_(0.9).add_(g, alpha=0.1)
            vv.mul_(0.999).addcmul_(g, g, value=0.001)
            ma.add_(mm / (vv.sqrt() + 1e-8), alpha=-1e-4)
            p.copy_(ma.to(p.dtype))
    return loss

for _ in range(2):
    lv = step()
torch.cuda.synchronize()
print(f"eager loss {lv.item():.5f}", flush=True)
cg = torch.cuda.CUDAGraph()
with torch.cuda.graph(cg):
    lv = step()
for r in range(4):
    cg.replay()
    torch.cuda.synchronize()
    bad = sum(1 for t in params + ms + vs + masters
              if not torch.isfinite(t.float()).all())
    print(f"[torch-control] replay {r}: loss={lv.item():.5f} bad={bad}",
          flush=True)
