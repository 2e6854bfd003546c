#!/usr/bin/env python3
"""A/B: hipBLASLt (default) vs rocBLAS under hipGraph capture, and a
pure-torch control, to pin the capture-replay corruption."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

mode = sys.argv[1] if len(sys.argv) > 1 else "lt"
if mode == "rocblas":
    torch.backends.cuda.preferred_blas_library("cublas")   # rocBLAS on ROCm


def hetu_run():
    from hetu_amd.engine.trainer import Trainer
    from hetu_amd.models.gpt import GPT_CONFIGS, build_gpt_train_graph
    cfg = GPT_CONFIGS["gpt2-345m"]
    B, S = 8, 1024
    torch.manual_seed(1234)
    dev = torch.device("cuda", 0)
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=torch.bfloat16, lr=1e-4)
    tr = Trainer(g, h, dev)
    pool = [torch.randint(0, cfg.vocab, (B, S), device=dev)
            for _ in range(4)]
    lp = [torch.randint(0, cfg.vocab, (B * S,), device=dev)
          for _ in range(4)]
    for i in range(6):
        lv = tr.step({h["input_ids"]: pool[i % 4], h["labels"]: lp[i % 4]})
        torch.cuda.synchronize()
        print(f"[{mode}] step {i}: loss={float(lv.float()):.4f}",
              flush=True)


def torch_control():
    """Same shape of work in plain torch: linear+bias tower, bf16, manual
    fp32-master adam, captured; replayed 5x."""
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    H = 1024
    layers = [torch.nn.Linear(H, 4 * H), torch.nn.Linear(4 * H, H)]
    mods = torch.nn.Sequential(*[m for m in layers]).to(dev, torch.bfloat16)
    masters = [p.detach().float().clone() for p in mods.parameters()]
    ms = [torch.zeros_like(m) for m in masters]
    vs = [torch.zeros_like(m) for m in masters]
    x = torch.randn(8 * 1024, H, dtype=torch.bfloat16, device=dev)

    def step():
        y = mods(x)
        loss = (y.float() ** 2).mean()
        grads = torch.autograd.grad(loss, list(mods.parameters()))
        with torch.no_grad():
            for p, g, mm, vv, ma in zip(mods.parameters(), grads, ms, vs,
                                        masters):
                mm.mul_(0.9).add_(g.float(), alpha=0.1)
                vv.mul_(0.999).addcmul_(g.float(), g.float(), value=0.001)
                ma.add_(mm / (vv.sqrt() + 1e-8), alpha=-1e-4)
                p.copy_(ma)
        return loss

    for i in range(2):
        lv = step()
    torch.cuda.synchronize()
    cg = torch.cuda.CUDAGraph()
    with torch.cuda.graph(cg):
        lv = step()
    for i in range(5):
        cg.replay()
        torch.cuda.synchronize()
        bad = [n for n, p in mods.named_parameters()
               if not torch.isfinite(p.float()).all()]
        badm = sum(0 if torch.isfinite(m).all() else 1 for m in ms + vs
                   + masters)
        print(f"[control] replay {i}: loss={float(lv.float()):.5f} "
              f"bad_params={bad} bad_states={badm}", flush=True)


if mode == "control":
    torch_control()
else:
    hetu_run()
