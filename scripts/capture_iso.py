#!/usr/bin/env python3
"""Isolate capture-replay corruption: (a) fused Adam alone under hipGraph,
(b) bias-grad reduce (bf16 column sum) alone, (c) adam via AdamStepOp with
the pinned bias-correction pattern."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
import hetu_amd.ops.functional as F  # noqa: E402

dev = torch.device("cuda", 0)
torch.manual_seed(0)


def test_adam_capture():
    N = 4096
    master = torch.randn(N, device=dev)
    m = torch.zeros(N, device=dev)
    v = torch.zeros(N, device=dev)
    out16 = torch.empty(N, dtype=torch.bfloat16, device=dev)
    grad = torch.randn(N, dtype=torch.bfloat16, device=dev)
    bc_host = torch.empty(2, dtype=torch.float32, pin_memory=True)
    bc_dev = torch.empty(2, dtype=torch.float32, device=dev)
    b1, b2 = 0.9, 0.999
    step = 1
    bc_host[0] = 1 - b1 ** step
    bc_host[1] = 1 - b2 ** step
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        bc_dev.copy_(bc_host, non_blocking=True)
        F.adam_step(master, grad, m, v, 1e-3, b1, b2, 1e-8, 0.0, step,
                    out16, bc_dev)
    for step in range(1, 6):
        bc_host[0] = 1 - b1 ** step
        bc_host[1] = 1 - b2 ** step
        g.replay()
        torch.cuda.synchronize()
        ok = (torch.isfinite(master).all() and torch.isfinite(m).all()
              and torch.isfinite(v).all() and torch.isfinite(out16.float()).all())
        print(f"adam replay {step}: finite={bool(ok)} bc_dev={bc_dev.tolist()}"
              f" m0={m[0].item():.5f} master0={master[0].item():.5f}",
              flush=True)


def test_reduce_capture():
    gy = torch.randn(8192, 1024, dtype=torch.bfloat16, device=dev)
    ref = gy.reshape(-1, 1024).sum(0)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        out = gy.reshape(-1, 1024).sum(0)
    for i in range(4):
        g.replay()
        torch.cuda.synchronize()
        err = (out.float() - ref.float()).abs().max().item()
        print(f"reduce replay {i}: finite={bool(torch.isfinite(out.float()).all())} "
              f"err={err:.4e}", flush=True)


def test_adamop_capture():
    from hetu_amd.graph.ops.optim import AdamStepOp
    from hetu_amd.graph.op import Op
    op_if = AdamStepOp()

    class FakeOp:
        attrs = {"lr": 1e-3, "beta1": 0.9, "beta2": 0.999, "eps": 1e-8}
    param = torch.randn(4096, dtype=torch.bfloat16, device=dev)
    grad = torch.randn(4096, dtype=torch.bfloat16, device=dev)
    # step 0 eager (state init happens outside capture, like the trainer)
    op_if.compute(FakeOp(), [param, grad], None)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        op_if.compute(FakeOp(), [param, grad], None)
    for s in range(3, 7):
        AdamStepOp.set_replay_step(s)
        g.replay()
        torch.cuda.synchronize()
        st = op_if.state
        ok = (torch.isfinite(st["master"]).all()
              and torch.isfinite(st["m"]).all()
              and torch.isfinite(st["v"]).all()
              and torch.isfinite(param.float()).all())
        print(f"adamop replay step={s}: finite={bool(ok)} "
              f"bc_dev={st['bc_dev'].tolist()}", flush=True)


if __name__ == "__main__":
    test_adam_capture()
    test_reduce_capture()
    test_adamop_capture()
