"""hipBLASLt layout sweep for the 7B GEMM shapes: is x @ W.T (current
weight layout [N,K]) the fastest variant, or do NN / TN layouts win?"""
import sys

import torch

SHAPES = [  # (M, K, N) as in y[M,N] = x[M,K] @ W
    (32768, 4096, 12288),   # qkv
    (32768, 4096, 4096),    # attn proj
    (32768, 4096, 16384),   # mlp fc
    (32768, 16384, 4096),   # mlp proj
    (32768, 4096, 50304),   # lm head
]


def bench(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    torch.manual_seed(0)
    dev = "cuda"
    print(f"{'shape':>22} {'NT(x@W.T)':>10} {'NN(x@Wt)':>10} "
          f"{'TN(xT.T@)':>10} {'linear':>10}  TF/s")
    for M, K, N in SHAPES:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)   # [N,K]
        wt = w.t().contiguous()                                    # [K,N]
        xt = x.t().contiguous()                                    # [K,M]
        fl = 2.0 * M * K * N
        t_nt = bench(lambda: x @ w.t())
        t_nn = bench(lambda: x @ wt)
        t_tn = bench(lambda: xt.t() @ wt)
        t_lin = bench(lambda: torch.nn.functional.linear(x, w))
        tf = [fl / t / 1e9 for t in (t_nt, t_nn, t_tn, t_lin)]
        print(f"{(M,K,N)!s:>22} {t_nt:10.3f} {t_nn:10.3f} {t_tn:10.3f} "
              f"{t_lin:10.3f}  {tf[0]:.0f}/{tf[1]:.0f}/{tf[2]:.0f}/"
              f"{tf[3]:.0f}")
        # wgrad variant: gy[M,N], dW = gy.T @ x  ([N,M]x[M,K])
        gy = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        t_w1 = bench(lambda: gy.t() @ x)
        t_w2 = bench(lambda: (x.t() @ gy).t())
        print(f"{'  wgrad':>22} {t_w1:10.3f} {t_w2:10.3f}   "
              f"{fl / t_w1 / 1e9:.0f}/{fl / t_w2 / 1e9:.0f} TF/s")


if __name__ == "__main__":
    main()
