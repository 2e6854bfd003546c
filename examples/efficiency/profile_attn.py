"""Attention micro-benchmark (reference examples/efficiency/profile_attn.py):
time the in-tree CDNA4 FA2 kernels against torch SDPA across shapes.

GPU:  python examples/efficiency/profile_attn.py
CPU:  python examples/efficiency/profile_attn.py --allow-cpu   (shape check
      only: the HIP kernels need an MI355X, so CPU times torch SDPA alone)

Prints one line per (B, H, S, D) config: fwd / fwd+bwd ms and achieved
TFLOP/s (causal attention FLOPs = 2*2*B*H*S^2*D for fwd, ~2.5x for bwd).
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

SHAPES = [  # (batch, heads, seq, head_dim)
    (4, 32, 2048, 128),
    (2, 32, 4096, 128),
    (1, 32, 8192, 128),
]


def _time(fn, warmup=3, iters=10, cuda=True):
    for _ in range(warmup):
        fn()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if cuda:
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--allow-cpu", action="store_true")
    ap.add_argument("--iters", type=int, default=10)
    args = ap.parse_args()
    cuda = torch.cuda.is_available()
    if not cuda and not args.allow_cpu:
        sys.exit("no GPU; pass --allow-cpu for a shape-check run")
    dev = torch.device("cuda" if cuda else "cpu")
    dtype = torch.bfloat16 if cuda else torch.float32
    shapes = SHAPES if cuda else [(1, 2, 256, 128)]

    from hetu_amd.ops import functional as F
    for (B, H, S, D) in shapes:
        q, k, v = (torch.randn(B, H, S, D, device=dev, dtype=dtype)
                   for _ in range(3))
        flops_fwd = 4 * B * H * S * S * D / 2          # causal
        # torch SDPA baseline
        ms_sdpa = _time(lambda: torch.nn.functional
                        .scaled_dot_product_attention(q, k, v,
                                                      is_causal=True),
                        iters=args.iters, cuda=cuda)
        line = (f"B{B} H{H} S{S} D{D}: "
                f"sdpa fwd {ms_sdpa:7.2f} ms "
                f"({flops_fwd / ms_sdpa / 1e9:6.1f} TF/s)")
        if cuda:
            o, lse = F.flash_attn_fwd(q, k, v, causal=True)
            ms_fa = _time(lambda: F.flash_attn_fwd(q, k, v, causal=True),
                          iters=args.iters, cuda=cuda)
            do = torch.randn_like(o)
            ms_bwd = _time(lambda: F.flash_attn_bwd(
                do, q, k, v, o, lse, causal=True),
                iters=args.iters, cuda=cuda)
            line += (f" | fa2 fwd {ms_fa:7.2f} ms "
                     f"({flops_fwd / ms_fa / 1e9:6.1f} TF/s)"
                     f" | fa2 bwd {ms_bwd:7.2f} ms")
        else:
            o, lse = F.flash_attn_fwd(q, k, v, causal=True)
            ref = torch.nn.functional.scaled_dot_product_attention(
                q, k, v, is_causal=True)
            err = (o - ref).abs().max().item()
            line += f" | fa fallback max-err {err:.2e}"
        print(line)
    print("PROFILE_ATTN_OK")


if __name__ == "__main__":
    main()
