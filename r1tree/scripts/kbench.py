#!/usr/bin/env python3
"""Kernel micro-benchmarks on MI355X: TF/s / TB/s per hand kernel, with a
torch (rocBLAS/hipBLASLt or eager) comparison where applicable."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
import hetu_amd.ops.functional as F  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = torch.device("cuda", 0)
    print("== GEMM bf16 TN (C[M,N]=A[M,K]@W[N,K]^T) ==")
    for (M, N, K) in [(4096, 4096, 4096), (8192, 8192, 8192),
                      (16384, 4096, 4096), (16384, 16384, 4096),
                      (8192, 50304 // 128 * 128, 4096)]:
        a = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        fl = 2.0 * M * N * K
        t_hip = bench(lambda: F.ext().gemm_bf16(a, w, True))
        t_blas = bench(lambda: torch.matmul(a, w.t()))
        print(f"  {M}x{N}x{K}: hip {fl/t_hip/1e12:7.1f} TF | "
              f"hipBLASLt {fl/t_blas/1e12:7.1f} TF")

    print("== flash attention fwd/bwd (B,H,S,D) ==")
    for (B, H, S, D) in [(4, 32, 2048, 128), (1, 32, 8192, 128),
                         (8, 32, 4096, 128)]:
        q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev)
        k = torch.randn_like(q)
        v = torch.randn_like(q)
        scale = D ** -0.5
        # causal flops: ~0.5 * 4 * B*H*S^2*D
        fl = 0.5 * 4.0 * B * H * S * S * D
        t_f = bench(lambda: F.ext().flash_attn_fwd(q, k, v, True, scale))
        o, lse = F.ext().flash_attn_fwd(q, k, v, True, scale)
        do = torch.randn_like(o)
        t_b = bench(lambda: F.ext().flash_attn_bwd(do, q, k, v, o, lse,
                                                   True, scale), iters=10)
        try:
            t_sdpa = bench(lambda: torch.nn.functional.
                           scaled_dot_product_attention(q, k, v,
                                                        is_causal=True))
        except Exception:
            t_sdpa = float("nan")
        print(f"  B{B} H{H} S{S}: fwd {fl/t_f/1e12:6.1f} TF "
              f"(sdpa {fl/t_sdpa/1e12:6.1f}) | bwd {2.5*fl/t_b/1e12:6.1f} TF")

    print("== memory-bound kernels (TB/s) ==")
    x = torch.randn(16384, 4096, dtype=torch.bfloat16, device=dev)
    w1 = torch.randn(4096, dtype=torch.bfloat16, device=dev)
    t = bench(lambda: F.ext().rmsnorm_fwd(x, w1, 1e-6))
    print(f"  rmsnorm fwd 16k x 4k: {2*x.numel()*2/t/1e12:6.2f} TB/s")
    y, rstd = F.ext().rmsnorm_fwd(x, w1, 1e-6)
    t = bench(lambda: F.ext().rmsnorm_bwd(y, x, w1, rstd))
    print(f"  rmsnorm bwd: {4*x.numel()*2/t/1e12:6.2f} TB/s")
    big = torch.randn(8192, 16384, dtype=torch.bfloat16, device=dev)
    t = bench(lambda: F.ext().swiglu_fwd(big))
    print(f"  swiglu fwd 8k x 16k: {1.5*big.numel()*2/t/1e12:6.2f} TB/s")
    t = bench(lambda: torch.nn.functional.silu(big))  # torch reference pass
    print(f"  (torch silu same bytes: {2*big.numel()*2/t/1e12:6.2f} TB/s)")
    logits = torch.randn(8192, 50304, dtype=torch.bfloat16, device=dev)
    labels = torch.randint(0, 50304, (8192,), device=dev)
    t = bench(lambda: F.ext().softmax_ce_fwd(logits, labels, -100))
    print(f"  CE fwd 8k x 50k: {logits.numel()*2/t/1e12:6.2f} TB/s")
    p = torch.randn(1 << 26, device=dev)
    g = torch.randn(1 << 26, dtype=torch.bfloat16, device=dev)
    m = torch.zeros_like(p)
    vv = torch.zeros_like(p)
    o16 = torch.zeros(1 << 26, dtype=torch.bfloat16, device=dev)
    e = p.new_empty(0)
    t = bench(lambda: F.ext().adam_step(p, g, m, vv, 1e-4, 0.9, 0.999, 1e-8,
                                        0.0, 5, o16, e))
    bytes_ = p.numel() * (4 * 3 * 2 + 2 + 2)  # rw p/m/v + r g + w out16
    print(f"  fused adam 64M: {bytes_/t/1e12:6.2f} TB/s")


if __name__ == "__main__":
    main()
