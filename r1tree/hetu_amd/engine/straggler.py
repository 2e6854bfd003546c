"""Malleus straggler detection + hetero re-balancing.

Reference parity: engine/straggler.py:20-75 (timed profile kernels, per-rank
logs) and engine/strategy.py:99 (the straggler-aware strategy model that
re-balances work).  MI355X-native: the probe is a fixed bf16 GEMM workload
timed with hipEvents; times allgather over gloo/RCCL; the re-balancer
assigns non-uniform per-rank micro-batch shares inversely proportional to
slowdown (the Malleus hetero-dp resolution; hetero pp splits reuse
PipelineSpec.partition_layers overrides)."""
from __future__ import annotations

import time
from typing import List, Optional

import torch

from ..parallel.comm import CommBackend, comm_backend


def profile_rank_speed(device: torch.device, iters: int = 8,
                       n: int = 2048) -> float:
    """Seconds for a fixed GEMM workload on this rank."""
    if device.type == "cuda":
        a = torch.randn(n, n, dtype=torch.bfloat16, device=device)
        b = torch.randn(n, n, dtype=torch.bfloat16, device=device)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            a = a @ b
        torch.cuda.synchronize()
        return time.perf_counter() - t0
    a = torch.randn(256, 256)
    b = torch.randn(256, 256)
    t0 = time.perf_counter()
    for _ in range(iters):
        a = a @ b
    return time.perf_counter() - t0


def gather_speeds(comm: Optional[CommBackend], my_time: float) -> List[float]:
    comm = comm or comm_backend()
    if comm.world_size <= 1:
        return [my_time]
    t = torch.tensor([my_time], device=comm.device
                     if comm.device.type == "cuda" else "cpu")
    out = comm.allgather(t, list(range(comm.world_size)), dim=0)
    return [float(x) for x in out]


def detect_stragglers(times: List[float], threshold: float = 1.5
                      ) -> List[int]:
    """Ranks slower than threshold x median."""
    med = sorted(times)[len(times) // 2]
    return [i for i, t in enumerate(times) if t > threshold * med]


def rebalance_micro_batches(times: List[float], total_mb: int
                            ) -> List[int]:
    """Non-uniform per-rank micro-batch counts ~ 1/time (Malleus hetero
    data assignment); every rank keeps >= 1 and the total is preserved."""
    inv = [1.0 / max(t, 1e-9) for t in times]
    s = sum(inv)
    raw = [v / s * total_mb for v in inv]
    out = [max(1, int(r)) for r in raw]
    # distribute the remainder to the fastest ranks
    order = sorted(range(len(times)), key=lambda i: times[i])
    i = 0
    while sum(out) < total_mb:
        out[order[i % len(order)]] += 1
        i += 1
    while sum(out) > total_mb:
        j = order[-1 - (i % len(order))]
        if out[j] > 1:
            out[j] -= 1
        i += 1
    return out
