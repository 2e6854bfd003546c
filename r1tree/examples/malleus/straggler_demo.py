#!/usr/bin/env python3
"""Malleus-style straggler handling demo (reference examples/malleus).

Each rank profiles a fixed GEMM workload; rank speeds are allgathered, a
synthetic straggler is injected on rank 0 (sleep per step, standing in for
the reference's workload_heavy_compute.cu occupancy kernels), and the
planner re-balances per-rank micro-batch shares.  Gradients are then
partial-reduced so fast ranks never block on the straggler.

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
       --master-addr 127.0.0.1 examples/malleus/straggler_demo.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.engine.straggler import (detect_stragglers,  # noqa: E402
                                       gather_speeds, profile_rank_speed,
                                       rebalance_micro_batches)
from hetu_amd.parallel.comm import comm_backend  # noqa: E402
from hetu_amd.parallel.preduce import PartialReduce  # noqa: E402
from hetu_amd.rpc.kv_store import KVStore  # noqa: E402


def main():
    comm = comm_backend()
    rank, ws = comm.rank, comm.world_size
    device = comm.device
    slowdown = 2.0 if rank == 0 and ws > 1 else 1.0   # synthetic straggler

    t = profile_rank_speed(device) * slowdown
    speeds = gather_speeds(comm, t)
    stragglers = detect_stragglers(speeds, threshold=1.5)
    shares = rebalance_micro_batches(speeds, total_mb=8 * ws)
    if rank == 0:
        print(f"speeds={['%.4f' % s for s in speeds]} "
              f"stragglers={stragglers} shares={shares}")

    # training loop with partial reduce: fast ranks sync among whoever
    # arrives within the window
    model = torch.nn.Linear(64, 64).to(device)
    pr = None
    if ws > 1:
        kv = KVStore("127.0.0.1", 29717, is_server=(rank == 0),
                     world_size=ws)
        pr = PartialReduce(kv, comm, min_size=max(2, ws - len(stragglers)),
                           window_s=0.3)
    for step in range(4):
        if rank in stragglers:
            time.sleep(0.2)                      # injected slowness
        x = torch.randn(shares[rank], 64, device=device)
        loss = model(x).pow(2).mean()
        loss.backward()
        if pr is not None:
            for p in model.parameters():
                p.grad, members = pr.preduce(p.grad, token=f"g{step}")
            if rank == 0:
                print(f"step {step}: reduced over ranks {members}")
        with torch.no_grad():
            for p in model.parameters():
                p -= 0.01 * p.grad
                p.grad = None


if __name__ == "__main__":
    main()
