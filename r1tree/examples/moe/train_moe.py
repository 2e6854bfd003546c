#!/usr/bin/env python3
"""GPT-MoE training with expert parallelism + hierarchical all-to-all
(reference HetuMoE examples).

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
       --master-addr 127.0.0.1 examples/moe/train_moe.py
Set HETU_AMD_MOE_NODE_SIZE to enable the 3-phase hierarchical a2a.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.engine.runner import prepare_run_context  # noqa: E402
from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,  # noqa
                                  push_graph)
from hetu_amd.graph.ops import api as ht  # noqa: E402
from hetu_amd.graph.ops.optim import Adam  # noqa: E402
from hetu_amd.nn.moe import MoEMLP  # noqa: E402
from hetu_amd.nn.parallel import ParallelSpec  # noqa: E402
from hetu_amd.parallel.comm import comm_backend  # noqa: E402


def main():
    comm = comm_backend()
    device = comm.device
    spec = ParallelSpec(dp=comm.world_size) if comm.world_size > 1 else None
    N, H, F, E = 512, 256, 1024, 8
    g = DefineAndRunGraph("moe")
    push_graph(g)
    try:
        x = ht.placeholder((N, H), name="x",
                           ds=spec.ds_tokens(0) if spec else None,
                           device_group=spec.device_group if spec else None)
        tgt = ht.placeholder((N, H), name="tgt",
                             ds=spec.ds_tokens(0) if spec else None,
                             device_group=spec.device_group if spec else None)
        moe = MoEMLP(H, F, E, spec=spec, k=2, gate_type="topk")
        loss = ht.mse_loss(moe(x), tgt)
        train_op = Adam(lr=1e-3).minimize(loss)
    finally:
        pop_graph()
    ctx = prepare_run_context(g, device)
    torch.manual_seed(3 + comm.rank)
    for step in range(30):
        xd = torch.randn(N, H, device=device)
        td = torch.randn(N, H, device=device)
        lv, _ = g.run([loss, train_op], {x: xd, tgt: td}, ctx=ctx)
        if comm.rank == 0 and step % 10 == 0:
            print(f"step {step} loss {float(lv):.4f}")


if __name__ == "__main__":
    main()
