#!/usr/bin/env python3
"""Malleus heterogeneous data parallelism: pipelines with DIFFERENT tensor
parallelism train one model (reference examples/malleus + engine/strategy).

3 ranks: pipeline A = tp2 on ranks {0,1} (fast GPUs), pipeline B = tp1 on
rank {2} (straggler); the batch splits 2:1 by speed and gradients reduce
across pipelines with split-allreduce groups
(hetu_amd/parallel/hetero.py, DistributedStatesUnion).

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 3 \
       --master-addr 127.0.0.1 examples/malleus/hetero_train.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.engine.runner import prepare_run_context  # noqa: E402
from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph  # noqa
from hetu_amd.nn.parallel import ParallelSpec  # noqa: E402
from hetu_amd.parallel.comm import comm_backend  # noqa: E402
from hetu_amd.parallel.hetero import HeteroSpec  # noqa: E402


def main():
    comm = comm_backend()
    assert comm.world_size == 3, "demo uses 3 ranks (tp2 + tp1 pipelines)"
    device = comm.device
    hs = HeteroSpec(pipelines=[
        ParallelSpec(dp=1, tp=2, device_group=[0, 1]),
        ParallelSpec(dp=1, tp=1, device_group=[2])],
        weights=[2 / 3, 1 / 3])
    pi = hs.my_pipeline()
    spec = hs.pipelines[pi]
    cfg = GPTConfig(n_layer=4, n_head=4, n_kv_head=4, hidden=256,
                    ffn_hidden=1024, vocab=50304, max_seq=512)
    gb = 12
    my_rows = hs.micro_batches(gb)[pi]
    S = 512
    g, h = build_gpt_train_graph(cfg, micro_batch=my_rows, seq_len=S,
                                 dtype=torch.float32, lr=1e-4, spec=spec,
                                 hetero=hs)
    ctx = prepare_run_context(g, device)
    torch.manual_seed(99)        # same global batch on every rank
    for step in range(int(os.environ.get("STEPS", "10"))):
        ids = torch.randint(0, cfg.vocab, (gb, S))
        lab = torch.randint(0, cfg.vocab, (gb, S))
        rows = slice(0, my_rows) if pi == 0 else slice(gb - my_rows, gb)
        lv, _ = g.run([h["loss"], h["train_op"]],
                      {h["input_ids"]: ids[rows].to(device),
                       h["labels"]: lab[rows].reshape(-1).to(device)},
                      ctx=ctx)
        if comm.rank in (0, 2):
            print(f"[pipe {pi}] step {step} loss {float(lv):.4f}")


if __name__ == "__main__":
    main()
