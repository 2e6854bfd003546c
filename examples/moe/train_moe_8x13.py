#!/usr/bin/env python3
"""BASELINE config 4: GPT-MoE 8 x 1.3B experts with HetuMoE hierarchical
all-to-all over xGMI (reference HetuMoE, arXiv:2203.14685).

Run (8 GPUs, expert parallelism over the node):
  HETU_AMD_MOE_NODE_SIZE=4 python -m torch.distributed.run --nnodes=1 \
    --nproc-per-node 8 --master-addr 127.0.0.1 \
    examples/moe/train_moe_8x13.py

HETU_AMD_MOE_NODE_SIZE splits the EP group into `node_size`-wide islands
for the 3-phase hierarchical a2a (intra-island gather -> layout transform
-> inter-island a2a -> reverse); on one fully-connected xGMI node the flat
a2a is usually best — the hierarchical path is for multi-node EP.
MODEL=gpt-tiny runs a smoke config on CPU.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.engine.trainer import Trainer  # noqa: E402
from hetu_amd.models.gpt import GPT_CONFIGS, build_gpt_train_graph  # noqa
from hetu_amd.nn.parallel import ParallelSpec  # noqa: E402
from hetu_amd.parallel.comm import comm_backend  # noqa: E402


def main():
    comm = comm_backend()
    device = comm.device
    name = os.environ.get("MODEL", "gpt-moe-8x1.3b")
    cfg = GPT_CONFIGS[name]
    S = int(os.environ.get("SEQ_LEN", "2048" if "1.3b" in name else "32"))
    B = int(os.environ.get("MICRO_BATCH", "4" if "1.3b" in name else "2"))
    spec = ParallelSpec(dp=comm.world_size) if comm.world_size > 1 else \
        ParallelSpec()
    torch.manual_seed(1234 + comm.rank)
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=torch.bfloat16
                                 if device.type == "cuda"
                                 else torch.float32,
                                 lr=1e-4, spec=spec)
    tr = Trainer(g, h, device)
    steps = int(os.environ.get("STEPS", "20"))
    t0 = time.time()
    for step in range(steps):
        ids = torch.randint(0, cfg.vocab, (B, S), device=device)
        lab = torch.randint(0, cfg.vocab, (B * S,), device=device)
        loss = tr.step({h["input_ids"]: ids, h["labels"]: lab})
        if comm.rank == 0 and step % 5 == 0:
            tok = B * S * comm.world_size * (step + 1)
            print(f"step {step} loss {float(loss.float()):.4f} "
                  f"{tok / (time.time() - t0):.0f} tok/s")


if __name__ == "__main__":
    main()
