#!/usr/bin/env python3
"""HotSPa-style multi-strategy training (reference examples/hotspa).

Sequence lengths are bucketed; each bucket trains under its best parallel
strategy and the trainer hot-switches parameters (+ Adam state) between
strategies with one batched p2p when the bucket changes
(hetu_amd/parallel/switch.py).

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
       --master-addr 127.0.0.1 examples/hotspa/hot_switch_train.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.engine.hot_switch_trainer import (HotSwitchTrainer,  # noqa
                                                bucket_for_seq_len)
from hetu_amd.models.llama import LLAMA_CONFIGS, build_llama_train_graph
from hetu_amd.nn.parallel import ParallelSpec  # noqa: E402
from hetu_amd.parallel.comm import comm_backend  # noqa: E402

BUCKETS = {"short": 512, "long": 2048}           # seq-len ceiling per bucket
STRATEGIES = {"short": dict(dp=2, tp=1),         # short seqs: data parallel
              "long": dict(dp=1, tp=2)}          # long seqs: tensor parallel


def main():
    comm = comm_backend()
    device = comm.device
    mcfg = LLAMA_CONFIGS["llama-tiny"]

    shapes = {}

    def build(key):
        spec = ParallelSpec(**STRATEGIES[key])
        S = BUCKETS[key]
        B = 8 // spec.dp
        shapes[key] = (B, S)
        g, h = build_llama_train_graph(mcfg, B, S, dtype=torch.float32,
                                       lr=1e-4, spec=spec)
        return g, h

    trainer = HotSwitchTrainer(build, device)
    torch.manual_seed(7 + comm.rank)
    seq_lens = [400, 380, 1800, 2000, 300, 1900]   # mixed workload
    for sl in seq_lens:
        key = bucket_for_seq_len(sl, BUCKETS)
        g, h, _ = trainer.switch_to(key)
        B, S = shapes[key]
        ids = torch.randint(0, mcfg.vocab, (B, S), device=device)
        labels = torch.randint(0, mcfg.vocab, (B * S,), device=device)
        loss = trainer.step(key, {h["input_ids"]: ids, h["labels"]: labels})
        if comm.rank == 0:
            print(f"seq_len {sl} -> bucket {key}: loss {float(loss):.4f}")


if __name__ == "__main__":
    main()
