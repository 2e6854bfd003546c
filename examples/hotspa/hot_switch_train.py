#!/usr/bin/env python3
"""HotSPa-style multi-strategy training (reference examples/hotspa).

Sequence lengths are bucketed; each bucket trains under its best parallel
strategy and the trainer hot-switches parameters (+ Adam state) between
strategies with one batched p2p when the bucket changes
(hetu_amd/parallel/switch.py).

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
       --master-addr 127.0.0.1 examples/hotspa/hot_switch_train.py
"""
import argparse
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
    ap = argparse.ArgumentParser()
    # BASELINE config 5 is llama-13b on 8 GPUs; llama-tiny is the CPU/CI
    # default so the same driver runs everywhere
    ap.add_argument("--model", default="llama-tiny",
                    choices=list(LLAMA_CONFIGS))
    ap.add_argument("--dtype", default=None,
                    help="default: bf16 on GPU, fp32 on CPU")
    args = ap.parse_args()
    comm = comm_backend()
    device = comm.device
    mcfg = LLAMA_CONFIGS[args.model]
    dt = {"bf16": torch.bfloat16, "fp32": torch.float32,
          None: torch.bfloat16 if device.type == "cuda"
          else torch.float32}[args.dtype]

    shapes = {}

    def build(key):
        spec = ParallelSpec(**STRATEGIES[key])
        S = BUCKETS[key]
        B = 8 // spec.dp
        shapes[key] = (B, S)
        g, h = build_llama_train_graph(mcfg, B, S, dtype=dt,
                                       lr=1e-4, spec=spec)
        return g, h

    trainer = HotSwitchTrainer(build, device)
    ws = comm.world_size
    for k in STRATEGIES:                      # scale strategies to world
        STRATEGIES[k] = {kk: (ws if v == 2 and ws > 2 else v)
                         for kk, v in STRATEGIES[k].items()}
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
