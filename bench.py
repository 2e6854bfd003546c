#!/usr/bin/env python3
"""Driver benchmark contract: flagship training step, tokens/sec whole-job.

Metric (BASELINE.json): tokens/sec (whole node), GPT-3 7B, at 1/2/4/8
MI355X; synthetic data, random-init weights, bf16 compute. For N>1 the
driver launches this under torch.distributed.run with one rank per GPU
(RCCL); per-GPU work is fixed (weak scaling).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default="gpt3-7b")
    ap.add_argument("--micro-batch", type=int, default=4)
    ap.add_argument("--seq-len", type=int, default=2048)
    ap.add_argument("--capture", default="auto")
    args = ap.parse_args()

    if args.capture != "auto":
        os.environ["HETU_AMD_CAPTURE"] = args.capture

    import hetu_amd  # noqa: F401
    from hetu_amd.engine.trainer import Trainer
    from hetu_amd.models.gpt import GPT_CONFIGS, build_gpt_train_graph
    from hetu_amd.parallel.comm import comm_backend
    from hetu_amd.parallel.dstates import DistributedStates

    assert torch.cuda.is_available(), "bench requires a GPU"
    comm = comm_backend()
    rank, ws = comm.rank, comm.world_size
    device = comm.device
    torch.cuda.set_device(device)

    cfg = GPT_CONFIGS[args.model]
    B, S = args.micro_batch, args.seq_len

    # Data-parallel SPMD annotation at build time: inputs split on dim 0,
    # parameters duplicated; minimize() inserts the grad allreduce comm ops.
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=torch.bfloat16, lr=1e-4, dp=ws)

    trainer = Trainer(g, h, device)

    torch.manual_seed(1234 + rank)
    pool = [torch.randint(0, cfg.vocab, (B, S), device=device)
            for _ in range(4)]
    lpool = [torch.randint(0, cfg.vocab, (B * S,), device=device)
             for _ in range(4)]

    def feed(i):
        return {h["input_ids"]: pool[i % 4], h["labels"]: lpool[i % 4]}

    for i in range(args.warmup):
        trainer.step(feed(i))
    loss = None

    comm.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        loss = trainer.step(feed(i))
    comm.barrier()
    torch.cuda.synchronize()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    # max over ranks
    if ws > 1:
        et = torch.tensor([elapsed], device=device)
        import torch.distributed as dist
        dist.all_reduce(et, op=dist.ReduceOp.MAX)
        elapsed = et.item()

    ms_per_step = elapsed / args.steps * 1000
    tokens_per_step = B * S * ws
    tok_s = tokens_per_step * args.steps / elapsed

    if rank == 0:
        out = {
            "metric": "tokens/sec (whole node) GPT-3 7B",
            "value": tok_s,
            "unit": "tokens/s",
            "n_gpus": ws,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "loss": float(loss.float().item()) if loss is not None else None,
            "config": {"model": args.model, "global_batch": B * ws,
                       "seq_len": S, "parallelism": f"dp{ws}",
                       "capture": trainer._cuda_graph is not None},
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
