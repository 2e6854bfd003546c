#!/usr/bin/env python3
"""Elastic training demo (reference examples/ampelos +
heturpc_elastic_server): run N workers; kill one mid-run and the rest
vote a restore step, re-rendezvous and continue as N-1
(hetu_amd/engine/elastic_loop.py).

Run 4 workers (the launcher hosts the KV store):
  python examples/elastic/elastic_train.py --world 4 --die-rank 3 \
      --die-at 3
"""
import argparse
import os
import subprocess
import sys
from datetime import timedelta

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

WORKER = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "_worker.py")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--world", type=int, default=4)
    ap.add_argument("--die-rank", type=int, default=-1)
    ap.add_argument("--die-at", type=int, default=-1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--kv-port", type=int, default=29770)
    args = ap.parse_args()
    import torch.distributed as dist
    store = dist.TCPStore("127.0.0.1", args.kv_port, 1, True,
                          timeout=timedelta(seconds=300))
    procs = []
    for r in range(args.world):
        env = dict(os.environ, ELASTIC_RANK=str(r),
                   ELASTIC_WORLD=str(args.world),
                   KV_PORT=str(args.kv_port), STEPS=str(args.steps),
                   CKPT_DIR=os.environ.get("CKPT_DIR", "/tmp/elastic_ckpt"))
        if r == args.die_rank:
            env["DIE_AT"] = str(args.die_at)
        procs.append(subprocess.Popen([sys.executable, WORKER], env=env))
    for r, p in enumerate(procs):
        rc = p.wait()
        print(f"rank {r} exited rc={rc}")
    del store


if __name__ == "__main__":
    main()
