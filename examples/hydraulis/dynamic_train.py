#!/usr/bin/env python3
"""Hydraulis-style dynamic strategy dispatch (reference examples/hydraulis):
each incoming batch of ragged sequences is bucketed and each bucket is
dispatched to the strategy whose fitted cost curve a*s^2 + b*s + c wins at
that length (hetu_amd/engine/dynamic_planner.py).

Run: python examples/hydraulis/dynamic_train.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.engine.dynamic_planner import DynamicPlanner  # noqa: E402
from hetu_amd.galvatron.cost_model import ModelShape, Strategy  # noqa


def main():
    shape = ModelShape(n_layer=12, hidden=1024, ffn_hidden=4096,
                       vocab=50304, n_head=16, kind="gpt")
    cands = [Strategy(dp=8), Strategy(dp=4, tp=2),
             Strategy(dp=2, tp=4), Strategy(dp=1, tp=8)]
    buckets = [512, 1024, 2048, 4096]
    planner = DynamicPlanner(shape, n_gpus=8, candidates=cands,
                             buckets=buckets)
    torch.manual_seed(0)
    for step in range(5):
        seq_lens = (torch.randint(5, 13, (32,)) ** 2 * 16
                    ).clamp(max=4096).tolist()
        plan = planner.plan(seq_lens)
        print(f"batch {step}: " + "  ".join(
            f"bucket<={b}: {cands[ci].name()}"
            for b, ci in sorted(plan.items())))


if __name__ == "__main__":
    main()
