"""LobRA-style multi-tenant LoRA fine-tuning demo.

Reference parity: examples/lobra (multi-task LoRA over one frozen base,
static rank planner, proportional batch scheduler).  Three "tenants" with
different data sizes fine-tune their own adapters over a shared frozen
two-layer parallel-linear base: the balance planner assigns ranks from the
tenants' demand weights, the stride scheduler interleaves their
micro-batches, and each step trains exactly one tenant's adapters.

Run:  python examples/lobra/train_multi_lora.py [--steps 60]
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

from hetu_amd.engine.runner import prepare_run_context
from hetu_amd.graph.graph import DefineAndRunGraph, pop_graph, push_graph
from hetu_amd.graph.ops import api as ht
from hetu_amd.graph.ops.optim import Adam
from hetu_amd.nn.parallel import ColumnParallelLinear, ParallelSpec, \
    RowParallelLinear
from hetu_amd.peft.multi_task import (MultiLoRALinear, TaskBatchScheduler,
                                      balance_plan)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=60)
    ap.add_argument("--rank-budget", type=int, default=24)
    args = ap.parse_args()
    torch.manual_seed(0)

    # tenants: (data size, demand weight for the planner)
    tenants = {"news": 600, "code": 300, "chat": 100}
    ranks = balance_plan({t: float(n) for t, n in tenants.items()},
                         rank_budget=args.rank_budget, r_min=2, r_max=16)
    print(f"planner ranks: {ranks}")

    d_in, d_h, d_out = 32, 64, 16
    g = DefineAndRunGraph("lobra")
    push_graph(g)
    try:
        spec = ParallelSpec()
        x = ht.placeholder((8, d_in), name="x")
        tgt = ht.placeholder((8, d_out), name="tgt")
        fc1 = ColumnParallelLinear(d_in, d_h, spec, bias=False,
                                   dtype=torch.float32, name="fc1")
        fc2 = RowParallelLinear(d_h, d_out, spec, bias=False,
                                dtype=torch.float32, name="fc2")
        l1 = MultiLoRALinear(fc1, ranks, name="l1")
        l2 = MultiLoRALinear(fc2, ranks, name="l2")
        losses, train_ops = {}, {}
        for t in tenants:
            h = ht.relu(l1(x, t))
            y = l2(h, t)
            losses[t] = ht.mse_loss(y, tgt)
            train_ops[t] = Adam(lr=5e-3).minimize(
                losses[t],
                params=l1.task_parameters(t) + l2.task_parameters(t))
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)

    # fixed per-tenant synthetic tasks (distinct targets)
    data = {t: (torch.randn(8, d_in),
                torch.randn(8, d_out) * (i + 1))
            for i, t in enumerate(tenants)}
    sched = TaskBatchScheduler(tenants)
    first, last = {}, {}
    for step in range(args.steps):
        t = sched.next()
        xd, td = data[t]
        lv, _ = g.run([losses[t], train_ops[t]], {x: xd, tgt: td}, ctx=ctx)
        first.setdefault(t, float(lv))
        last[t] = float(lv)
        if step % 10 == 0:
            print(f"step {step:3d} tenant={t:5s} loss={float(lv):.4f}")
    for t in tenants:
        print(f"tenant {t}: first={first[t]:.4f} last={last[t]:.4f}")
        assert last[t] < first[t], f"tenant {t} did not improve"
    print("LOBRA_OK")


if __name__ == "__main__":
    main()
