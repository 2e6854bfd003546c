"""HotSPa-style multi-strategy trainer: a plan pool of executable graphs,
one per parallel strategy, with live param/optimizer migration on switch.

Reference parity: DefineAndRunGraph's exec-graph plan pool + hot switch
(define_and_run_graph.cc:1174,1380-1460) and the seq-len-bucket switch
driver (examples/hotspa/llama_hot_switch_trainer.py:58-75): batches are
bucketed by max sequence length and each bucket trains under the strategy
that suits it (e.g. long-seq buckets under tp/cp-heavy layouts, short ones
under dp-heavy).
"""
from __future__ import annotations

from typing import Callable, Dict, Optional, Tuple

import torch

from ..engine.runner import prepare_run_context
from ..parallel.comm import comm_backend
from ..parallel.switch import switch_graph_params


class HotSwitchTrainer:
    """build_fn(strategy_key) -> (graph, handles); strategies are built
    lazily, parameters migrate from the active graph on first use and on
    every switch.

    Run levels (reference graph.h:33-39 RunLevel + switch_exec_graph.h:42-54
    SWITCH_MODE/LEVEL):
      * step(key, feed)                — full COMPUTE+GRAD+UPDATE step
      * step(key, feed, level="grad")  — accumulate grads only (no update)
      * step(key, feed, level="update")— accumulate + apply ALL pending
    A switch with pending accumulated grads migrates them to the new
    layout too (SWITCH_ACCUMULATE_GRAD): partial-over-dp grads are reduced
    in the OLD layout first, then resharded like parameters."""

    def __init__(self, build_fn: Callable[[str], Tuple], device,
                 comm=None):
        self.build_fn = build_fn
        self.device = device
        self.comm = comm or comm_backend(device)
        self.pool: Dict[str, Tuple] = {}       # key -> (graph, handles, ctx)
        self.active: Optional[str] = None
        self.switches = 0
        self._accum: Dict[str, torch.Tensor] = {}   # param name -> grad

    # ---- grad plumbing ---------------------------------------------------
    def _grad_tensors(self, g):
        """(param, pre-update grad tensor) pairs from the update ops."""
        out = []
        for op in g.ops:
            if op.type in ("AdamStep", "ZeroAdamStep", "SGDStep"):
                out.append((op.inputs[0], op.inputs[1]))
        return out

    def _reduce_partial(self, g, grads_by_name):
        """Reduce partial-over-dp grads in the CURRENT layout so they can
        reshard like parameters."""
        from ..graph.ops.comm import _ranks
        for p, gt in self._grad_tensors(g):
            name = p.name.split(":")[0]
            if name not in grads_by_name or gt.ds is None \
                    or gt.ds.partial <= 1:
                continue
            dg = p.device_group
            my = dg.index(self.comm.rank) if dg else 0
            ranks = _ranks(dg, gt.ds.group_devices_along(-2), my)
            grads_by_name[name] = self.comm.allreduce(
                grads_by_name[name], ranks)

    def _get(self, key: str):
        if key not in self.pool:
            g, h = self.build_fn(key)
            ctx = prepare_run_context(g, self.device)
            self.pool[key] = (g, h, ctx)
        return self.pool[key]

    def switch_to(self, key: str):
        if key == self.active:
            return self.pool[key]
        g, h, ctx = self._get(key)
        if self.active is not None:
            ga = self.pool[self.active][0]
            if self._accum:
                # SWITCH_ACCUMULATE_GRAD: reduce partials, reshard the
                # accumulated grads alongside the params
                self._reduce_partial(ga, self._accum)
                by_name_a = {p.name.split(":")[0]: p
                             for p in ga.parameters}
                new_accum = {}
                from ..parallel.switch import plan_entries, switch_params
                plan = []
                for p in g.parameters:
                    name = p.name.split(":")[0]
                    if name not in self._accum:
                        continue
                    pa = by_name_a.get(name)
                    dst = torch.zeros(tuple(p.shape), dtype=torch.float32,
                                      device=self.device)
                    # section-aware (fused qkv interleave) like params
                    plan.extend(plan_entries(pa, p, self._accum[name],
                                             dst))
                    new_accum[name] = dst
                switch_params(plan, self.comm)
                self._accum = new_accum
            switch_graph_params(ga, g, self.comm)
            self.switches += 1
        self.active = key
        return g, h, ctx

    def step(self, key: str, feed: Dict, level: str = "full"):
        g, h, ctx = self.switch_to(key)
        if level == "full" and not self._accum:
            loss, _ = g.run([h["loss"], h["train_op"]], feed, ctx=ctx)
            return loss
        pairs = self._grad_tensors(g)
        fetches = [h["loss"]] + [gt for _, gt in pairs]
        vals = g.run(fetches, feed, ctx=ctx)
        loss, gvals = vals[0], vals[1:]
        for (p, _), gv in zip(pairs, gvals):
            name = p.name.split(":")[0]
            if name in self._accum:
                self._accum[name] = self._accum[name] + gv.float()
            else:
                self._accum[name] = gv.float().clone()
        if level == "grad":
            return loss
        # UPDATE: seed the grad tensors with the accumulated sums, run the
        # optimizer subgraph only (plan cut at the seeds)
        seeds = {}
        for p, gt in pairs:
            name = p.name.split(":")[0]
            seeds[gt.id] = self._accum[name].to(gt.dtype
                                                if gt.dtype is not None
                                                else torch.float32)
        g.run([h["train_op"]], {}, ctx=ctx, seed_values=seeds)
        self._accum = {}
        return loss


def bucket_for_seq_len(seq_len: int, buckets) -> str:
    """Smallest bucket holding seq_len (hotspa seq-len bucket dispatch).
    `buckets` is either a list of int ceilings (key = str(ceiling)) or a
    {name: ceiling} dict (key = name)."""
    if isinstance(buckets, dict):
        for k, b in sorted(buckets.items(), key=lambda kv: kv[1]):
            if seq_len <= b:
                return k
        return max(buckets, key=buckets.get)
    for b in sorted(buckets):
        if seq_len <= b:
            return str(b)
    return str(max(buckets))
