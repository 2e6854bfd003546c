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
    every switch."""

    def __init__(self, build_fn: Callable[[str], Tuple], device,
                 comm=None):
        self.build_fn = build_fn
        self.device = device
        self.comm = comm or comm_backend(device)
        self.pool: Dict[str, Tuple] = {}       # key -> (graph, handles, ctx)
        self.active: Optional[str] = None
        self.switches = 0

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
            switch_graph_params(ga, g, self.comm)
            self.switches += 1
        self.active = key
        return g, h, ctx

    def step(self, key: str, feed: Dict):
        g, h, ctx = self.switch_to(key)
        loss, _ = g.run([h["loss"], h["train_op"]], feed, ctx=ctx)
        return loss


def bucket_for_seq_len(seq_len: int, buckets) -> str:
    """Smallest bucket holding seq_len (hotspa seq-len bucket dispatch)."""
    for b in sorted(buckets):
        if seq_len <= b:
            return str(b)
    return str(max(buckets))
