"""PartialReduce: dynamic partner grouping + partial all-reduce.

Reference parity: /root/reference/hetu/v1/python/hetu/preduce.py:8-41
(`PartialReduce.get_partner` → ps-lite `preduce_handler.cc` forms a group
from whichever workers arrive within a time window, then the group runs an
all-reduce) — the straggler-tolerant gradient sync used by Malleus-style
training.

MI355X-native: coordination goes through the TCPStore-backed KVStore (the
same store that bootstraps RCCL) instead of a zmq PS; the reduce itself is
an RCCL all-reduce over a cached ProcessGroup of exactly the partner set,
so fast workers are never blocked on the slowest rank.

Protocol per (token, round, generation):
  1. every arriving rank atomically increments the arrival counter and
     records its rank;
  2. the FIRST arrival coordinates: it waits until everyone is in, or the
     window elapses with at least `min_size` arrivals, then publishes the
     member list;
  3. ranks in the published list all-reduce together; later arrivals retry
     in generation g+1 and form their own group.
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional, Tuple

import torch

from ..rpc.kv_store import KVStore
from .comm import CommBackend, comm_backend


class PartialReduce:
    def __init__(self, kv: KVStore, comm: Optional[CommBackend] = None,
                 min_size: int = 2, window_s: float = 0.5,
                 poll_s: float = 0.005):
        self.kv = kv
        self.comm = comm or comm_backend()
        self.min_size = min(min_size, self.comm.world_size)
        self.window = window_s
        self.poll = poll_s
        self._round: Dict[str, int] = {}

    def _form_group(self, base: str) -> List[int]:
        """Register under `base`; return this generation's member ranks
        (may not include self — then the caller retries next generation)."""
        idx = self.kv.add(f"{base}/n", 1)
        self.kv.put(f"{base}/member/{idx}", self.comm.rank)
        if idx == 1:  # coordinator
            t0 = time.time()
            while True:
                n = self.kv.add(f"{base}/n", 0)
                if n >= self.comm.world_size:
                    break
                if n >= self.min_size and time.time() - t0 > self.window:
                    break
                time.sleep(self.poll)
            n = self.kv.add(f"{base}/n", 0)
            members = sorted(
                self.kv.get(f"{base}/member/{i}")
                for i in range(1, n + 1))
            self.kv.put(f"{base}/group", members)
            return members
        self.kv.wait([f"{base}/group"])
        return self.kv.get(f"{base}/group")

    def get_partner(self, token: str = "grad") -> List[int]:
        """Block until this rank is part of a formed group; returns the
        sorted partner ranks (reference preduce.py:get_partner)."""
        rnd = self._round.get(token, 0)
        self._round[token] = rnd + 1
        gen = 0
        while True:
            base = f"preduce/{token}/{rnd}/{gen}"
            members = self._form_group(base)
            if self.comm.rank in members:
                return members
            gen += 1

    def preduce(self, tensor: torch.Tensor, token: str = "grad"
                ) -> Tuple[torch.Tensor, List[int]]:
        """Partial all-reduce (mean) over the dynamically formed group."""
        members = self.get_partner(token)
        if len(members) > 1:
            tensor = self.comm.allreduce(tensor, members, op="avg")
        return tensor, members
