"""Elastic training controller: status state machine + consistency vote.

Reference parity: rpc/heturpc_elastic_server.py (SERVER_STATUS state
machine :39-560, `Consistent` vote :389, WorkerStop :470-485, dead-worker
detection :463-476) and the Ampelos re-planning flow
(engine/strategy_ampelos.py:906).

MI355X-native shape: the controller is a thread next to rank 0 (no
separate gRPC service); all state rides the TCPStore KVStore.  On failure:
  RUNNING -> VOTING    (survivors report their last completed step)
  VOTING  -> RESHAPING (controller picks the min common step = the
                        consistent restore point, re-runs the Galvatron
                        search for the surviving world, publishes the plan)
  RESHAPING -> RUNNING (workers ack, reload from the restore step)
Workers poll `worker_poll(step)` once per training step.
"""
from __future__ import annotations

import json
import threading
import time
from typing import Callable, Dict, List, Optional

from .kv_store import FailureDetector, KVStore

RUNNING, VOTING, RESHAPING = "running", "voting", "reshaping"


class ElasticController:
    def __init__(self, kv: KVStore, world_size: int,
                 replan: Callable[[List[int]], Dict],
                 heartbeat_timeout_s: float = 5.0):
        """replan(alive_ranks) -> plan dict (strategy/batch/...)."""
        self.kv = kv
        self.world = world_size
        self.replan = replan
        self.epoch = 0                       # reshape generation
        self.kv.put("elastic/status", RUNNING)
        self.kv.put("elastic/epoch", 0)
        self.detector = FailureDetector(kv, world_size,
                                        timeout_s=heartbeat_timeout_s,
                                        on_failure=self._on_failure)
        self._lock = threading.Lock()

    def start(self, interval_s: float = 1.0):
        self.detector.start(interval_s)

    def stop(self):
        self.detector.stop()

    def _on_failure(self, dead: List[int]):
        with self._lock:
            alive = [r for r in range(self.world) if r not in dead]
            self.epoch += 1
            self.kv.put("elastic/alive", alive)
            self.kv.put("elastic/status", VOTING)
            self.kv.put("elastic/epoch", self.epoch)
            # collect votes (survivors report last completed step)
            votes = {}
            deadline = time.time() + 30.0
            while len(votes) < len(alive) and time.time() < deadline:
                for r in alive:
                    if r in votes:
                        continue
                    key = f"elastic/vote/{self.epoch}/{r}"
                    try:
                        if self.kv.has(key):
                            votes[r] = int(self.kv.get(key))
                    except Exception:  # noqa: BLE001
                        pass
                time.sleep(0.02)
            restore = min(votes.values()) if votes else 0
            plan = self.replan(alive)
            plan = dict(plan, restore_step=restore, alive=alive,
                        epoch=self.epoch)
            self.kv.put("elastic/status", RESHAPING)
            self.kv.put(f"elastic/plan/{self.epoch}", plan)

    def resume(self):
        """Called once workers have reconfigured: back to RUNNING."""
        self.kv.put("elastic/status", RUNNING)


class ElasticWorker:
    """Per-rank client: call poll(step) once per training step; returns a
    reshape plan when the cluster must reconfigure, else None."""

    def __init__(self, kv: KVStore, rank: int):
        self.kv = kv
        self.rank = rank
        self._voted_epoch = 0

    def poll(self, completed_step: int) -> Optional[Dict]:
        try:
            status = self.kv.get("elastic/status")
        except Exception:  # noqa: BLE001
            return None
        if status == RUNNING:
            return None
        epoch = int(self.kv.get("elastic/epoch"))
        if status == VOTING and self._voted_epoch < epoch:
            self.kv.put(f"elastic/vote/{epoch}/{self.rank}", completed_step)
            self._voted_epoch = epoch
        # wait for the published plan of this epoch (non-fatal timeout:
        # the caller polls again)
        key = f"elastic/plan/{epoch}"
        try:
            self.kv.wait([key], timeout_s=30.0)
        except Exception:  # noqa: BLE001
            return None
        plan = self.kv.get(key)
        if isinstance(plan, str):
            plan = json.loads(plan)
        if self.rank not in plan["alive"]:
            return {"stop": True, **plan}     # WorkerStop semantics
        return plan
