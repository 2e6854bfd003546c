"""Cluster bootstrap + KV store + heartbeat / failure detection.

Reference parity: the gRPC DeviceController stack — rank assignment and KV
(python/hetu/rpc/heturpc_polling_server.py, kv_store/client.py:101),
heartbeat tracking (`last_heartbeat`, :291) and the elastic server's dead-
worker detection (heturpc_elastic_server.py:463-476).

MI355X-native choice: torch.distributed.TCPStore already provides the
rendezvous + KV + named-barrier primitives RCCL bootstrap needs, so the
DeviceController is a thin layer over it (no extra gRPC service to deploy);
heartbeats are timestamped KV entries scanned by a monitor thread.
"""
from __future__ import annotations

import json
import threading
import time
from datetime import timedelta
from typing import Callable, Dict, List, Optional

import torch.distributed as dist


class KVStore:
    """put/get (str/json), counters, named barriers over a TCPStore."""

    def __init__(self, host: str = "127.0.0.1", port: int = 29700,
                 is_server: bool = False, world_size: int = 1,
                 timeout_s: float = 60.0):
        self.world_size = world_size
        self.store = dist.TCPStore(host, port, world_size, is_server,
                                   timeout=timedelta(seconds=timeout_s))

    def put(self, key: str, value) -> None:
        if not isinstance(value, str):
            value = json.dumps(value)
        self.store.set(key, value)

    def get(self, key: str, json_load: bool = True):
        v = self.store.get(key).decode()
        if json_load:
            try:
                return json.loads(v)
            except (json.JSONDecodeError, ValueError):
                return v
        return v

    def has(self, key: str) -> bool:
        return self.store.check([key])

    def wait(self, keys: List[str], timeout_s: Optional[float] = None):
        if timeout_s is not None:
            self.store.wait(keys, timedelta(seconds=timeout_s))
        else:
            self.store.wait(keys)

    def add(self, key: str, amount: int = 1) -> int:
        return self.store.add(key, amount)

    def barrier(self, name: str, n: int, timeout_s: float = 60.0):
        """Named barrier (reference heturpc named barriers :266)."""
        arrived = self.add(f"barrier/{name}/count", 1)
        if arrived == n:
            self.put(f"barrier/{name}/go", "1")
        self.wait([f"barrier/{name}/go"], timeout_s)


class HeartbeatClient:
    """Worker side: periodic timestamped heartbeat into the KV store."""

    def __init__(self, kv: KVStore, rank: int, interval_s: float = 1.0):
        self.kv = kv
        self.rank = rank
        self.interval = interval_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self):
        def loop():
            while not self._stop.is_set():
                self.kv.put(f"heartbeat/{self.rank}", time.time())
                self._stop.wait(self.interval)
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)


class FailureDetector:
    """Coordinator side: scans heartbeats; on timeout marks ranks dead and
    invokes the callback (Ampelos-style re-planning hook,
    engine/strategy_ampelos.py:906)."""

    def __init__(self, kv: KVStore, world_size: int,
                 timeout_s: float = 5.0,
                 on_failure: Optional[Callable[[List[int]], None]] = None):
        self.kv = kv
        self.world_size = world_size
        self.timeout_s = timeout_s
        self.on_failure = on_failure
        self.dead: List[int] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def scan_once(self) -> List[int]:
        now = time.time()
        newly = []
        for r in range(self.world_size):
            if r in self.dead:
                continue
            try:
                ts = float(self.kv.get(f"heartbeat/{r}"))
            except Exception:  # noqa: BLE001  (no heartbeat yet)
                continue
            if now - ts > self.timeout_s:
                newly.append(r)
        if newly:
            self.dead.extend(newly)
            self.kv.put("dead_ranks", self.dead)
            if self.on_failure:
                self.on_failure(list(self.dead))
        return newly

    def start(self, interval_s: float = 1.0):
        def loop():
            while not self._stop.is_set():
                self.scan_once()
                self._stop.wait(interval_s)
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)


def replan_after_failure(model_shape, seq_len: int, global_batch: int,
                         alive_ranks: List[int]):
    """Elastic recovery policy: re-run the Galvatron search for the
    surviving device count (Ampelos re-planning semantics) and return the
    new strategy + rank list."""
    from ..galvatron.search import search
    n = len(alive_ranks)
    # shrink the batch if it no longer divides
    gb = global_batch
    while gb % n != 0 and gb > 1:
        gb -= 1
    st, res = search(model_shape, seq_len, n, gb)
    return st, gb, alive_ranks
