"""KV store, heartbeat/failure detection, elastic re-planning
(reference heturpc servers + Ampelos)."""
import threading
import time

import pytest

from hetu_amd.rpc.kv_store import (FailureDetector, HeartbeatClient, KVStore,
                                   replan_after_failure)


def test_kv_put_get_barrier():
    kv = KVStore(port=29711, is_server=True, world_size=1)
    kv.put("x", {"a": 1})
    assert kv.get("x") == {"a": 1}
    kv.put("s", "plain")
    assert kv.get("s") == "plain"
    assert kv.add("ctr", 3) == 3
    # barrier with n=2 from two threads
    done = []

    def member():
        # own client connection: a TCPStore client socket is not
        # thread-safe to share across concurrently-blocking calls
        kv2 = KVStore(port=29711, is_server=False, world_size=1)
        kv2.barrier("b1", 2)
        done.append(1)
    t = threading.Thread(target=member)
    t.start()
    kv.barrier("b1", 2)
    t.join(timeout=10)
    assert len(done) == 1


def test_failure_detection_and_replan():
    kv = KVStore(port=29712, is_server=True, world_size=1)
    hb = [HeartbeatClient(kv, r, interval_s=0.1) for r in range(3)]
    for h in hb:
        h.start()
    time.sleep(0.4)
    failed = []
    det = FailureDetector(kv, 3, timeout_s=0.5,
                          on_failure=lambda d: failed.append(list(d)))
    assert det.scan_once() == []
    hb[1].stop()          # rank 1 dies
    time.sleep(1.0)
    newly = det.scan_once()
    assert newly == [1]
    assert kv.get("dead_ranks") == [1]
    for h in hb:
        h.stop()
    # elastic re-plan for survivors
    from hetu_amd.galvatron.cost_model import ModelShape
    shape = ModelShape(n_layer=8, hidden=512, ffn_hidden=2048, vocab=1000,
                       n_head=8)
    st, gb, alive = replan_after_failure(shape, 128, 8, [0, 2])
    assert st.world == 2 and gb % 2 == 0
