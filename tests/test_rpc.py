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


def test_elastic_controller_reshape():
    """Failure -> vote -> consistent restore step -> published plan
    (reference heturpc_elastic_server state machine)."""
    import threading
    import time
    from hetu_amd.rpc.elastic import ElasticController, ElasticWorker
    from hetu_amd.rpc.kv_store import HeartbeatClient, KVStore

    port = 29721
    kv0 = KVStore("127.0.0.1", port, is_server=True, world_size=1)
    plans = []

    def replan(alive):
        return {"strategy": f"dp{len(alive)}", "alive_n": len(alive)}

    ctl = ElasticController(kv0, 3, replan, heartbeat_timeout_s=1.0)
    workers = [KVStore("127.0.0.1", port, world_size=1) for _ in range(3)]
    hbs = [HeartbeatClient(workers[r], r, interval_s=0.2) for r in range(3)]
    for hb in hbs:
        hb.start()
    time.sleep(0.6)
    ctl.start(interval_s=0.3)

    # rank 2 dies; ranks 0/1 keep polling at steps 7 and 9
    hbs[2].stop()
    results = {}

    def run_worker(r, step):
        w = ElasticWorker(workers[r], r)
        deadline = time.time() + 20
        while time.time() < deadline:
            plan = w.poll(step)
            if plan is not None:
                results[r] = plan
                return
            time.sleep(0.1)

    t0 = threading.Thread(target=run_worker, args=(0, 7))
    t1 = threading.Thread(target=run_worker, args=(1, 9))
    t0.start(); t1.start()
    t0.join(25); t1.join(25)
    ctl.stop()
    for hb in hbs[:2]:
        hb.stop()
    assert 0 in results and 1 in results, results
    for r in (0, 1):
        assert results[r]["restore_step"] == 7        # min common step
        assert results[r]["alive"] == [0, 1]
        assert results[r]["strategy"] == "dp2"


def test_pssh_launcher_local_transport(tmp_path):
    """Multi-'node' launch over an injected local transport (the ssh van
    of pssh_start.py): 2 hosts x 1 proc rendezvous over torchrun and
    train; a killed host respawns (pssh_start_elastic pool)."""
    import sys
    from hetu_amd.rpc.pssh import Host, PsshLauncher
    script = tmp_path / "w.py"
    script.write_text(
        "import os, torch, torch.distributed as dist\n"
        "from datetime import timedelta\n"
        "os.environ.setdefault('GLOO_SOCKET_IFNAME', 'lo')\n"
        "dist.init_process_group('gloo',"
        " timeout=timedelta(seconds=60))\n"
        "t = torch.ones(4) * (dist.get_rank() + 1)\n"
        "dist.all_reduce(t)\n"
        "print('PSSH_SUM', t[0].item(), dist.get_world_size())\n"
        "dist.destroy_process_group()\n")
    hosts = [Host("127.0.0.1", 1), Host("127.0.0.1", 1)]
    la = PsshLauncher(hosts, str(script), master_port=29793,
                      ssh_cmd=["bash", "-c"],
                      env_extra={"GLOO_SOCKET_IFNAME": "lo"})
    la.start()
    codes = la.wait(timeout_s=180)
    outs = [la.output(i) for i in range(2)]
    assert all(c == 0 for c in codes), (codes, outs)
    assert any("PSSH_SUM 3.0 2" in o for o in outs), outs
    # respawn path: relaunch host 1 (full group restart for the test)
    la2 = PsshLauncher(hosts, str(script), master_port=29795,
                       ssh_cmd=["bash", "-c"],
                       env_extra={"GLOO_SOCKET_IFNAME": "lo"})
    la2.start()
    la2.respawn(1)
    codes = la2.wait(timeout_s=180)
    assert codes[1] == 0, la2.output(1)
