"""Elastic recovery end-to-end (reference heturpc_elastic_server.py:463-560
semantics): kill one of 4 gloo ranks mid-run; the survivors vote the
restore step, re-rendezvous as a 3-rank world, reload the checkpoint and
finish training."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

ELASTIC_WORKER = r"""
import os, sys, json, torch
sys.path.insert(0, os.environ["HETU_REPO"])
from hetu_amd.engine.elastic_loop import run_elastic_training
from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph
from hetu_amd.nn.parallel import ParallelSpec

rank = int(os.environ["ELASTIC_RANK"])
world = int(os.environ["ELASTIC_WORLD"])
cfg = GPTConfig(n_layer=1, n_head=2, n_kv_head=2, hidden=32,
                ffn_hidden=64, vocab=64, max_seq=8)
S = 8


def build_fn(ws, comm):
    spec = ParallelSpec(dp=ws, device_group=list(range(ws)))
    return build_gpt_train_graph(cfg, micro_batch=2, seq_len=S,
                                 dtype=torch.float32, lr=1e-2, spec=spec)


def feed_fn(step):
    gen = torch.Generator().manual_seed(1000 + 31 * step + rank)
    ids = torch.randint(0, 64, (2, S), generator=gen)
    labels = torch.randint(0, 64, (2 * S,), generator=gen)
    return ids, labels.reshape(-1)


die_at = int(os.environ.get("DIE_AT", "-1"))
res = run_elastic_training(
    build_fn, feed_fn, total_steps=7,
    ckpt_dir=os.environ["CKPT_DIR"],
    kv_host="127.0.0.1", kv_port=int(os.environ["KV_PORT"]),
    rank=rank, world=world,
    rendezvous_port=int(os.environ["RDV_PORT"]),
    heartbeat_timeout_s=2.0,
    die_at=die_at if die_at >= 0 else None)
print("ELASTIC:" + json.dumps({"rank": rank, **res}))
"""


def test_kill_one_of_four_resumes_with_three(tmp_path):
    import torch.distributed as dist
    from datetime import timedelta
    kv_port, rdv_port = 29751, 29860
    # host the KV store in the test process (the reference's external
    # DeviceController service)
    server_store = dist.TCPStore("127.0.0.1", kv_port, 1, True,
                                 timeout=timedelta(seconds=180))
    env0 = {**os.environ, "HETU_REPO": REPO, "CKPT_DIR": str(tmp_path),
            "KV_PORT": str(kv_port), "RDV_PORT": str(rdv_port),
            "GLOO_SOCKET_IFNAME": "lo"}
    procs = []
    for r in range(4):
        env = dict(env0, ELASTIC_RANK=str(r), ELASTIC_WORLD="4")
        if r == 3:
            env["DIE_AT"] = "3"          # rank 3 dies before step 3
        procs.append(subprocess.Popen([sys.executable, "-c",
                                       ELASTIC_WORKER], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    results = {}
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=240)
        if r == 3:
            assert p.returncode == 17, f"rank3 should die: {out}\n{err}"
            continue
        assert p.returncode in (0, -6), f"rank {r}: rc={p.returncode}\n" \
                                        f"{out[-2000:]}\n{err[-3000:]}"
        for ln in out.splitlines():
            if ln.startswith("ELASTIC:"):
                results[r] = json.loads(ln[len("ELASTIC:"):])
    del server_store
    assert set(results) == {0, 1, 2}, results
    for r, res in results.items():
        assert res["final_world"] == 3, res
        assert res["epoch"] == 1, res
        # trained through all 7 steps (0..6) despite the failure
        losses = {int(k): v for k, v in res["losses"].items()}
        assert max(losses) == 6, sorted(losses)
        assert min(losses) == 0
    # survivors agree on the loss trajectory after the reshape
    l0 = {int(k): v for k, v in results[0]["losses"].items()}
    l1 = {int(k): v for k, v in results[1]["losses"].items()}
    for s in range(3, 7):
        assert abs(l0[s] - l1[s]) < 1e-6
