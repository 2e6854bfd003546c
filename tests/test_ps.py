"""Parameter server + embedding cache tests (reference parity:
hetu/v1/ps-lite pull/push + hetu_cache LRU/LFU/LFUOpt + preduce).

CPU: single-process exact-parity of CachedEmbedding vs plain SGD on the
table; sharded pull/push and PartialReduce on 2 gloo ranks.
"""
import json
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _local_backend():
    import hetu_amd.parallel.comm as C
    if C._BACKEND is None or C._BACKEND.world_size != 1:
        C._BACKEND = C.CommBackend()
    return C._BACKEND


@pytest.mark.parametrize("policy", ["lru", "lfu", "lfuopt"])
def test_cached_embedding_matches_sgd(policy):
    """staleness=0 write-through: cached training == uncached SGD even when
    the working set exceeds cache capacity (eviction path exercised)."""
    _local_backend()
    from hetu_amd.ps import CachedEmbedding, ShardedEmbeddingTable
    torch.manual_seed(0)
    t = ShardedEmbeddingTable(100, 8, lr=0.1, seed=3)
    ref = t.gather_full().clone()
    ce = CachedEmbedding(t, capacity=16, policy=policy, staleness=0)
    for _ in range(8):
        ids = torch.randint(0, 100, (4, 6))
        out = ce(ids)
        assert (out.detach() - ref[ids]).abs().max() < 1e-5
        (out ** 2).sum().backward()
        g = 2 * ref[ids]
        uniq, inv = torch.unique(ids.reshape(-1), return_inverse=True)
        acc = torch.zeros(uniq.numel(), 8)
        acc.index_add_(0, inv, g.reshape(-1, 8))
        ref[uniq] -= 0.1 * acc
        assert (t.gather_full() - ref).abs().max() < 1e-5
    assert 0 < ce.hit_rate < 1


def test_cached_embedding_stale_flush():
    """staleness>0: pushes are deferred but nothing is lost — after the
    final flush the table equals init - lr * (sum of all grads)."""
    _local_backend()
    from hetu_amd.ps import CachedEmbedding, ShardedEmbeddingTable
    torch.manual_seed(1)
    t = ShardedEmbeddingTable(60, 4, lr=0.05, seed=9)
    init = t.gather_full().clone()
    ce = CachedEmbedding(t, capacity=8, policy="lru", staleness=3)
    total = torch.zeros(60, 4)
    for _ in range(7):
        ids = torch.randint(0, 60, (10,))
        out = ce(ids)
        (out.sum()).backward()       # grad = 1 per element
        for i in ids.tolist():
            total[i] += 1.0
    ce.flush()
    expect = init - 0.05 * total
    assert (t.gather_full() - expect).abs().max() < 1e-5


def test_adagrad_push():
    _local_backend()
    from hetu_amd.ps import ShardedEmbeddingTable
    t = ShardedEmbeddingTable(50, 4, lr=0.1, optimizer="adagrad", seed=1)
    r0 = t.gather_full().clone()
    t.push(torch.tensor([1, 1, 3]), torch.ones(3, 4))
    r = t.gather_full()
    # id 1 accumulates grad 2 -> step lr*2/sqrt(4); id 3 grad 1 -> lr*1/1
    assert abs((r0[1] - r[1]).max().item() - 0.2 / 2.0) < 1e-6
    assert abs((r0[3] - r[3]).max().item() - 0.1 / 1.0) < 1e-6
    assert (r0[2] - r[2]).abs().max() == 0


def test_lfuopt_admission_refusal():
    from hetu_amd.ops.functional import ext
    c = ext().EmbedCache(2, "lfuopt")
    c.query(torch.tensor([1, 2]))
    c.admit(torch.tensor([1, 2]))
    for _ in range(3):
        c.query(torch.tensor([1, 2]))          # freq up
    c.query(torch.tensor([9]))                 # ghost freq 1
    slots, _, _ = c.admit(torch.tensor([9]))
    assert slots.tolist() == [-1]              # refused: 1 <= min freq
    for _ in range(5):
        c.query(torch.tensor([9]))
    slots, ev, _ = c.admit(torch.tensor([9]))
    assert slots.tolist() != [-1] and ev.numel() == 1


WORKER = r"""
import json, os, sys, torch
sys.path.insert(0, os.environ["HETU_REPO"])
from hetu_amd.parallel.comm import comm_backend
from hetu_amd.ps import ShardedEmbeddingTable, CachedEmbedding
comm = comm_backend()
rank, ws = comm.rank, comm.world_size
t = ShardedEmbeddingTable(40, 4, lr=0.1, seed=5)
full0 = t.gather_full().clone()
# pull parity: every rank pulls the same ids and must see the same rows
ids = torch.tensor([0, 1, 7, 38, 7])
rows = t.pull(ids)
assert (rows - full0[ids]).abs().max() < 1e-6, "pull mismatch"
# push: disjoint ids per rank, grads=1; afterwards both updated
my = torch.tensor([rank, 10 + rank])
t.push(my, torch.ones(2, 4))
comm.barrier()
full = t.gather_full()
exp = full0.clone()
for r in range(ws):
    exp[[r, 10 + r]] -= 0.1
assert (full - exp).abs().max() < 1e-6, "push mismatch"
# cached embedding across workers: each trains its own ids, flush syncs
ce = CachedEmbedding(t, capacity=8, staleness=0)
out = ce(torch.tensor([20 + rank]))
out.sum().backward()
comm.barrier()
got = t.gather_full()
assert abs((exp[20 + rank] - got[20 + rank]).max().item() - 0.1) < 1e-6
# partial reduce: both ranks arrive -> one group of 2, mean
from hetu_amd.rpc.kv_store import KVStore
from hetu_amd.parallel.preduce import PartialReduce
kv = KVStore("127.0.0.1", int(os.environ["PR_PORT"]),
             is_server=(rank == 0), world_size=ws)
pr = PartialReduce(kv, comm, min_size=2)
x = torch.full((4,), float(rank + 1))
y, members = pr.preduce(x)
assert members == list(range(ws)), members
assert (y - 1.5).abs().max() < 1e-6, y
print("PSOK:" + json.dumps(rank))
"""


def test_ps_two_ranks():
    env0 = {**os.environ, "HETU_REPO": REPO, "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29612", "PR_PORT": "29613",
            "GLOO_SOCKET_IFNAME": "lo"}
    procs = []
    for r in range(2):
        env = dict(env0, RANK=str(r), WORLD_SIZE="2", LOCAL_RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, "-c", WORKER],
                                      env=env, stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=300)
        ok = (p.returncode == 0 or p.returncode == -6) and "PSOK:" in out
        assert ok, f"rank {r} failed rc={p.returncode}:\n{out}\n{err}"


def test_ps_server_transport_sparse_training():
    """ps-lite-style server transport: 2 server shards over TCP, a worker
    pushes sparse grads and the server-side optimizer converges an
    embedding toward targets (reference kv_app.h Push/Pull +
    PSFhandle_embedding server-side update)."""
    import torch
    from hetu_amd.ps.server import PSClient, PSServer
    s1 = PSServer().start()
    s2 = PSServer().start()
    cli = PSClient([("127.0.0.1", s1.port), ("127.0.0.1", s2.port)])
    try:
        N, D = 64, 8
        cli.register("emb", N, D, optimizer="sgd", lr=0.5, init_std=0.01)
        torch.manual_seed(0)
        target = torch.randn(N, D)
        for step in range(200):
            ids = torch.randint(0, N, (32,))
            rows = cli.pull("emb", ids)
            grads = rows - target[ids]          # d/drow of 0.5||row-t||^2
            cli.push("emb", ids, grads)
        rows = cli.pull("emb", torch.arange(N))
        err = (rows - target).norm() / target.norm()
        assert err < 0.05, float(err)
        # adagrad path
        cli.register("emb2", N, D, optimizer="adagrad", lr=1.0)
        ids = torch.arange(N)
        for step in range(100):
            rows = cli.pull("emb2", ids)
            cli.push("emb2", ids, rows - target)
        err2 = (cli.pull("emb2", ids) - target).norm() / target.norm()
        assert err2 < 0.2, float(err2)
    finally:
        cli.stop_servers()
        cli.close()
