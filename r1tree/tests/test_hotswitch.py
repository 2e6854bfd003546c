"""Parallelism hot switching (HotSPa): tp2 -> dp2 live migration of params
AND Adam state must continue the exact single-process training trajectory.
(reference switch_exec_graph.cc BufferBatchedIsendIrecvExec)"""
import json
import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json, os, sys, torch
sys.path.insert(0, os.environ["HETU_REPO"])
from hetu_amd.models.llama import LlamaConfig, build_llama_train_graph
from hetu_amd.nn.parallel import ParallelSpec
from hetu_amd.engine.runner import prepare_run_context
from hetu_amd.parallel.switch import switch_graph_params
from hetu_amd.parallel.comm import comm_backend

ws = int(os.environ.get("WORLD_SIZE", "1"))
rank = int(os.environ.get("RANK", "0"))
cfg = LlamaConfig(n_layer=2, n_head=4, n_kv_head=4, hidden=64,
                  ffn_hidden=128, vocab=312, max_seq=16)
gen = torch.Generator().manual_seed(99)
ids = torch.randint(0, cfg.vocab, (4, 16), generator=gen)
labels = torch.randint(0, cfg.vocab, (4, 16), generator=gen)
losses = []

if ws == 1:
    g, h = build_llama_train_graph(cfg, 4, 16, dtype=torch.float32, lr=1e-3)
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    for i in range(4):
        lv, _ = g.run([h["loss"], h["train_op"]],
                      {h["input_ids"]: ids,
                       h["labels"]: labels.reshape(-1)}, ctx=ctx)
        losses.append(float(lv))
    print("LOSSES:" + json.dumps(losses))
else:
    comm = comm_backend(torch.device("cpu"))
    # phase 1: tp2
    spec_a = ParallelSpec(dp=1, tp=2)
    ga, ha = build_llama_train_graph(cfg, 4, 16, dtype=torch.float32,
                                     lr=1e-3, spec=spec_a)
    ctx = prepare_run_context(ga, torch.device("cpu"))
    for i in range(2):
        lv, _ = ga.run([ha["loss"], ha["train_op"]],
                       {ha["input_ids"]: ids,
                        ha["labels"]: labels.reshape(-1)}, ctx=ctx)
        losses.append(float(lv))
    # phase 2: switch to dp2 (params + Adam state migrate live)
    spec_b = ParallelSpec(dp=2, tp=1)
    gb, hb = build_llama_train_graph(cfg, 2, 16, dtype=torch.float32,
                                     lr=1e-3, spec=spec_b)
    # poison B's params so only a real migration can pass
    for p in gb.parameters:
        p.get_data().mul_(0.0)
    switch_graph_params(ga, gb, comm)
    ctxb = prepare_run_context(gb, torch.device("cpu"))
    my_ids = ids[rank * 2:(rank + 1) * 2]
    my_labels = labels[rank * 2:(rank + 1) * 2].reshape(-1)
    for i in range(2):
        lv, _ = gb.run([hb["loss"], hb["train_op"]],
                       {hb["input_ids"]: my_ids,
                        hb["labels"]: my_labels}, ctx=ctxb)
        losses.append(float(lv))
    if rank == 0:
        print("LOSSES:" + json.dumps(losses))
"""


def test_hot_switch_tp2_to_dp2():
    env0 = {**os.environ, "HETU_REPO": REPO, "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29601", "GLOO_SOCKET_IFNAME": "lo"}
    procs = []
    for r in range(2):
        env = dict(env0, RANK=str(r), WORLD_SIZE="2", LOCAL_RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, "-c", WORKER],
                                      env=env, stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    sw = None
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=300)
        assert p.returncode in (0, -6), f"rank {r}: {out}\n{err}"
        for line in out.splitlines():
            if line.startswith("LOSSES:"):
                sw = json.loads(line[7:])
    assert sw is not None
    p = subprocess.run([sys.executable, "-c", WORKER],
                       env={**os.environ, "HETU_REPO": REPO,
                            "WORLD_SIZE": "1"},
                       capture_output=True, text=True, timeout=300)
    single = None
    for line in p.stdout.splitlines():
        if line.startswith("LOSSES:"):
            single = json.loads(line[7:])
    assert single is not None, p.stderr
    assert np.allclose(sw, single, rtol=5e-4, atol=2e-4), \
        f"switch {sw} vs single {single}"
