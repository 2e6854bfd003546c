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


GRAD_SWITCH_WORKER = r"""
import json, os, sys, torch
sys.path.insert(0, os.environ["HETU_REPO"])
from hetu_amd.models.llama import LlamaConfig, build_llama_train_graph
from hetu_amd.nn.parallel import ParallelSpec
from hetu_amd.engine.hot_switch_trainer import HotSwitchTrainer
from hetu_amd.parallel.comm import comm_backend

ws = int(os.environ.get("WORLD_SIZE", "1"))
rank = int(os.environ.get("RANK", "0"))
cfg = LlamaConfig(n_layer=2, n_head=4, n_kv_head=4, hidden=64,
                  ffn_hidden=128, vocab=312, max_seq=16)
gen = torch.Generator().manual_seed(31)
ids = torch.randint(0, cfg.vocab, (8, 16), generator=gen)
labels = torch.randint(0, cfg.vocab, (8, 16), generator=gen)
dev = torch.device("cpu")

if ws == 1:
    # reference trajectory: accumulate batch0+batch1 grads, one update
    from hetu_amd.engine.hot_switch_trainer import HotSwitchTrainer

    def build(key):
        return build_llama_train_graph(cfg, 4, 16, dtype=torch.float32,
                                       lr=1e-3)
    tr = HotSwitchTrainer(build, dev, comm=comm_backend(dev))
    h = tr._get("x")[1]
    l0 = tr.step("x", {h["input_ids"]: ids[:4],
                       h["labels"]: labels[:4].reshape(-1)}, level="grad")
    gsum = {k: v.double().sum().item() for k, v in tr._accum.items()}
    print("GSUM:" + json.dumps(gsum))
    l1 = tr.step("x", {h["input_ids"]: ids[4:],
                       h["labels"]: labels[4:].reshape(-1)},
                 level="update")
    l2 = tr.step("x", {h["input_ids"]: ids[:4],
                       h["labels"]: labels[:4].reshape(-1)})
    g = tr.pool["x"][0]
    wsum = sum(p.get_data().double().sum().item() for p in g.parameters)
    print("LOSSES:" + json.dumps([float(l0), float(l1), float(l2), wsum]))
else:
    comm = comm_backend(dev)

    def build(key):
        if key == "tp2":
            return build_llama_train_graph(cfg, 4, 16,
                                           dtype=torch.float32, lr=1e-3,
                                           spec=ParallelSpec(dp=1, tp=2))
        return build_llama_train_graph(cfg, 2, 16, dtype=torch.float32,
                                       lr=1e-3,
                                       spec=ParallelSpec(dp=2, tp=1))
    tr = HotSwitchTrainer(build, dev, comm=comm)
    # batch 0 under tp2: grads accumulate, NO update
    h = tr._get("tp2")[1]
    l0 = tr.step("tp2", {h["input_ids"]: ids[:4],
                         h["labels"]: labels[:4].reshape(-1)},
                 level="grad")
    # switch mid-accumulation -> dp2 (params + Adam + pending grads move)
    tr.switch_to("dp2")
    if rank == 0:
        gsum = {k: v.double().sum().item() for k, v in tr._accum.items()}
        print("GSUM:" + json.dumps(gsum))
    h2 = tr._get("dp2")[1]
    my = ids[4:][rank * 2:(rank + 1) * 2]
    myl = labels[4:][rank * 2:(rank + 1) * 2].reshape(-1)
    l1 = tr.step("dp2", {h2["input_ids"]: my, h2["labels"]: myl},
                 level="update")
    g = tr.pool["dp2"][0]
    wsum = sum(p.get_data().double().sum().item() for p in g.parameters)
    my0 = ids[:4][rank * 2:(rank + 1) * 2]
    my0l = labels[:4][rank * 2:(rank + 1) * 2].reshape(-1)
    l2 = tr.step("dp2", {h2["input_ids"]: my0, h2["labels"]: my0l})
    if rank == 0:
        print("LOSSES:" + json.dumps([float(l0), float(l1), float(l2),
                                      wsum]))
"""


def test_grad_switch_mid_accumulation():
    """SWITCH_ACCUMULATE_GRAD: grads accumulated under tp2 migrate through
    a hot switch to dp2 and the combined update matches the no-switch
    single-process trajectory (reference switch_exec_graph.h:42-54)."""
    env0 = {**os.environ, "HETU_REPO": REPO, "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29611", "GLOO_SOCKET_IFNAME": "lo"}
    env = dict(env0)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    sp = subprocess.run([sys.executable, "-c", GRAD_SWITCH_WORKER],
                        env=env, capture_output=True, text=True,
                        timeout=300)
    assert sp.returncode == 0, sp.stderr
    single = json.loads(sp.stdout.split("LOSSES:")[1].splitlines()[0])
    single_gs = json.loads(sp.stdout.split("GSUM:")[1].splitlines()[0])
    procs = []
    for r in range(2):
        env = dict(env0, RANK=str(r), WORLD_SIZE="2", LOCAL_RANK=str(r))
        procs.append(subprocess.Popen(
            [sys.executable, "-c", GRAD_SWITCH_WORKER], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    sw = gs = None
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=300)
        assert p.returncode in (0, -6), f"rank {r}: {out}\n{err}"
        for ln in out.splitlines():
            if ln.startswith("LOSSES:"):
                sw = json.loads(ln[len("LOSSES:"):])
            elif ln.startswith("GSUM:"):
                gs = json.loads(ln[len("GSUM:"):])
    assert sw is not None and gs is not None
    assert abs(sw[0] - single[0]) < 1e-5, (sw, single)
    # STRONG check: the batch-0 grads accumulated under tp2 and migrated
    # through the switch equal the single-process accumulated grads (a
    # param-level check would be confounded by Adam's first-step
    # sign(m/sqrt(v)) amplification of fp32 reorder noise)
    for k, v in single_gs.items():
        assert abs(gs[k] - v) < 1e-4, (k, gs[k], v)
    # smoke: the combined update keeps training on the same trajectory
    assert abs(sw[3] - single[3]) < 5.0
