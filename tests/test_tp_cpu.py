"""Tensor-parallel correctness on CPU (gloo, world_size=2): a tp=2 Llama run
(and a tp=2 sequence-parallel run) must match the single-process tp=1 run
bit-for-bit up to fp32 tolerance.  Mirrors the reference CI's
ds-parallel-config GPT runs (tests/ci_test/) at miniature scale."""
import json
import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json, os, sys
sys.path.insert(0, os.environ["HETU_REPO"])
import torch
import hetu_amd as ht
from hetu_amd.models.llama import LlamaConfig, build_llama_train_graph
from hetu_amd.nn.parallel import ParallelSpec
from hetu_amd.engine.runner import prepare_run_context

ws = int(os.environ.get("WORLD_SIZE", "1"))
rank = int(os.environ.get("RANK", "0"))
sp = os.environ.get("HETU_TEST_SP", "0") == "1"
zero = os.environ.get("HETU_TEST_ZERO", "0") == "1"
cp = int(os.environ.get("HETU_TEST_CP", "1"))
dp = int(os.environ.get("HETU_TEST_DP", "1"))
tp = ws // (dp * cp)
cfg = LlamaConfig(n_layer=2, n_head=4, n_kv_head=4, hidden=64,
                  ffn_hidden=128, vocab=312, max_seq=16)
spec = ParallelSpec(dp=dp, tp=tp, cp=cp, sequence_parallel=sp)
B = 4 // dp
SL = 16 // cp                       # local seq chunk per cp rank
g, h = build_llama_train_graph(cfg, micro_batch=B, seq_len=SL,
                               dtype=torch.float32, lr=1e-3, spec=spec,
                               zero=zero)
ctx = prepare_run_context(g, torch.device("cpu"))
gen = torch.Generator().manual_seed(99)
ids = torch.randint(0, cfg.vocab, (4, 16), generator=gen)
labels = torch.randint(0, cfg.vocab, (4 * 16,), generator=gen)
di, ci = spec.my_dp_index(), spec.my_cp_index()
my_ids = ids[di * B:(di + 1) * B, ci * SL:(ci + 1) * SL]
my_labels = labels.reshape(4, 16)[di * B:(di + 1) * B,
                                  ci * SL:(ci + 1) * SL].reshape(-1)
losses = []
for i in range(5):
    lv, _ = g.run([h["loss"], h["train_op"]],
                  {h["input_ids"]: my_ids, h["labels"]: my_labels}, ctx=ctx)
    losses.append(float(lv.item()))
if rank == 0:
    print("LOSSES:" + json.dumps(losses))
"""


def _launch_once(ws, extra_env, port):
    base_env = dict(os.environ)
    base_env["HETU_REPO"] = REPO
    base_env["MASTER_ADDR"] = "127.0.0.1"
    base_env["MASTER_PORT"] = str(port)
    base_env.update(extra_env)
    procs = []
    for rank in range(ws):
        env = dict(base_env)
        if ws > 1:
            env["RANK"] = str(rank)
            env["WORLD_SIZE"] = str(ws)
            env["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, "-c", WORKER], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    losses = None
    for rank, p in enumerate(procs):
        out, err = p.communicate(timeout=300)
        # gloo teardown may SIGABRT (-6) after a clean run
        ok = p.returncode in (0, -6)
        assert ok, f"rank {rank} failed:\n{out}\n{err}"
        for line in out.splitlines():
            if line.startswith("LOSSES:"):
                losses = json.loads(line[len("LOSSES:"):])
    assert losses is not None
    return losses


def _launch(ws, extra_env, port):
    """Retry once: gloo occasionally SIGABRTs in teardown after a clean
    run (non-deterministic; results already printed)."""
    try:
        return _launch_once(ws, extra_env, port)
    except AssertionError:
        return _launch_once(ws, extra_env, port + 40)


@pytest.fixture(scope="module")
def single_losses():
    return _launch(1, {"HETU_TEST_DP": "1"}, 29531)


def test_tp2_matches_single(single_losses):
    tp_losses = _launch(2, {"HETU_TEST_DP": "1"}, 29532)
    assert np.allclose(tp_losses, single_losses, rtol=2e-4, atol=1e-4), \
        f"tp2 {tp_losses} vs single {single_losses}"
    assert tp_losses[-1] < tp_losses[0]


def test_tp2_sequence_parallel_matches_single(single_losses):
    sp_losses = _launch(2, {"HETU_TEST_DP": "1", "HETU_TEST_SP": "1"}, 29533)
    assert np.allclose(sp_losses, single_losses, rtol=2e-4, atol=1e-4), \
        f"tp2+sp {sp_losses} vs single {single_losses}"


def test_dp2_llama_matches_single(single_losses):
    dp_losses = _launch(2, {"HETU_TEST_DP": "2"}, 29534)
    assert np.allclose(dp_losses, single_losses, rtol=2e-4, atol=1e-4), \
        f"dp2 {dp_losses} vs single {single_losses}"


def test_dp2_zero_matches_single(single_losses):
    """ZeRO-sharded optimizer states (reduce-scatter + all-gather) must
    produce the same training trajectory as plain dp."""
    z_losses = _launch(2, {"HETU_TEST_DP": "2", "HETU_TEST_ZERO": "1"},
                       29535)
    assert np.allclose(z_losses, single_losses, rtol=2e-4, atol=1e-4), \
        f"dp2+zero {z_losses} vs single {single_losses}"


def test_cp2_ring_attention_matches_single(single_losses):
    """Context parallelism: cp=2 ring attention over gloo must match the
    single-process run (seq chunks per rank)."""
    cp_losses = _launch(2, {"HETU_TEST_DP": "1", "HETU_TEST_CP": "2"}, 29536)
    assert np.allclose(cp_losses, single_losses, rtol=5e-4, atol=2e-4), \
        f"cp2 {cp_losses} vs single {single_losses}"


def test_cp2_tp2_matches_single(single_losses):
    """cp2 x tp2 (4 ranks): ring attention across cp groups whose members
    are tensor-parallel (heads sharded) — both layouts must compose."""
    losses = _launch(4, {"HETU_TEST_DP": "1", "HETU_TEST_CP": "2"}, 29538)
    assert np.allclose(losses, single_losses, rtol=5e-4, atol=2e-4), \
        f"cp2xtp2 {losses} vs single {single_losses}"


def test_dp2_cp2_matches_single(single_losses):
    """dp2 x cp2 (4 ranks): data-parallel replicas each running a cp ring."""
    losses = _launch(4, {"HETU_TEST_DP": "2", "HETU_TEST_CP": "2"}, 29539)
    assert np.allclose(losses, single_losses, rtol=5e-4, atol=2e-4), \
        f"dp2xcp2 {losses} vs single {single_losses}"


def test_dp2_tp2_zero_matches_single(single_losses):
    """dp2 x tp2 with ZeRO (4 ranks): optimizer-state sharding over the
    dp groups composes with tensor parallelism."""
    losses = _launch(4, {"HETU_TEST_DP": "2", "HETU_TEST_ZERO": "1"},
                     29540)
    assert np.allclose(losses, single_losses, rtol=5e-4, atol=2e-4), \
        f"dp2xtp2+zero {losses} vs single {single_losses}"


def test_dp2_tp2_sp_matches_single(single_losses):
    """dp2 x tp2 with sequence parallelism (4 ranks)."""
    losses = _launch(4, {"HETU_TEST_DP": "2", "HETU_TEST_SP": "1"}, 29545)
    assert np.allclose(losses, single_losses, rtol=5e-4, atol=2e-4), \
        f"dp2xtp2+sp {losses} vs single {single_losses}"
