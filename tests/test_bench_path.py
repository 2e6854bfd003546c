"""Multi-process CPU smoke of the DRIVER'S bench contract (bench.py):
the exact code path the round-end scaling run takes, at world_size 2 over
gloo — auto strategy broadcast, dp branch, pipeline branch."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")


def _run_ws2(extra, port):
    procs = []
    env0 = {**os.environ, "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port), "GLOO_SOCKET_IFNAME": "lo"}
    for r in range(2):
        env = dict(env0, RANK=str(r), WORLD_SIZE="2", LOCAL_RANK=str(r))
        procs.append(subprocess.Popen(
            [sys.executable, BENCH, "--allow-cpu", "--model", "gpt-tiny",
             "--gpus", "2", "--steps", "2", "--warmup", "1",
             "--seq-len", "32", "--global-batch", "8"] + extra,
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True))
    line = None
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=600)
        ok = p.returncode in (0, -6)
        assert ok, f"rank {r}: rc={p.returncode}\n{out}\n{err}"
        for ln in out.splitlines():
            if ln.startswith("{"):
                line = json.loads(ln)
    return line


def test_bench_ws2_auto():
    out = _run_ws2([], 29661)
    assert out is not None
    assert out["n_gpus"] == 2
    assert out["value"] > 0
    assert out["config"]["parallelism"].startswith("dp")
    assert out["loss"] is not None


def test_bench_ws2_pp2():
    out = _run_ws2(["--parallel", "pp2_dp1", "--micro-batch", "1"], 29671)
    assert out is not None
    assert out["config"]["parallelism"] == "dp1_pp2"
    assert out["value"] > 0


def test_bench_ws8_auto():
    """8-rank gloo smoke of the exact round-end N=8 invocation shape:
    auto search at ws=8, strategy broadcast to all ranks, whole-job
    aggregation."""
    procs = []
    env0 = {**os.environ, "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29671", "GLOO_SOCKET_IFNAME": "lo"}
    for r in range(8):
        env = dict(env0, RANK=str(r), WORLD_SIZE="8", LOCAL_RANK=str(r))
        procs.append(subprocess.Popen(
            [sys.executable, BENCH, "--allow-cpu", "--model", "gpt-tiny",
             "--gpus", "8", "--steps", "2", "--warmup", "1",
             "--seq-len", "32", "--global-batch", "16"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            text=True))
    line = None
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=900)
        ok = p.returncode in (0, -6)
        assert ok, f"rank {r}: rc={p.returncode}\n{out}\n{err}"
        for ln in out.splitlines():
            if ln.startswith("{"):
                line = json.loads(ln)
    assert line is not None and line["n_gpus"] == 8
    assert line["value"] > 0
