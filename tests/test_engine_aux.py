"""Engine aux: SFT masking, straggler rebalance, dynamic planner, config."""
import torch

from hetu_amd.engine.sft_trainer import build_sft_example
from hetu_amd.engine.straggler import detect_stragglers, rebalance_micro_batches
from hetu_amd.engine.dynamic_planner import DynamicPlanner
from hetu_amd.engine.trainer_config import TrainingConfig
from hetu_amd.galvatron.cost_model import ModelShape, Strategy


def test_sft_example_masks_prompt():
    x, y = build_sft_example([1, 2, 3], [4, 5], seq_len=8)
    assert x.tolist()[:4] == [1, 2, 3, 4]
    assert y.tolist()[:4] == [-100, -100, 4, 5]
    assert all(v == -100 for v in y.tolist()[4:])


def test_straggler_detection_and_rebalance():
    times = [1.0, 1.0, 3.0, 1.0]
    assert detect_stragglers(times) == [2]
    mb = rebalance_micro_batches(times, 12)
    assert sum(mb) == 12
    assert mb[2] < mb[0]           # straggler gets less work


def test_dynamic_planner_prefers_big_strategy_for_long_seq():
    shape = ModelShape(n_layer=8, hidden=1024, ffn_hidden=4096, vocab=32000,
                       n_head=16)
    cands = [Strategy(dp=8, micro_batch=1),
             Strategy(dp=2, tp=4, micro_batch=1)]
    pl = DynamicPlanner(shape, 8, cands, buckets=[512, 4096])
    plan = pl.plan([128, 256, 4096, 3000])
    assert set(plan.keys()) <= {512, 4096}
    assert 512 in plan and 4096 in plan


def test_config_yaml(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text("model: gpt2-345m\nglobal_batch: 32\ntp: 2\n"
                 "precision: bf16\nunknown_key: 5\n")
    cfg = TrainingConfig.from_yaml(str(p))
    assert cfg.model == "gpt2-345m" and cfg.tp == 2
    assert cfg.dtype() == torch.bfloat16


def test_pipeline_memory_snapshots(monkeypatch):
    """HETU_AMD_MEM_PROFILE=1 records a snapshot per micro-batch fwd/bwd
    (reference per-micro-batch CUDAProfiler memory info)."""
    import subprocess
    import sys
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = r"""
import os, sys, torch
sys.path.insert(0, os.environ["HETU_REPO"])
os.environ["HETU_AMD_MEM_PROFILE"] = "1"
from hetu_amd.models.llama import LlamaConfig, build_llama_pipeline_stage
from hetu_amd.parallel.pipeline import PipelineSpec, PipelineRunner
cfg = LlamaConfig(n_layer=2, n_head=4, n_kv_head=4, hidden=64,
                  ffn_hidden=96, vocab=211, max_seq=16)
pspec = PipelineSpec(pp=1, dp=1, tp=1)
stage = build_llama_pipeline_stage(cfg, pspec, micro_batch=1, seq_len=16,
                                   dtype=torch.float32, lr=1e-3)
runner = PipelineRunner(pspec, stage, torch.device("cpu"))
h = stage.h
mbs = [{h["input_ids"]: torch.randint(0, 211, (1, 16)),
        h["labels"]: torch.randint(0, 211, (16,))} for _ in range(3)]
runner.step(mbs)
assert runner.mem_snapshots is not None
rep = runner.mem_snapshots.report()
assert "fwd_mb0" in rep and "bwd_mb" in rep, rep
print("SNAP OK")
"""
    p = subprocess.run([sys.executable, "-c", code],
                       env={**os.environ, "HETU_REPO": repo},
                       capture_output=True, text=True, timeout=300)
    assert p.returncode == 0 and "SNAP OK" in p.stdout, \
        f"{p.stdout}\n{p.stderr}"


def test_check_numeric_guard(monkeypatch):
    """HETU_AMD_CHECK_NUMERIC: the executor raises at the first op whose
    output goes non-finite, naming that op (reference CheckNumeric)."""
    import pytest
    import hetu_amd as ht
    from hetu_amd.graph import executor as ex

    monkeypatch.setattr(ex, "_CHECK_NUMERIC", True)
    with ht.graph("define_and_run") as g:
        x = ht.placeholder((4,), name="x")
        y = ht.log(x)                     # log(-1) -> NaN
        y.producer.name = "bad_log"
        z = ht.add(y, y)
        with pytest.raises(ex.NonFiniteError) as ei:
            g.run([z], {x: torch.tensor([1.0, -1.0, 2.0, 3.0])})
        assert ei.value.op_name == "bad_log"
        assert ei.value.out_index == 0
        # clean inputs pass untouched
        (out,) = g.run([z], {x: torch.tensor([1.0, 1.0, 2.0, 3.0])})
        assert torch.isfinite(out).all()


def test_torch_profiler_trace(tmp_path, monkeypatch):
    """HETU_AMD_TORCH_PROFILE writes a chrome trace of steps 2-4
    (reference trainer wires torch.profiler for host tracing)."""
    import glob
    import hetu_amd as ht
    from hetu_amd.engine.trainer import Trainer

    monkeypatch.setenv("HETU_AMD_TORCH_PROFILE", str(tmp_path))
    with ht.graph("define_and_run") as g:
        x = ht.placeholder((4, 8), name="x")
        w = ht.variable(torch.randn(8, 8), name="w")
        loss = ht.reduce_mean(ht.matmul(x, w))
        opt = ht.Adam(lr=1e-3)
        train_op = opt.minimize(loss)
        tr = Trainer(g, {"loss": loss, "train_op": train_op},
                     torch.device("cpu"))
        for _ in range(6):
            tr.step({x: torch.randn(4, 8)})
    traces = glob.glob(str(tmp_path / "*.json")) + \
        glob.glob(str(tmp_path / "**" / "*.json"), recursive=True)
    assert traces, "no chrome trace written"


def test_training_config_from_ds_parallel_json(tmp_path):
    """A TrainingConfig yaml can point at a ds_parallel_config JSON; the
    strategy (pp/dp/tp/zero) is read from the file."""
    from hetu_amd.engine.trainer_config import TrainingConfig
    from hetu_amd.utils.ds_config import (generate_ds_parallel_config,
                                          write_ds_parallel_config)
    dsp = str(tmp_path / "ds.json")
    write_ds_parallel_config(
        generate_ds_parallel_config([(2, 2), (2, 2)], num_layers=4), dsp)
    yml = tmp_path / "run.yaml"
    yml.write_text(f"model: gpt-tiny\nds_parallel_config: {dsp}\n")
    tc = TrainingConfig.from_yaml(str(yml))
    assert (tc.pp, tc.dp, tc.tp) == (2, 2, 2)
    assert tc.zero is True


def test_lr_schedule_exact_scaling():
    """A schedule multiplier of 0.5 must produce EXACTLY the trajectory
    of half the base lr (the multiplier folds into the Adam update), and
    schedules compose with the Trainer step loop."""
    import hetu_amd as ht
    from hetu_amd.engine.lr_schedule import (cosine_with_warmup,
                                             linear_warmup)
    from hetu_amd.engine.trainer import Trainer
    from hetu_amd.graph.ops.optim import AdamStepOp

    def build(lr):
        torch.manual_seed(3)
        with ht.graph("define_and_run") as g:
            x = ht.placeholder((4, 8), name="x")
            w = ht.variable(torch.randn(8, 8), name="w")
            loss = ht.reduce_mean(ht.pow(ht.matmul(x, w), 2))
            train = ht.Adam(lr=lr).minimize(loss)
        return g, {"loss": loss, "train_op": train}, x, w

    try:
        xd = torch.randn(4, 8)
        g1, h1, x1, w1 = build(0.05)
        t1 = Trainer(g1, h1, torch.device("cpu"))
        for _ in range(4):
            t1.step({x1: xd})
        g2, h2, x2, w2 = build(0.10)
        t2 = Trainer(g2, h2, torch.device("cpu"),
                     lr_schedule=lambda s: 0.5)
        for _ in range(4):
            t2.step({x2: xd})
        assert torch.allclose(w1.get_data(), w2.get_data(), atol=1e-6)
        # warmup actually ramps: step-0 update is smaller than lr
        m = linear_warmup(4)
        assert m(0) == 0.25 and m(3) == 1.0 and m(10) == 1.0
        c = cosine_with_warmup(2, 10, min_ratio=0.1)
        assert c(0) == 0.5 and abs(c(10) - 0.1) < 1e-6
    finally:
        AdamStepOp.set_lr_scale(1.0)


def test_training_config_lr_schedule(tmp_path):
    from hetu_amd.engine.trainer_config import TrainingConfig
    yml = tmp_path / "r.yaml"
    yml.write_text("lr: 0.001\nlr_warmup_steps: 10\nlr_decay: cosine\n"
                   "steps: 100\n")
    tc = TrainingConfig.from_yaml(str(yml))
    f = tc.lr_schedule()
    assert f(0) < f(9) == 1.0 and f(99) <= 0.11
    assert TrainingConfig().lr_schedule() is None


def test_leveled_logger(monkeypatch, capsys):
    import importlib
    monkeypatch.setenv("HETU_AMD_LOG_LEVEL", "DEBUG")
    from hetu_amd.utils import logging as hl
    lg = hl.get_logger("hetu_amd.test")
    lg.debug("dbg %d", 1)
    lg.warning("warn")
    err = capsys.readouterr().err
    assert "dbg 1" in err and "warn" in err and "r0" in err


def test_package_import_hygiene():
    """Importing the package surface must not initialize distributed
    state or CUDA (workers decide that), and the public namespaces
    resolve."""
    import subprocess
    import sys
    code = (
        "import hetu_amd as ht\n"
        "from hetu_amd import nn, optim, models, data, parallel, utils\n"
        "from hetu_amd.engine import Trainer, TrainingConfig\n"
        "import torch, torch.distributed as dist\n"
        "assert not dist.is_initialized()\n"
        "assert not torch.cuda.is_initialized()\n"
        "print('HYGIENE_OK')\n")
    p = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=120)
    assert p.returncode == 0 and "HYGIENE_OK" in p.stdout, \
        f"{p.stdout}\n{p.stderr}"
