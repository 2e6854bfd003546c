"""LoRA adapters: base frozen, adapters train, merge == adapter path."""
import torch

from hetu_amd.graph.graph import DefineAndRunGraph, push_graph, pop_graph
from hetu_amd.graph.ops import api as ht
from hetu_amd.graph.ops.optim import Adam
from hetu_amd.nn.parallel import ColumnParallelLinear, ParallelSpec
from hetu_amd.peft.lora import LoRALinear
from hetu_amd.engine.runner import prepare_run_context


def _build():
    g = DefineAndRunGraph("lora")
    push_graph(g)
    try:
        spec = ParallelSpec()
        x = ht.placeholder((4, 16), name="x")
        tgt = ht.placeholder((4, 8), name="tgt")
        base = ColumnParallelLinear(16, 8, spec, bias=False,
                                    dtype=torch.float32, name="base")
        lora = LoRALinear(base, r=4, alpha=8.0)
        y = lora(x)
        loss = ht.mse_loss(y, tgt)
        opt = Adam(lr=1e-2)
        train_op = opt.minimize(loss)
    finally:
        pop_graph()
    return g, x, tgt, y, loss, train_op, base, lora


def test_lora_trains_adapters_only():
    torch.manual_seed(0)
    g, x, tgt, y, loss, train_op, base, lora = _build()
    # only A and B are trainable
    names = sorted(p.name.split(":")[0] for p in g.parameters)
    assert names == ["lora.A", "lora.B"], names
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    xd, td = torch.randn(4, 16), torch.randn(4, 8)
    w_before = base.weight.get_data().clone()
    losses = []
    for _ in range(30):
        lv, _ = g.run([loss, train_op], {x: xd, tgt: td}, ctx=ctx)
        losses.append(float(lv))
    assert losses[-1] < losses[0] * 0.9
    assert torch.equal(base.weight.get_data(), w_before)  # frozen


def test_lora_merge_matches():
    torch.manual_seed(1)
    g, x, tgt, y, loss, train_op, base, lora = _build()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    xd, td = torch.randn(4, 16), torch.randn(4, 8)
    for _ in range(5):
        g.run([loss, train_op], {x: xd, tgt: td}, ctx=ctx)
    y_adapter = g.run([y], {x: xd}, ctx=ctx)[0].clone()
    lora.merge()
    # after merging, the BASE path alone equals the adapter path
    y_base = base.weight.get_data() @ xd.t()
    assert torch.allclose(y_adapter, y_base.t(), rtol=1e-4, atol=1e-5)
    lora.unmerge()


def test_multi_task_lora_isolation():
    """LobRA: per-task adapters over one frozen base — training task a
    leaves task b's adapters and the base untouched."""
    from hetu_amd.graph.ops.optim import Adam as _Adam
    from hetu_amd.peft.multi_task import MultiLoRALinear

    torch.manual_seed(2)
    g = DefineAndRunGraph("mlora")
    push_graph(g)
    try:
        spec = ParallelSpec()
        x = ht.placeholder((4, 16), name="x")
        tgt = ht.placeholder((4, 8), name="tgt")
        base = ColumnParallelLinear(16, 8, spec, bias=False,
                                    dtype=torch.float32, name="base")
        ml = MultiLoRALinear(base, {"a": 4, "b": 2}, alpha=8.0)
        ya, yb = ml(x, "a"), ml(x, "b")
        la = ht.mse_loss(ya, tgt)
        lb = ht.mse_loss(yb, tgt)
        opa = _Adam(lr=1e-2)
        ta = opa.minimize(la, params=ml.task_parameters("a"))
    finally:
        pop_graph()
    names = sorted(p.name.split(":")[0] for p in g.parameters)
    assert names == ["mlora.a.A", "mlora.a.B", "mlora.b.A", "mlora.b.B"]
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    xd, td = torch.randn(4, 16), torch.randn(4, 8)
    w0 = base.weight.get_data().clone()
    b_A0 = ml.A["b"].get_data().clone()
    losses = []
    for _ in range(30):
        lv, _ = g.run([la, ta], {x: xd, tgt: td}, ctx=ctx)
        losses.append(float(lv))
    assert losses[-1] < losses[0] * 0.9
    assert torch.equal(base.weight.get_data(), w0)
    assert torch.equal(ml.A["b"].get_data(), b_A0)


def test_lobra_planners_and_scheduler():
    from hetu_amd.peft.multi_task import (TaskBatchScheduler, balance_plan,
                                          group_plan, prune_plan)

    # balance: proportional under budget, exact total
    plan = balance_plan({"a": 3.0, "b": 1.0}, rank_budget=16, r_min=2)
    assert sum(plan.values()) == 16 and plan["a"] > plan["b"] >= 2
    # group: k groups, shared rank = member max
    groups = group_plan({"a": 8, "b": 8, "c": 2, "d": 2}, 2)
    assert len(groups) == 2
    all_tasks = sorted(t for _, g_ in groups for t in g_)
    assert all_tasks == ["a", "b", "c", "d"]
    for gmax, members in groups:
        assert gmax == max(8 if m in ("a", "b") else 2 for m in members)
    # prune: halves lowest utility-per-rank until within budget
    pruned = prune_plan({"a": 16, "b": 16}, {"a": 10.0, "b": 1.0}, 24)
    assert sum(pruned.values()) <= 24 and pruned["b"] < pruned["a"]
    # scheduler: frequencies track data sizes; deterministic
    sched = TaskBatchScheduler({"big": 3, "small": 1})
    seq = sched.schedule(40)
    assert seq.count("big") == 30 and seq.count("small") == 10
    assert TaskBatchScheduler({"big": 3, "small": 1}).schedule(40) == seq
