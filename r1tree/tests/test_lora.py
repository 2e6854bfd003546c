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
