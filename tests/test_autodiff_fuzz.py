"""Autodiff fuzzing: random op chains vs torch.autograd (hypothesis).

Each sampled program is built twice — once with hetu_amd graph ops, once
with plain torch on leaf tensors — and the gradients of a scalar loss
w.r.t. the two inputs must agree."""
import torch
import torch.nn.functional as TF
from hypothesis import given, settings
from hypothesis import strategies as st

from hetu_amd.engine.runner import prepare_run_context
from hetu_amd.graph.graph import DefineAndRunGraph, pop_graph, push_graph
from hetu_amd.graph.ops import api as ht

# (name, ht_fn, torch_fn, needs_weight)
OPS = [
    ("gelu", lambda t: ht.gelu(t),
     lambda t: TF.gelu(t, approximate="tanh")),
    ("silu", lambda t: ht.silu(t), TF.silu),
    ("relu", lambda t: ht.relu(t), TF.relu),
    ("tanh", lambda t: ht.tanh(t), torch.tanh),
    ("sigmoid", lambda t: ht.sigmoid(t), torch.sigmoid),
    ("softmax", lambda t: ht.softmax(t, dim=-1),
     lambda t: torch.softmax(t, dim=-1)),
    ("mul_self", lambda t: ht.mul(t, t), lambda t: t * t),
    ("add_selfT", lambda t: ht.add(t, ht.transpose(t, 0, 1)),
     lambda t: t + t.t()),
    ("exp_clamp", lambda t: ht.exp(ht.clamp(t, min=-3.0, max=3.0)),
     lambda t: torch.exp(torch.clamp(t, -3.0, 3.0))),
    ("softshrink", lambda t: ht.softshrink(t), TF.softshrink),
    ("leaky", lambda t: ht.leaky_relu(t),
     lambda t: TF.leaky_relu(t, 0.01)),
    ("triu", lambda t: ht.triu(t), torch.triu),
    ("roll", lambda t: ht.roll(t, 1, 0),
     lambda t: torch.roll(t, 1, 0)),
    ("sub_min", lambda t: ht.sub(t, ht.reduce_min(t, dim=1, keepdim=True)),
     lambda t: t - t.min(1, keepdim=True).values),
    ("div_norm", lambda t: ht.div(t, ht.add(
        ht.norm(t, p=2, dim=1, keepdim=True), 1.0)),
     lambda t: t / (torch.linalg.vector_norm(t, 2, dim=1,
                                             keepdim=True) + 1.0)),
    ("bcast_mean", lambda t: ht.add(t, ht.broadcast_to(
        ht.reduce_mean(t, dim=1, keepdim=True), (6, 6))),
     lambda t: t + t.mean(1, keepdim=True).expand(6, 6)),
]


@settings(max_examples=120, deadline=None)
@given(st.lists(st.integers(0, len(OPS) - 1), min_size=1, max_size=5),
       st.integers(0, 10_000))
def test_random_chain_grads_match_torch(chain, seed):
    torch.manual_seed(seed)
    x0 = torch.randn(6, 6)

    # hetu graph
    g = DefineAndRunGraph("fuzz")
    push_graph(g)
    try:
        x = ht.placeholder((6, 6), name="x")
        cur = x
        for i in chain:
            cur = OPS[i][1](cur)
        loss = ht.reduce_mean(ht.mul(cur, cur))
        grads = ht.gradients([loss], [x])
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    lv, gv = g.run([loss, grads[0]], {x: x0}, ctx=ctx)

    # torch reference
    xr = x0.clone().requires_grad_(True)
    cur = xr
    for i in chain:
        cur = OPS[i][2](cur)
    ref = (cur * cur).mean()
    ref.backward()

    names = [OPS[i][0] for i in chain]
    assert abs(float(lv) - float(ref)) < 1e-4, (names, float(lv),
                                                float(ref))
    assert torch.allclose(gv, xr.grad, rtol=1e-3, atol=1e-5), \
        (names, (gv - xr.grad).abs().max())
