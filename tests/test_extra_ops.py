"""Extended op families vs torch references (fwd + grads)."""
import torch

from hetu_amd.graph.graph import DefineAndRunGraph, push_graph, pop_graph
from hetu_amd.graph.ops import api as ht
from hetu_amd.engine.runner import prepare_run_context


def _run(build, feeds, wrt=None):
    g = DefineAndRunGraph("t")
    push_graph(g)
    try:
        phs, out = build()
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    fetches = [out]
    if wrt is not None:
        grads = g.gradients([out], [phs[i] for i in wrt])
        fetches += grads
    vals = g.run(fetches, dict(zip(phs, feeds)), ctx=ctx)
    return vals


def test_einsum_fwd_bwd():
    a = torch.randn(4, 5, requires_grad=True)
    b = torch.randn(5, 6, requires_grad=True)

    def build():
        x = ht.placeholder((4, 5), name="a")
        y = ht.placeholder((5, 6), name="b")
        return [x, y], ht.reduce_sum(ht.einsum("ij,jk->ik", x, y))
    out, ga, gb = _run(build, [a.detach(), b.detach()], wrt=[0, 1])
    ref = torch.einsum("ij,jk->ik", a, b).sum()
    ref.backward()
    assert torch.allclose(out, ref.detach(), rtol=1e-5)
    assert torch.allclose(ga, a.grad, rtol=1e-5)
    assert torch.allclose(gb, b.grad, rtol=1e-5)


def test_conv2d_bwd():
    x = torch.randn(2, 3, 8, 8, requires_grad=True)
    w = torch.randn(4, 3, 3, 3, requires_grad=True)

    def build():
        xp = ht.placeholder((2, 3, 8, 8), name="x")
        wp = ht.placeholder((4, 3, 3, 3), name="w")
        return [xp, wp], ht.reduce_sum(ht.conv2d(xp, wp, padding=1))
    out, gx, gw = _run(build, [x.detach(), w.detach()], wrt=[0, 1])
    ref = torch.nn.functional.conv2d(x, w, padding=1).sum()
    ref.backward()
    assert torch.allclose(out, ref.detach(), rtol=1e-4)
    assert torch.allclose(gx, x.grad, rtol=1e-4, atol=1e-5)
    assert torch.allclose(gw, w.grad, rtol=1e-4, atol=1e-5)


def test_pools_and_norms():
    x = torch.randn(2, 3, 8, 8)

    def build():
        xp = ht.placeholder((2, 3, 8, 8), name="x")
        return [xp], ht.max_pool2d(xp, 2)
    (out,) = _run(build, [x])
    assert torch.allclose(out, torch.nn.functional.max_pool2d(x, 2))

    w = torch.ones(3)
    b = torch.zeros(3)

    def build2():
        xp = ht.placeholder((2, 3, 8, 8), name="x")
        wp = ht.placeholder((3,), name="w")
        bp = ht.placeholder((3,), name="b")
        return [xp, wp, bp], ht.batch_norm(xp, wp, bp)
    (out2,) = _run(build2, [x, w, b])
    ref2 = torch.nn.functional.batch_norm(x, None, None, w, b, training=True)
    assert torch.allclose(out2, ref2, rtol=1e-4, atol=1e-5)


def test_losses():
    p = torch.rand(8).clamp(0.01, 0.99)
    t = torch.rand(8).round()

    def build():
        xp = ht.placeholder((8,), name="x")
        tp = ht.placeholder((8,), name="t")
        return [xp, tp], ht.binary_cross_entropy(xp, tp)
    (out,) = _run(build, [p, t])
    assert torch.allclose(out, torch.nn.functional.binary_cross_entropy(p, t),
                          rtol=1e-5)


def test_manipulation_ops():
    x = torch.randn(4, 6)

    def b1():
        xp = ht.placeholder((4, 6), name="x")
        return [xp], ht.triu(xp, 1)
    (out,) = _run(b1, [x])
    assert torch.equal(out, torch.triu(x, 1))

    def b2():
        xp = ht.placeholder((4, 6), name="x")
        return [xp], ht.reduce_sum(ht.clamp(xp, min=-0.5, max=0.5))
    out, gx = _run(b2, [x], wrt=[0])
    xr = x.clone().requires_grad_(True)
    xr.clamp(-0.5, 0.5).sum().backward()
    assert torch.allclose(gx, xr.grad)

    idx = torch.randint(0, 6, (4, 3))

    def b3():
        xp = ht.placeholder((4, 6), name="x")
        ip = ht.placeholder((4, 3), dtype=torch.int64, name="i")
        return [xp, ip], ht.reduce_sum(ht.gather(xp, 1, ip))
    out, gx = _run(b3, [x, idx], wrt=[0])
    xr = x.clone().requires_grad_(True)
    xr.gather(1, idx).sum().backward()
    assert torch.allclose(gx, xr.grad)

    def b4():
        xp = ht.placeholder((4, 6), name="x")
        return [xp], ht.pad(xp, [1, 2], value=3.0)
    (out,) = _run(b4, [x])
    assert out.shape == (4, 9) and float(out[0, 0]) == 3.0

    def b5():
        xp = ht.placeholder((4, 6), name="x")
        return [xp], ht.roll(xp, 2, 1)
    (out,) = _run(b5, [x])
    assert torch.equal(out, torch.roll(x, 2, 1))


def test_bulk_unary_family():
    """Abs/Ceil/Floor/Round/Sin/Cos/Reciprocal + remaining activations
    (reference graph/ops/<Name>.cc families) with autograd."""
    import torch.nn.functional as F
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,
                                      push_graph)
    from hetu_amd.graph.ops import api as ht
    g = DefineAndRunGraph("u")
    push_graph(g)
    try:
        x = ht.placeholder((4, 5), name="x")
        outs = {"abs": ht.abs_(x), "ceil": ht.ceil(x), "floor": ht.floor(x),
                "round": ht.round_(x), "sin": ht.sin(x), "cos": ht.cos(x),
                "recip": ht.reciprocal(x), "lrelu": ht.leaky_relu(x),
                "mish": ht.mish(x), "elu": ht.elu(x),
                "hshrink": ht.hardshrink(x), "hsig": ht.hardsigmoid(x),
                "hswish": ht.hardswish(x), "htanh": ht.hardtanh(x),
                "lsig": ht.logsigmoid(x), "splus": ht.softplus(x),
                "sshrink": ht.softshrink(x)}
        grads = ht.gradients([ht.reduce_sum(outs["mish"])], [x])
        sp = ht.split(x, 5, dim=1)
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    xd = torch.randn(4, 5).abs() + 0.5
    keys = list(outs)
    res = g.run([outs[k] for k in keys] + grads + sp, {x: xd}, ctx=ctx)
    ref = {"abs": xd.abs(), "ceil": xd.ceil(), "floor": xd.floor(),
           "round": xd.round(), "sin": xd.sin(), "cos": xd.cos(),
           "recip": xd.reciprocal(), "lrelu": F.leaky_relu(xd, 0.01),
           "mish": F.mish(xd), "elu": F.elu(xd),
           "hshrink": F.hardshrink(xd), "hsig": F.hardsigmoid(xd),
           "hswish": F.hardswish(xd), "htanh": F.hardtanh(xd),
           "lsig": F.logsigmoid(xd), "splus": F.softplus(xd),
           "sshrink": F.softshrink(xd)}
    for i, k in enumerate(keys):
        assert torch.allclose(res[i], ref[k], atol=1e-6), k
    xr = xd.clone().requires_grad_(True)
    F.mish(xr).sum().backward()
    assert torch.allclose(res[len(keys)], xr.grad, atol=1e-5)
    assert torch.allclose(res[len(keys) + 1], xd[:, :1])


def test_outer_dot_diagonal():
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,
                                      push_graph)
    from hetu_amd.graph.ops import api as ht
    g = DefineAndRunGraph("odd")
    push_graph(g)
    try:
        a = ht.placeholder((4,), name="a")
        b = ht.placeholder((4,), name="b")
        o, dt = ht.outer(a, b), ht.dot(a, b)
        m = ht.placeholder((3, 3), name="m")
        d = ht.diagonal(m)
        go = ht.gradients([ht.reduce_sum(o)], [a])
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    av, bv, mv = torch.randn(4), torch.randn(4), torch.randn(3, 3)
    ro, rdt, rd, rga = g.run([o, dt, d, go[0]],
                             {a: av, b: bv, m: mv}, ctx=ctx)
    assert torch.allclose(ro, torch.outer(av, bv))
    assert torch.allclose(rdt, torch.dot(av, bv))
    assert torch.allclose(rd, mv.diagonal())
    assert torch.allclose(rga, bv.sum().expand(4), atol=1e-6) or \
        torch.allclose(rga, torch.full((4,), bv.sum().item()), atol=1e-6)


def test_dropout2d_bool_rangemask_asstrided():
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,
                                      push_graph)
    from hetu_amd.graph.ops import api as ht
    g = DefineAndRunGraph("x")
    push_graph(g)
    try:
        x = ht.placeholder((2, 3, 4), name="x")
        d = ht.dropout2d(x, 0.5, seed=7)
        b = ht.bool_(x)
        r = ht.range_mask(x, 0.0, 1.0)
        a = ht.as_strided(x, (2, 3), (12, 4), 0)
        gs = ht.gradients([ht.reduce_sum(a)], [x])
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    ctx.training = True
    xd = torch.randn(2, 3, 4)
    rd, rb, rr, ra, rg = g.run([d, b, r, a, gs[0]], {x: xd}, ctx=ctx)
    assert rb.dtype == torch.bool
    assert set(rr.unique().tolist()) <= {0.0, 1.0}
    assert torch.allclose(ra, torch.as_strided(xd, (2, 3), (12, 4)))
    exp = torch.zeros_like(xd)
    exp.as_strided((2, 3), (12, 4)).add_(torch.ones(2, 3))
    assert torch.allclose(rg, exp)
    m = (rd != 0)
    assert all(m[n, c].all() or (~m[n, c]).all()
               for n in range(2) for c in range(3))  # whole-channel drops


def test_varlen_attention_matches_per_segment_ref():
    """Packed-varlen attention: segments attend only within themselves
    (reference ParallelAttention packed path)."""
    import math
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,
                                      push_graph)
    from hetu_amd.graph.ops import api as ht
    torch.manual_seed(0)
    T, H, D = 48, 2, 64
    cu = torch.tensor([0, 16, 40, 48], dtype=torch.int32)
    g = DefineAndRunGraph("vl")
    push_graph(g)
    try:
        q = ht.placeholder((T, H, D), name="q")
        k = ht.placeholder((T, H, D), name="k")
        v = ht.placeholder((T, H, D), name="v")
        c = ht.placeholder((4,), dtype=torch.int32, name="cu")
        o = ht.varlen_attention(q, k, v, c)
        gs = ht.gradients([ht.reduce_sum(o)], [q, k, v])
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    qd = torch.randn(T, H, D)
    kd = torch.randn(T, H, D)
    vd = torch.randn(T, H, D)
    res = g.run([o] + gs, {q: qd, k: kd, v: vd, c: cu}, ctx=ctx)
    ref = torch.empty(T, H, D)
    for s0, s1 in zip(cu[:-1].tolist(), cu[1:].tolist()):
        qs = qd[s0:s1].permute(1, 0, 2)
        ks = kd[s0:s1].permute(1, 0, 2)
        vs = vd[s0:s1].permute(1, 0, 2)
        S = s1 - s0
        sc = (qs @ ks.transpose(-1, -2)) / math.sqrt(D)
        mask = torch.ones(S, S, dtype=torch.bool).tril()
        sc = sc.masked_fill(~mask, float("-inf"))
        ref[s0:s1] = (torch.softmax(sc, -1) @ vs).permute(1, 0, 2)
    assert (res[0] - ref).abs().max() < 1e-5
    # cross-segment isolation: perturbing segment 0 must not change the
    # grads of segment 1's tokens
    qd2 = qd.clone()
    qd2[:16] += 1.0
    res2 = g.run(gs, {q: qd2, k: kd, v: vd, c: cu}, ctx=ctx)
    assert torch.allclose(res[2][16:40], res2[1][16:40], atol=1e-5)


def test_non_contiguous_inputs():
    """Ops must accept non-contiguous (transposed/sliced) feeds
    (reference tests/test_non_contig_ops.py)."""
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,
                                      push_graph)
    from hetu_amd.graph.ops import api as ht
    g = DefineAndRunGraph("nc")
    push_graph(g)
    try:
        x = ht.placeholder((8, 16), name="x")
        w = ht.placeholder((4, 16), name="w")
        y = ht.linear(x, w)
        s = ht.softmax(y)
        r = ht.reduce_sum(ht.mul(s, s))
        grads = ht.gradients([r], [x])
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    base_x = torch.randn(16, 8).t()           # transposed view
    base_w = torch.randn(16, 8)[:, ::2].t()   # strided slice view
    assert not base_x.is_contiguous() and not base_w.is_contiguous()
    out = g.run([s, grads[0]], {x: base_x, w: base_w}, ctx=ctx)
    ref = torch.softmax(base_x.contiguous()
                        @ base_w.contiguous().t(), dim=-1)
    assert torch.allclose(out[0], ref, atol=1e-6)
    assert out[1].shape == (8, 16)


def test_matdot_and_dynamic_concat():
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,
                                      push_graph)
    from hetu_amd.graph.ops import api as ht
    g = DefineAndRunGraph("md")
    push_graph(g)
    try:
        a = ht.placeholder((4, 5), name="a")
        b = ht.placeholder((4,), name="b")
        y = ht.mat_dot(a, b)
        gs = ht.gradients([ht.reduce_sum(y)], [a, b])
        c1 = ht.placeholder((2, 3), name="c1")
        c2 = ht.placeholder((4, 3), name="c2")
        cc = ht.dynamic_concat([c1, c2], dim=0)
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    av, bv = torch.randn(4, 5), torch.randn(4)
    r, ga, gb, rc = g.run(
        [y, gs[0], gs[1], cc],
        {a: av, b: bv, c1: torch.ones(1, 3), c2: torch.ones(4, 3)},
        ctx=ctx)
    assert torch.allclose(r, av * bv[:, None])
    assert torch.allclose(ga, bv[:, None].expand(4, 5))
    assert torch.allclose(gb, av.sum(-1))
    assert rc.shape == (6, 3) and rc[1].abs().sum() == 0


def test_transposed_matmul_and_extremum_reduce_grads():
    """Gradients for all four matmul transpose modes and max-reduce
    (previously unimplemented corners)."""
    from hetu_amd.engine.runner import prepare_run_context
    from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,
                                      push_graph)
    from hetu_amd.graph.ops import api as ht
    torch.manual_seed(0)
    for ta, tb in [(False, True), (True, False), (True, True)]:
        g = DefineAndRunGraph("t")
        push_graph(g)
        try:
            A = ht.placeholder((4, 5) if not ta else (5, 4), name="a")
            Bp = ht.placeholder((6, 5) if tb else (5, 6), name="b")
            y = ht.matmul(A, Bp, trans_a=ta, trans_b=tb)
            gs = ht.gradients([ht.reduce_sum(ht.mul(y, y))], [A, Bp])
        finally:
            pop_graph()
        ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
        av = torch.randn(*([4, 5] if not ta else [5, 4])) \
            .requires_grad_(True)
        bv = torch.randn(*([6, 5] if tb else [5, 6])).requires_grad_(True)
        res = g.run(gs, {A: av.detach(), Bp: bv.detach()}, ctx=ctx)
        aa = av.t() if ta else av
        bb = bv.t() if tb else bv
        (aa @ bb).pow(2).sum().backward()
        assert torch.allclose(res[0], av.grad, atol=1e-5), (ta, tb)
        assert torch.allclose(res[1], bv.grad, atol=1e-5), (ta, tb)

    g = DefineAndRunGraph("m")
    push_graph(g)
    try:
        x = ht.placeholder((3, 4), name="x")
        y = ht.reduce_max(x, dim=1)
        gs = ht.gradients([ht.reduce_sum(ht.mul(y, y))], [x])
    finally:
        pop_graph()
    ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
    xv = torch.randn(3, 4).requires_grad_(True)
    r, = g.run(gs, {x: xv.detach()}, ctx=ctx)
    xv.amax(1).pow(2).sum().backward()
    assert torch.allclose(r, xv.grad, atol=1e-5)


def test_fused_mlp_parity():
    """FusedMLPOp (hipBLASLt-epilogue path on GPU; exact composition on
    CPU) vs the composed linear+gelu+linear graph: forward and all grads."""
    import torch
    from hetu_amd.graph.graph import DefineAndRunGraph, push_graph, pop_graph
    from hetu_amd.graph.ops import api as ht
    from hetu_amd.engine.runner import prepare_run_context
    torch.manual_seed(3)
    B, S, H, F = 2, 8, 16, 32
    xw = torch.randn(B, S, H)
    wfc = torch.randn(F, H) * 0.2
    b1 = torch.randn(F) * 0.1
    wproj = torch.randn(H, F) * 0.2
    b2 = torch.randn(H) * 0.1
    outs = {}
    for mode in ("fused", "composed"):
        g = DefineAndRunGraph(mode)
        push_graph(g)
        try:
            x = ht.placeholder((B, S, H), name="x")
            vfc = ht.variable(wfc.clone(), name="wfc")
            v1 = ht.variable(b1.clone(), name="b1")
            vpr = ht.variable(wproj.clone(), name="wproj")
            v2 = ht.variable(b2.clone(), name="b2")
            if mode == "fused":
                y = ht.fused_mlp(x, vfc, v1, vpr, v2)
            else:
                y = ht.add(ht.linear(ht.gelu(ht.linear(x, vfc, v1)), vpr),
                           v2)
            loss = ht.reduce_sum(ht.mul(y, y))
            gs = ht.gradients([loss], [x, vfc, v1, vpr, v2])
        finally:
            pop_graph()
        ctx = prepare_run_context(g, torch.device("cpu"), use_comm=False)
        vals = g.run([y] + gs, {x: xw.clone()}, ctx=ctx)
        outs[mode] = vals
    for i, (a, b) in enumerate(zip(outs["fused"], outs["composed"])):
        assert torch.allclose(a, b, atol=1e-4), (i, (a - b).abs().max())


def test_min_prod_norm_reduce_grads():
    x = torch.randn(3, 5, requires_grad=True)

    def build_min():
        p = ht.placeholder((3, 5), name="x")
        return [p], ht.reduce_sum(ht.reduce_min(p, dim=1))
    out, gx = _run(build_min, [x.detach()], wrt=[0])
    ref = x.min(1).values.sum()
    ref.backward()
    assert torch.allclose(out, ref.detach())
    assert torch.allclose(gx, x.grad, atol=1e-6)

    x2 = torch.randn(3, 4, requires_grad=True)
    x2d = x2.detach().clone()
    x2d[1, 2] = 0.0                          # zero inside the product

    def build_prod():
        p = ht.placeholder((3, 4), name="x")
        return [p], ht.reduce_sum(ht.reduce_prod(p, dim=1))
    out, gx = _run(build_prod, [x2d], wrt=[0])
    xr = x2d.clone().requires_grad_(True)
    ref = xr.prod(1).sum()
    ref.backward()
    assert torch.allclose(out, ref.detach(), atol=1e-6)
    assert torch.allclose(gx, xr.grad, atol=1e-5)

    x3 = torch.randn(4, 6, requires_grad=True)

    def build_norm():
        p = ht.placeholder((4, 6), name="x")
        return [p], ht.reduce_sum(ht.norm(p, p=2, dim=1))
    out, gx = _run(build_norm, [x3.detach()], wrt=[0])
    ref = torch.linalg.vector_norm(x3, 2, dim=1).sum()
    ref.backward()
    assert torch.allclose(out, ref.detach(), atol=1e-5)
    assert torch.allclose(gx, x3.grad, atol=1e-5)


def test_dense_softmax_ce_broadcast_group():
    logits = torch.randn(5, 7, requires_grad=True)
    labels = torch.softmax(torch.randn(5, 7), -1)

    def build():
        lg = ht.placeholder((5, 7), name="lg")
        lb = ht.placeholder((5, 7), name="lb")
        return [lg, lb], ht.softmax_cross_entropy(lg, lb)
    out, g = _run(build, [logits.detach(), labels], wrt=[0])
    ref = -(labels * torch.log_softmax(logits, -1)).sum(-1).mean()
    ref.backward()
    assert torch.allclose(out, ref.detach(), atol=1e-6)
    assert torch.allclose(g, logits.grad, atol=1e-6)

    # broadcast_to: grad sum-reduces back over the expanded dims
    b = torch.randn(1, 4, requires_grad=True)

    def build_b():
        p = ht.placeholder((1, 4), name="b")
        return [p], ht.reduce_sum(ht.mul(ht.broadcast_to(p, (3, 4)), 2.0))
    out, g = _run(build_b, [b.detach()], wrt=[0])
    ref = (b.expand(3, 4) * 2.0).sum()
    ref.backward()
    assert torch.allclose(out, ref.detach())
    assert torch.allclose(g, b.grad)

    # group: one scalar forcing several subgraphs, like/zeros helpers
    def build_g():
        p = ht.placeholder((2, 2), name="p")
        tok = ht.group(ht.ones_like(p), ht.zeros_like(p), ht.exp(p))
        return [p], tok
    (out,) = _run(build_g, [torch.randn(2, 2)])
    assert out.shape == () and float(out) == 0.0


def test_baddbmm_matvec():
    a = torch.randn(3, 4, 5, requires_grad=True)
    b = torch.randn(3, 5, 6, requires_grad=True)
    inp = torch.randn(3, 4, 6, requires_grad=True)

    def build():
        pa = ht.placeholder((3, 4, 5), name="a")
        pb = ht.placeholder((3, 5, 6), name="b")
        pi = ht.placeholder((3, 4, 6), name="i")
        return [pa, pb, pi], ht.reduce_sum(
            ht.baddbmm(pi, pa, pb, beta=0.5, alpha=2.0))
    out, ga, gb, gi = _run(build, [a.detach(), b.detach(), inp.detach()],
                           wrt=[0, 1, 2])
    ref = torch.baddbmm(inp, a, b, beta=0.5, alpha=2.0).sum()
    ref.backward()
    assert torch.allclose(out, ref.detach(), rtol=1e-4)
    assert torch.allclose(ga, a.grad, rtol=1e-4, atol=1e-5)
    assert torch.allclose(gb, b.grad, rtol=1e-4, atol=1e-5)
    assert torch.allclose(gi, inp.grad)

    m = torch.randn(4, 6, requires_grad=True)
    v = torch.randn(6, requires_grad=True)

    def build_mv():
        pm = ht.placeholder((4, 6), name="m")
        pv = ht.placeholder((6,), name="v")
        return [pm, pv], ht.reduce_sum(ht.matvec(pm, pv))
    out, gm, gv = _run(build_mv, [m.detach(), v.detach()], wrt=[0, 1])
    ref = (m @ v).sum()
    ref.backward()
    assert torch.allclose(out, ref.detach(), rtol=1e-5)
    assert torch.allclose(gm, m.grad, atol=1e-6)
    assert torch.allclose(gv, v.grad, atol=1e-5)
