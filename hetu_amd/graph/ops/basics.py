"""Core op set: sources, arithmetic, shape ops, reductions, GEMM family.

Re-designs the reference op families (/root/reference/hetu/graph/ops/ —
Arithmetics.cc, matmul.cc, Linear.cc, BatchMatMul.cc, Reshape/Transpose/
Slice/Concat, Reduce.cc, variable.cc, placeholder.cc, sum.cc) for torch-ROCm
execution. Each op implements infer_meta / compute / gradient and, where the
layout is nontrivial, deduce_states for SPMD propagation.
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch

from ...core.symbol import IntSymbol, resolve_dim, resolve_shape
from ...parallel.dstates import DistributedStates, ds_from_index_table
from ..op import Op, OpInterface
from ..tensor import Tensor, TensorMeta


def _g(t: Tensor):
    return t.graph


def _make(graph, iface, inputs, attrs=None, name="", **kw):
    return graph.make_op(iface, inputs, attrs or {}, name=name, **kw)


# ---------------------------------------------------------------------------
# Sources
# ---------------------------------------------------------------------------

class PlaceholderOp(OpInterface):
    type = "Placeholder"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(attrs["shape"], attrs["dtype"])]

    def compute(self, op, inputs, ctx):
        raise RuntimeError("placeholder must be fed")


class VariableOp(OpInterface):
    type = "Variable"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(attrs["shape"], attrs["dtype"])]

    def compute(self, op, inputs, ctx):
        data = op.outputs[0].get_data()
        if data is None:
            raise RuntimeError(f"variable {op.name} not initialized")
        return [data]

    def gradient(self, op, grad_outputs):
        return []


class ConstantOp(OpInterface):
    type = "Constant"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(attrs["shape"], attrs["dtype"])]

    def compute(self, op, inputs, ctx):
        a = op.attrs
        return [torch.full(resolve_shape(a["shape"]), a["value"],
                           dtype=a["dtype"], device=ctx.device)]


class OnesLikeOp(OpInterface):
    type = "OnesLike"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype)]

    def deduce_states(self, op):
        # gradient-seed semantics: d(sum)/d(partial contribution) == 1
        # everywhere, so a partial input yields a *duplicate* seed.
        src = op.inputs[0].ds
        if src is not None:
            states = dict(src.states)
            if -2 in states:
                states[-1] = states.get(-1, 1) * states.pop(-2)
            order = [-1 if d == -2 else d for d in src.order]
            dedup = []
            for d in order:
                if d not in dedup:
                    dedup.append(d)
            op.outputs[0].ds = DistributedStates(src.device_num, states, dedup)
        op.outputs[0].device_group = op.inputs[0].device_group

    def compute(self, op, inputs, ctx):
        return [torch.ones_like(inputs[0])]


class ZerosLikeOp(OpInterface):
    type = "ZerosLike"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        return [torch.zeros_like(inputs[0])]


def make_ones_like(graph, t: Tensor) -> Tensor:
    return _make(graph, OnesLikeOp(), [t], name=f"ones_like({t.name})").output()


class BroadcastToOp(OpInterface):
    """Expand to a target shape (reference Broadcast.cc); the gradient
    sum-reduces back over the broadcast dims (ReduceToShapeOp)."""
    type = "BroadcastTo"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(tuple(attrs["shape"]), inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        return [inputs[0].expand(*op.attrs["shape"]).contiguous()]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, ReduceToShapeOp(), [g[0], op.inputs[0]],
                      name="bcast_grad").output()]


# ---------------------------------------------------------------------------
# Elementwise arithmetic (broadcasting; grads reduce back to input shape)
# ---------------------------------------------------------------------------

def _bcast_shape(s1, s2):
    out = []
    l1, l2 = len(s1), len(s2)
    for i in range(max(l1, l2)):
        d1 = s1[l1 - 1 - i] if i < l1 else 1
        d2 = s2[l2 - 1 - i] if i < l2 else 1
        if isinstance(d1, IntSymbol) or isinstance(d2, IntSymbol):
            out.append(d1 if not (isinstance(d1, int) and d1 == 1) else d2)
        else:
            out.append(max(d1, d2))
    return tuple(reversed(out))


def _reduce_to_shape(graph, g: Tensor, target: Tensor) -> Tensor:
    if tuple(g.shape) == tuple(target.shape):
        return g
    return _make(graph, ReduceToShapeOp(), [g, target],
                 name=f"reduce_to({target.name})").output()


class ReduceToShapeOp(OpInterface):
    """Sum-reduce a broadcasted gradient back to the shape of inputs[1]."""
    type = "ReduceToShape"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[0].dtype)]

    def deduce_states(self, op):
        g, ref = op.inputs
        out = op.outputs[0]
        out.device_group = ref.device_group or g.device_group
        if g.ds is None:
            out.ds = ref.ds
            return
        n = g.ds.device_num
        # dims of g that get summed away: leading extras + broadcast dims
        gnd, rnd = g.ndim, ref.ndim
        reduced = set(range(gnd - rnd))
        for i in range(rnd):
            gd = gnd - rnd + i
            try:
                if (not isinstance(ref.shape[i], IntSymbol)
                        and not isinstance(g.shape[gd], IntSymbol)
                        and ref.shape[i] == 1 and g.shape[gd] != 1):
                    reduced.add(gd)
            except Exception:  # noqa: BLE001
                pass
        table = []
        counts: Dict[int, int] = {}
        npart = g.ds.partial
        for d in g.ds.split_dims():
            if d in reduced:
                npart *= g.ds.get_dim(d)
        for i in range(n):
            st = g.ds.map_device_to_state_index(i)
            ent: Dict[int, int] = {}
            ip = st.get(-2, 0)
            for d in g.ds.split_dims():
                if d in reduced:
                    ip = ip * g.ds.get_dim(d) + st.get(d, 0)
                else:
                    rd = d - (gnd - rnd)
                    if g.ds.get_dim(d) > 1:
                        ent[rd] = st.get(d, 0)
                        counts[rd] = g.ds.get_dim(d)
            if npart > 1:
                ent[-2] = ip
            table.append(ent)
        counts[-2] = npart
        out.ds = ds_from_index_table(n, table, counts)

    def compute(self, op, inputs, ctx):
        g, ref = inputs
        tgt = list(ref.shape)
        lead = g.ndim - len(tgt)
        if lead > 0:
            # leading-dim reduction through the hand colsum kernel: the
            # at::native column reduce returns garbage from the 2nd replay
            # of a captured step on some shapes (ROCm 7.2, see
            # profiles/r02_capture_replay_bug.md) — and colsum is faster.
            from ...ops import functional as F
            rest = g.shape[lead:]
            n = 1
            for d in rest:
                n *= int(d)
            dt = g.dtype
            g = F.colsum(g.reshape(-1, n)).to(dt).reshape(rest)
        for i, d in enumerate(tgt):
            if g.shape[i] != d:
                g = g.sum(i, keepdim=True)
        return [g]


class _BinaryOp(OpInterface):
    def infer_meta(self, attrs, inputs):
        dtype = inputs[0].dtype
        return [TensorMeta(_bcast_shape(inputs[0].shape, inputs[1].shape),
                           dtype)]

    def deduce_states(self, op):
        """Broadcast-aware elementwise SPMD rule: a dim split on one side
        must be matched by an equal split (same device->shard mapping) or a
        broadcast (absent / size-1) dim on the other; partial inputs are
        rejected (they must be comm'ed first)."""
        a, b = op.inputs[0], op.inputs[1]
        out = op.outputs[0]
        out.device_group = a.device_group or b.device_group
        if a.ds is None and b.ds is None:
            out.ds = None
            return
        if a.ds is None or b.ds is None:
            src_t = a if a.ds is not None else b
            # the un-annotated side is replicated; output inherits the
            # annotated layout (split dims right-aligned to output rank).
            out.ds = _bcast_ds(src_t.ds, src_t.ndim, out.meta.ndim)
            return
        if a.ds.partial > 1 or b.ds.partial > 1:
            raise ValueError(
                f"elementwise op {op.name} on partial input (a={a.ds}, "
                f"b={b.ds}): insert a comm op first")
        n = a.ds.device_num
        ond = out.meta.ndim
        table = []
        counts: Dict[int, int] = {}
        for i in range(n):
            sa = a.ds.map_device_to_state_index(i)
            sb = b.ds.map_device_to_state_index(i)
            ent: Dict[int, int] = {}
            for od in range(ond):
                ad = od - (ond - a.ndim)
                bd = od - (ond - b.ndim)
                na = a.ds.get_dim(ad) if ad >= 0 else 1
                nb = b.ds.get_dim(bd) if bd >= 0 else 1
                # broadcast dims (size 1) cannot be split
                if na > 1 and nb > 1:
                    if na != nb or sa.get(ad, 0) != sb.get(bd, 0):
                        raise ValueError(
                            f"misaligned splits on dim {od} of {op.name}")
                nn = max(na, nb)
                if nn > 1:
                    ent[od] = sa.get(ad, 0) if na > 1 else sb.get(bd, 0)
                    counts[od] = nn
            table.append(ent)
        out.ds = ds_from_index_table(n, table, counts)


def _bcast_ds(src: DistributedStates, src_tensor_ndim: int, out_ndim: int
              ) -> DistributedStates:
    """Shift src's split dims to the output rank (right-aligned)."""
    shift = out_ndim - src_tensor_ndim
    if shift == 0:
        return src
    states = {(d + shift if d >= 0 else d): c for d, c in src.states.items()}
    order = [(d + shift if d >= 0 else d) for d in src.order]
    return DistributedStates(src.device_num, states, order)


class AddOp(_BinaryOp):
    type = "Add"

    def compute(self, op, inputs, ctx):
        return [inputs[0] + inputs[1]]

    def gradient(self, op, g):
        gy = g[0]
        gr = _g(op.outputs[0])
        return [_reduce_to_shape(gr, gy, op.inputs[0]),
                _reduce_to_shape(gr, gy, op.inputs[1])]


class SubOp(_BinaryOp):
    type = "Sub"

    def compute(self, op, inputs, ctx):
        return [inputs[0] - inputs[1]]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_reduce_to_shape(gr, g[0], op.inputs[0]),
                _reduce_to_shape(gr, make_neg(gr, g[0]), op.inputs[1])]


class MulOp(_BinaryOp):
    type = "Mul"

    def compute(self, op, inputs, ctx):
        return [inputs[0] * inputs[1]]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        a, b = op.inputs
        ga = _make(gr, MulOp(), [g[0], b]).output()
        gb = _make(gr, MulOp(), [g[0], a]).output()
        return [_reduce_to_shape(gr, ga, a), _reduce_to_shape(gr, gb, b)]


class DivOp(_BinaryOp):
    type = "Div"

    def compute(self, op, inputs, ctx):
        return [inputs[0] / inputs[1]]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        a, b = op.inputs
        ga = _make(gr, DivOp(), [g[0], b]).output()
        # gb = -g * a / b^2
        gb_num = _make(gr, MulOp(), [g[0], a]).output()
        b2 = _make(gr, MulOp(), [b, b]).output()
        gb = make_neg(gr, _make(gr, DivOp(), [gb_num, b2]).output())
        return [_reduce_to_shape(gr, ga, a), _reduce_to_shape(gr, gb, b)]


class _ScalarOp(OpInterface):
    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype)]


class AddScalarOp(_ScalarOp):
    type = "AddScalar"

    def compute(self, op, inputs, ctx):
        return [inputs[0] + op.attrs["value"]]

    def gradient(self, op, g):
        return [g[0]]


class MulScalarOp(_ScalarOp):
    type = "MulScalar"

    def compute(self, op, inputs, ctx):
        return [inputs[0] * op.attrs["value"]]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, MulScalarOp(), [g[0]],
                      {"value": op.attrs["value"]}).output()]


class PowScalarOp(_ScalarOp):
    type = "PowScalar"

    def compute(self, op, inputs, ctx):
        return [inputs[0] ** op.attrs["value"]]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        p = op.attrs["value"]
        xp = _make(gr, PowScalarOp(), [op.inputs[0]], {"value": p - 1}).output()
        gx = _make(gr, MulOp(), [g[0], xp]).output()
        return [_make(gr, MulScalarOp(), [gx], {"value": p}).output()]


class NegOp(_ScalarOp):
    type = "Neg"

    def compute(self, op, inputs, ctx):
        return [-inputs[0]]

    def gradient(self, op, g):
        return [make_neg(_g(op.outputs[0]), g[0])]


def make_neg(graph, t):
    return _make(graph, NegOp(), [t]).output()


class _UnaryTorch(_ScalarOp):
    fn = None

    def compute(self, op, inputs, ctx):
        return [self.__class__.fn(inputs[0])]


class ExpOp(_UnaryTorch):
    type = "Exp"
    fn = torch.exp

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, MulOp(), [g[0], op.outputs[0]]).output()]


class LogOp(_UnaryTorch):
    type = "Log"
    fn = torch.log

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, DivOp(), [g[0], op.inputs[0]]).output()]


class SqrtOp(_UnaryTorch):
    type = "Sqrt"
    fn = torch.sqrt

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        half = _make(gr, MulScalarOp(), [op.outputs[0]], {"value": 2.0}).output()
        return [_make(gr, DivOp(), [g[0], half]).output()]


class RsqrtOp(_UnaryTorch):
    type = "Rsqrt"
    fn = torch.rsqrt

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        y3 = _make(gr, PowScalarOp(), [op.outputs[0]], {"value": 3.0}).output()
        gx = _make(gr, MulOp(), [g[0], y3]).output()
        return [_make(gr, MulScalarOp(), [gx], {"value": -0.5}).output()]


class AddNOp(OpInterface):
    """n-ary sum (reference sum.cc) — grad accumulation node."""
    type = "AddN"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        out = inputs[0].clone()
        for t in inputs[1:]:
            out += t
        return [out]

    def gradient(self, op, g):
        return [g[0]] * len(op.inputs)


def make_add_n(graph, ts: List[Tensor]) -> Tensor:
    return _make(graph, AddNOp(), ts, name="grad_sum").output()


# ---------------------------------------------------------------------------
# Shape ops
# ---------------------------------------------------------------------------

class ReshapeOp(OpInterface):
    type = "Reshape"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(attrs["shape"], inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        shape = resolve_shape(op.attrs["shape"])
        # allow a single -1
        return [inputs[0].reshape(shape)]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, ReshapeOp(), [g[0]],
                      {"shape": op.inputs[0].shape}).output()]


class TransposeOp(OpInterface):
    type = "Transpose"

    def infer_meta(self, attrs, inputs):
        d0, d1 = attrs["dim0"], attrs["dim1"]
        shape = list(inputs[0].shape)
        shape[d0], shape[d1] = shape[d1], shape[d0]
        return [TensorMeta(shape, inputs[0].dtype)]

    def deduce_states(self, op):
        src = op.inputs[0].ds
        if src is not None:
            d0, d1 = op.attrs["dim0"], op.attrs["dim1"]
            nd = op.inputs[0].ndim
            d0 = d0 % nd
            d1 = d1 % nd
            states = {}
            for d, n in src.states.items():
                nd_ = d
                if d == d0:
                    nd_ = d1
                elif d == d1:
                    nd_ = d0
                states[nd_] = n
            order = [d1 if d == d0 else d0 if d == d1 else d
                     for d in src.order]
            op.outputs[0].ds = DistributedStates(src.device_num, states, order)
        op.outputs[0].device_group = op.inputs[0].device_group

    def compute(self, op, inputs, ctx):
        return [inputs[0].transpose(op.attrs["dim0"], op.attrs["dim1"])
                .contiguous()]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, TransposeOp(), [g[0]],
                      {"dim0": op.attrs["dim0"], "dim1": op.attrs["dim1"]}
                      ).output()]


class SliceOp(OpInterface):
    """Slice along one dim: [start, start+length)."""
    type = "Slice"

    def infer_meta(self, attrs, inputs):
        shape = list(inputs[0].shape)
        shape[attrs["dim"]] = attrs["length"]
        return [TensorMeta(shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        a = op.attrs
        return [inputs[0].narrow(a["dim"], resolve_dim(a["start"]),
                                 resolve_dim(a["length"])).contiguous()]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, SliceGradOp(), [g[0], op.inputs[0]],
                      dict(op.attrs)).output()]


class SliceGradOp(OpInterface):
    type = "SliceGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        g, ref = inputs
        out = torch.zeros_like(ref)
        a = op.attrs
        out.narrow(a["dim"], resolve_dim(a["start"]),
                   resolve_dim(a["length"])).copy_(g)
        return [out]


class ConcatOp(OpInterface):
    type = "Concat"

    def infer_meta(self, attrs, inputs):
        dim = attrs["dim"]
        shape = list(inputs[0].shape)
        total = 0
        symbolic = False
        for t in inputs:
            d = t.shape[dim]
            if isinstance(d, IntSymbol):
                symbolic = True
                break
            total += d
        if symbolic:
            shape[dim] = inputs[0].shape[dim]
        else:
            shape[dim] = total
        return [TensorMeta(shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        return [torch.cat(inputs, dim=op.attrs["dim"])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        dim = op.attrs["dim"]
        grads = []
        start = 0
        for t in op.inputs:
            grads.append(_make(gr, SliceOp(), [g[0]],
                               {"dim": dim, "start": start,
                                "length": t.shape[dim]}).output())
            start += resolve_dim(t.shape[dim]) if not isinstance(
                t.shape[dim], IntSymbol) else 0
        return grads


class ContiguousOp(_ScalarOp):
    type = "Contiguous"

    def compute(self, op, inputs, ctx):
        return [inputs[0].contiguous()]

    def gradient(self, op, g):
        return [g[0]]


class CastOp(OpInterface):
    """dtype transfer (reference data_transfer.cc)."""
    type = "Cast"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, attrs["dtype"])]

    def compute(self, op, inputs, ctx):
        return [inputs[0].to(op.attrs["dtype"])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, CastOp(), [g[0]],
                      {"dtype": op.inputs[0].dtype}).output()]


# ---------------------------------------------------------------------------
# Reductions
# ---------------------------------------------------------------------------

class ReduceOp_(OpInterface):
    type = "Reduce"

    def infer_meta(self, attrs, inputs):
        dim = attrs["dim"]
        keepdim = attrs["keepdim"]
        shape = list(inputs[0].shape)
        if dim is None:
            shape = [1] * len(shape) if keepdim else []
        else:
            dims = [dim] if isinstance(dim, int) else list(dim)
            dims = [d % len(shape) for d in dims]
            if keepdim:
                for d in dims:
                    shape[d] = 1
            else:
                shape = [s for i, s in enumerate(shape) if i not in dims]
        dtype = inputs[0].dtype
        return [TensorMeta(shape, dtype)]

    def deduce_states(self, op):
        src = op.inputs[0].ds
        if src is not None:
            dim = op.attrs["dim"]
            keepdim = op.attrs["keepdim"]
            nd = op.inputs[0].ndim
            dims = (list(range(nd)) if dim is None
                    else [dim % nd] if isinstance(dim, int)
                    else [d % nd for d in dim])
            states = {}
            part = src.partial
            for d, n in src.states.items():
                if d >= 0 and d in dims:
                    if op.attrs["mode"] in ("sum", "mean"):
                        part *= n     # reduced over a split dim -> partial
                    else:
                        raise ValueError(
                            "max/min/prod reduce over a split dim has no "
                            "partial-sum representation")
                elif d >= 0:
                    nd_ = d - sum(1 for r in dims if r < d) if not keepdim else d
                    states[nd_] = n
                elif d == -1:
                    states[-1] = n
            if part > 1:
                states[-2] = part
            order = []
            for d in src.order:
                if d >= 0 and d in dims:
                    order.append(-2)
                elif d >= 0 and not keepdim:
                    order.append(d - sum(1 for r in dims if r < d))
                else:
                    order.append(d)
            op.outputs[0].ds = DistributedStates(src.device_num, states,
                                                 [d for d in order])
        op.outputs[0].device_group = op.inputs[0].device_group

    def _split_factor(self, op):
        """Product of input splits over the reduced dims: a mean over a
        split dim must divide by the GLOBAL count so that the partial-sum
        representation (sum over ranks == global mean) holds."""
        src = op.inputs[0].ds
        if src is None:
            return 1
        dim = op.attrs["dim"]
        nd = op.inputs[0].ndim
        dims = (list(range(nd)) if dim is None
                else [dim % nd] if isinstance(dim, int)
                else [d % nd for d in dim])
        f = 1
        for d in dims:
            f *= src.get_dim(d)
        return f

    def compute(self, op, inputs, ctx):
        x = inputs[0]
        mode = op.attrs["mode"]
        dim = op.attrs["dim"]
        keepdim = op.attrs["keepdim"]
        if mode == "sum":
            return [x.sum() if dim is None else x.sum(dim, keepdim=keepdim)]
        if mode == "mean":
            y = x.mean() if dim is None else x.mean(dim, keepdim=keepdim)
            f = self._split_factor(op)
            if f > 1:
                y = y / f
            return [y]
        if mode == "max":
            if dim is None:
                return [x.max()]
            return [x.max(dim, keepdim=keepdim).values]
        if mode == "min":
            if dim is None:
                return [x.min()]
            return [x.min(dim, keepdim=keepdim).values]
        if mode == "prod":
            if dim is None:
                return [x.prod()]
            assert isinstance(dim, int), "prod reduces one dim at a time"
            return [x.prod(dim, keepdim=keepdim)]
        raise ValueError(mode)

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        mode = op.attrs["mode"]
        if mode in ("max", "min"):
            # subgradient routed to the extremal positions
            # (split across ties, matching the mask/count convention)
            return [_make(gr, ReduceExtremumGradOp(),
                          [g[0], op.inputs[0], op.outputs[0]],
                          dict(op.attrs)).output()]
        if mode == "prod":
            return [_make(gr, ReduceProdGradOp(),
                          [g[0], op.inputs[0]],
                          dict(op.attrs)).output()]
        return [_make(gr, ReduceGradOp(), [g[0], op.inputs[0]],
                      dict(op.attrs)).output()]


class ReduceExtremumGradOp(OpInterface):
    """dx for max/min reduce: gy spread over argext positions, ties
    sharing equally."""
    type = "ReduceExtremumGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[1].dtype)]

    def compute(self, op, inputs, ctx):
        gy, x, y = inputs
        dim = op.attrs.get("dim")
        keepdim = op.attrs.get("keepdim", False)
        xf = x.float()
        if dim is None:
            mask = (xf == y.float()).to(xf.dtype)
            return [(mask / mask.sum().clamp(min=1)
                     * gy.float()).to(x.dtype)]
        ye = y.float() if keepdim else y.float().unsqueeze(dim)
        ge = gy.float() if keepdim else gy.float().unsqueeze(dim)
        mask = (xf == ye).to(xf.dtype)
        return [(mask / mask.sum(dim, keepdim=True).clamp(min=1)
                 * ge).to(x.dtype)]


class ReduceProdGradOp(OpInterface):
    """dx for prod reduce.  Exact including zeros: replays the reduction
    under torch.autograd on a detached leaf (cold path; reference
    Reduce.cc prod grad)."""
    type = "ReduceProdGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[1].dtype)]

    def compute(self, op, inputs, ctx):
        gy, x = inputs
        dim = op.attrs.get("dim")
        keepdim = op.attrs.get("keepdim", False)
        with torch.enable_grad():
            xl = x.detach().float().requires_grad_(True)
            y = xl.prod() if dim is None else xl.prod(dim, keepdim=keepdim)
            (dx,) = torch.autograd.grad(y, xl, gy.float())
        return [dx.to(x.dtype)]


class CheckFiniteOp(OpInterface):
    """All-finite flag over N tensors (reference CheckFinite.cu + the AMP
    inf-check in gradscaler.cc): output fp32 scalar 1.0 iff every element
    of every input is finite."""
    type = "CheckFinite"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta([], torch.float32)]

    def deduce_states(self, op):
        op.outputs[0].ds = None
        op.outputs[0].device_group = (op.inputs[0].device_group
                                      if op.inputs else None)

    def compute(self, op, inputs, ctx):
        dev = inputs[0].device if inputs else "cpu"
        ok = torch.ones((), device=dev)
        for t in inputs:
            ok = ok * torch.isfinite(t).all().to(ok.dtype)
        return [ok]


class ReduceGradOp(OpInterface):
    type = "ReduceGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[0].dtype)]

    def deduce_states(self, op):
        op.outputs[0].ds = op.inputs[1].ds
        op.outputs[0].device_group = op.inputs[1].device_group

    def compute(self, op, inputs, ctx):
        g, ref = inputs
        mode = op.attrs["mode"]
        dim = op.attrs["dim"]
        keepdim = op.attrs["keepdim"]
        nd = ref.ndim
        if dim is None:
            dims = list(range(nd))
        else:
            dims = [dim % nd] if isinstance(dim, int) else [d % nd for d in dim]
        if not keepdim:
            for d in sorted(dims):
                g = g.unsqueeze(d)
        out = g.expand_as(ref).contiguous()
        if mode == "mean":
            n = 1
            for d in dims:
                n *= ref.shape[d]
            src = op.inputs[1].ds
            if src is not None:
                for d in dims:
                    n *= src.get_dim(d)   # global count over split dims
            out = out / n
        return [out]


# ---------------------------------------------------------------------------
# GEMM family — matmul / linear / batched matmul
# hand-written MFMA path via ops.functional.linear
# ---------------------------------------------------------------------------

def _deduce_matmul_ds(op, x: Tensor, w: Tensor, out: Tensor,
                      x_k_dims, w_k_dims, x_pass, w_pass):
    """Shared DS deduction for matmul-like ops via per-device index tables.

    x_k_dims / w_k_dims: paired contraction dims (a split on any pair makes
    the output partial).  x_pass / w_pass: {input_dim: output_dim} maps for
    dims whose split carries through (batch/row/col dims)."""
    dsx, dsw = x.ds, w.ds
    if dsx is None and dsw is None:
        return
    n = (dsx or dsw).device_num
    if dsx is None:
        dsx = DistributedStates(n, {-1: n} if n > 1 else {})
    if dsw is None:
        dsw = DistributedStates(n, {-1: n} if n > 1 else {})
    if dsx.device_num != dsw.device_num:
        raise ValueError("matmul inputs on different-size device groups")
    kx = 1
    for dx_, dw_ in zip(x_k_dims, w_k_dims):
        if dsx.get_dim(dx_) != dsw.get_dim(dw_):
            raise ValueError(
                f"contraction-dim splits differ: x[{dx_}] "
                f"{dsx.get_dim(dx_)} vs w[{dw_}] {dsw.get_dim(dw_)}")
        kx *= dsx.get_dim(dx_)
    npart = dsx.partial * dsw.partial * kx
    counts = {-2: npart}
    for d_in, d_out in x_pass.items():
        counts[d_out] = counts.get(d_out, 1) * dsx.get_dim(d_in)
    for d_in, d_out in w_pass.items():
        counts[d_out] = counts.get(d_out, 1) * dsw.get_dim(d_in)
    table = []
    for i in range(n):
        sx = dsx.map_device_to_state_index(i)
        sw = dsw.map_device_to_state_index(i)
        ipart = sx.get(-2, 0) * dsw.partial + sw.get(-2, 0)
        for dx_, dw_ in zip(x_k_dims, w_k_dims):
            if sx.get(dx_, 0) != sw.get(dw_, 0):
                raise ValueError(
                    "contraction shard indices differ between operands")
            ipart = ipart * dsx.get_dim(dx_) + sx.get(dx_, 0)
        ent = {}
        for d_in, d_out in x_pass.items():
            if dsx.get_dim(d_in) > 1:
                ent[d_out] = sx.get(d_in, 0)
        for d_in, d_out in w_pass.items():
            if dsw.get_dim(d_in) > 1:
                ent[d_out] = ent.get(d_out, 0) * dsw.get_dim(d_in) \
                    + sw.get(d_in, 0)
        if npart > 1:
            ent[-2] = ipart
        table.append(ent)
    out.ds = ds_from_index_table(n, table, counts)
    out.device_group = x.device_group or w.device_group


class LinearOp(OpInterface):
    """y = x @ W^T (+ b); W stored [out_features, in_features] (torch
    convention). inputs: x [..., K], w [N, K], optional bias [N]."""
    type = "Linear"

    def infer_meta(self, attrs, inputs):
        x, w = inputs[0], inputs[1]
        shape = list(x.shape[:-1]) + [w.shape[0]]
        return [TensorMeta(shape, x.dtype)]

    def deduce_states(self, op):
        x, w = op.inputs[0], op.inputs[1]
        nd = x.ndim
        _deduce_matmul_ds(op, x, w, op.outputs[0],
                          x_k_dims=[nd - 1], w_k_dims=[1],
                          x_pass={d: d for d in range(nd - 1)},
                          w_pass={0: nd - 1})

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        x, w = inputs[0], inputs[1]
        b = inputs[2] if len(inputs) > 2 else None
        return [F.linear(x, w, b, trans_w=True)]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        gy = g[0]
        x, w = op.inputs[0], op.inputs[1]
        # dx = gy @ W          : Linear with w not transposed
        dx = _make(gr, MatMul2DOp(), [gy, w], {"trans_a": False,
                                               "trans_b": False},
                   name="linear_dx").output()
        # dw = gy^T @ x  (flattened over leading dims)
        dw = _make(gr, MatMulGradWOp(), [gy, x], name="linear_dw").output()
        grads = [dx, dw]
        if len(op.inputs) > 2:
            db = _make(gr, ReduceLeadingOp(), [gy, op.inputs[2]],
                       name="linear_db").output()
            grads.append(db)
        return grads


class MatMul2DOp(OpInterface):
    """General matmul on last two dims with optional transposes.
    a [..., M, K] @ b [K, N] -> [..., M, N]."""
    type = "MatMul"

    def infer_meta(self, attrs, inputs):
        a, b = inputs
        ta, tb = attrs.get("trans_a", False), attrs.get("trans_b", False)
        ash = list(a.shape)
        bsh = list(b.shape)
        m = ash[-1] if ta else ash[-2] if len(ash) >= 2 else 1
        n = bsh[-2] if tb else bsh[-1]
        if len(ash) == 2 and len(bsh) == 2:
            shape = [m, n]
        else:
            shape = list(ash[:-1]) + [n]
        return [TensorMeta(shape, a.dtype)]

    def deduce_states(self, op):
        a, b = op.inputs
        ta = op.attrs.get("trans_a", False)
        tb = op.attrs.get("trans_b", False)
        nda, ndb = a.ndim, b.ndim
        a_k = nda - 2 if ta else nda - 1
        a_row = nda - 1 if ta else nda - 2
        x_pass = {d: d for d in range(nda - 2)}
        x_pass[a_row] = nda - 2
        _deduce_matmul_ds(
            op, a, b, op.outputs[0],
            x_k_dims=[a_k],
            w_k_dims=[ndb - 1 if tb else ndb - 2],
            x_pass=x_pass,
            w_pass={(ndb - 2 if tb else ndb - 1): nda - 1})

    def compute(self, op, inputs, ctx):
        a, b = inputs
        if op.attrs.get("trans_a", False):
            a = a.transpose(-1, -2)
        if op.attrs.get("trans_b", False):
            b = b.transpose(-1, -2)
        return [torch.matmul(a, b)]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        a, b = op.inputs
        ta = op.attrs.get("trans_a", False)
        tb = op.attrs.get("trans_b", False)
        gy = g[0]
        if not ta and not tb:
            da = _make(gr, MatMul2DOp(), [gy, b],
                       {"trans_a": False, "trans_b": True}).output()
            db = _make(gr, MatMulGradBOp(), [a, gy]).output()
            return [da, db]
        # y = op_a(a) @ op_b(b): standard transposed-matmul adjoints
        # (2-D operands; the batched leading-dim case routes through the
        # untransposed path above)
        if not ta and tb:       # y = a  @ b^T
            da = _make(gr, MatMul2DOp(), [gy, b], {}).output()
            db = _make(gr, MatMul2DOp(), [gy, a],
                       {"trans_a": True}).output()
        elif ta and not tb:     # y = a^T @ b
            da = _make(gr, MatMul2DOp(), [b, gy],
                       {"trans_b": True}).output()
            db = _make(gr, MatMul2DOp(), [a, gy], {}).output()
        else:                   # y = a^T @ b^T
            da = _make(gr, MatMul2DOp(), [b, gy],
                       {"trans_a": True, "trans_b": True}).output()
            db = _make(gr, MatMul2DOp(), [gy, a],
                       {"trans_a": True, "trans_b": True}).output()
        return [da, db]


class MatMulGradWOp(OpInterface):
    """dw = gy^T @ x with gy [..., N], x [..., K] flattened -> [N, K]."""
    type = "MatMulGradW"

    def infer_meta(self, attrs, inputs):
        gy, x = inputs
        return [TensorMeta([gy.shape[-1], x.shape[-1]], x.dtype)]

    def deduce_states(self, op):
        gy, x = op.inputs
        nd = gy.ndim
        # ALL leading (token) dims of gy/x are the contraction; gy cols ->
        # out dim 0; x cols -> out dim 1
        _deduce_matmul_ds(op, gy, x, op.outputs[0],
                          x_k_dims=list(range(nd - 1)),
                          w_k_dims=list(range(x.ndim - 1)),
                          x_pass={nd - 1: 0},
                          w_pass={x.ndim - 1: 1})

    def compute(self, op, inputs, ctx):
        gy, x = inputs
        gy2 = gy.reshape(-1, gy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        return [torch.matmul(gy2.t(), x2)]


class MatMulGradBOp(OpInterface):
    """db = a^T @ gy flattened: a [..., K] gy [..., N] -> [K, N]."""
    type = "MatMulGradB"

    def infer_meta(self, attrs, inputs):
        a, gy = inputs
        return [TensorMeta([a.shape[-1], gy.shape[-1]], a.dtype)]

    def deduce_states(self, op):
        a, gy = op.inputs
        _deduce_matmul_ds(op, a, gy, op.outputs[0],
                          x_k_dims=list(range(a.ndim - 1)),
                          w_k_dims=list(range(gy.ndim - 1)),
                          x_pass={a.ndim - 1: 0},
                          w_pass={gy.ndim - 1: 1})

    def compute(self, op, inputs, ctx):
        a, gy = inputs
        a2 = a.reshape(-1, a.shape[-1])
        g2 = gy.reshape(-1, gy.shape[-1])
        return [torch.matmul(a2.t(), g2)]


class ReduceLeadingOp(OpInterface):
    """Sum over all leading dims to match inputs[1]'s (1-D) shape — bias
    gradient."""
    type = "ReduceLeading"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[0].dtype)]

    def deduce_states(self, op):
        gy, ref = op.inputs
        out = op.outputs[0]
        if gy.ds is not None:
            n = gy.ds.device_num
            table = []
            ncol = gy.ds.get_dim(gy.ndim - 1)
            # token splits become partial
            npart = gy.ds.partial
            for d in gy.ds.split_dims():
                if d != gy.ndim - 1:
                    npart *= gy.ds.get_dim(d)
            for i in range(n):
                st = gy.ds.map_device_to_state_index(i)
                ipart = st.get(-2, 0)
                for d in gy.ds.split_dims():
                    if d != gy.ndim - 1:
                        ipart = ipart * gy.ds.get_dim(d) + st.get(d, 0)
                ent = {}
                if ncol > 1:
                    ent[0] = st.get(gy.ndim - 1, 0)
                if npart > 1:
                    ent[-2] = ipart
                table.append(ent)
            out.ds = ds_from_index_table(n, table, {0: ncol, -2: npart})
        out.device_group = ref.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        gy = inputs[0]
        out = F.colsum(gy.reshape(-1, gy.shape[-1]))
        return [out.to(gy.dtype)]


class BatchMatMulOp(OpInterface):
    type = "BatchMatMul"

    def infer_meta(self, attrs, inputs):
        a, b = inputs
        shape = list(a.shape[:-1]) + [b.shape[-1]]
        return [TensorMeta(shape, a.dtype)]

    def compute(self, op, inputs, ctx):
        return [torch.matmul(inputs[0], inputs[1])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        a, b = op.inputs
        gy = g[0]
        da = _make(gr, BatchMatMulTNOp(), [gy, b], {"mode": "nt"}).output()
        db = _make(gr, BatchMatMulTNOp(), [a, gy], {"mode": "tn"}).output()
        return [da, db]


class BatchMatMulTNOp(OpInterface):
    type = "BatchMatMulTN"

    def infer_meta(self, attrs, inputs):
        a, b = inputs
        if attrs["mode"] == "nt":   # a @ b^T
            shape = list(a.shape[:-1]) + [b.shape[-2]]
        else:                        # a^T @ b
            shape = list(a.shape[:-2]) + [a.shape[-1], b.shape[-1]]
        return [TensorMeta(shape, a.dtype)]

    def compute(self, op, inputs, ctx):
        # On GPU both transposed batched modes materialize the transpose
        # and run the NN path: the hipBLASLt strided-batched NT kernel
        # reads ~1 MB past the end of the transposed operand on the MoE
        # grad shapes ([8,640,2048] @ [8,2048,8192]^T bf16) — isolated on
        # MI355X via serialized repro (fault address = b.ptr + numel*2 +
        # 1MB; both inputs clone cleanly).  Normally the overread lands in
        # allocator slack; at a mapping boundary it faults.  The NN route
        # (same shapes as the forward expert bmm) is fault-free.
        a, b = inputs
        if op.attrs["mode"] == "nt":
            bt = b.transpose(-1, -2)
            if a.is_cuda:
                bt = bt.contiguous()
            return [torch.matmul(a, bt)]
        at = a.transpose(-1, -2)
        if a.is_cuda:
            at = at.contiguous()
        return [torch.matmul(at, b)]
