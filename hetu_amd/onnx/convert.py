"""ONNX graph export/import over the minimal proto layer.

Supported op subset (both directions): Linear/Gemm, MatMul, Add, Sub,
Mul, Div (+Scalar variants as rank-0 initializers), Relu, Gelu, Sigmoid,
Tanh, Exp, Log, Sqrt, Abs, Neg, Pow, Softmax, Reshape, Transpose, Concat,
Slice, Where, Bool, Reduce{Sum,Mean,Max,Min,Prod}, LayerNormalization,
Gather (embedding), Cast.  Reference parity:
hetu/v1/python/hetu/onnx/{onnx2hetu,hetu2onnx}.
"""
from __future__ import annotations

import struct
from typing import Dict, List, Sequence, Tuple

import numpy as np
import torch

from . import proto as P


_TORCH2DT = {torch.float32: P.DT_FLOAT, torch.int64: P.DT_INT64,
             torch.int32: P.DT_INT32, torch.bool: P.DT_BOOL}
_DT2TORCH = {v: k for k, v in _TORCH2DT.items()}


def _tensor_raw(t: torch.Tensor) -> bytes:
    return t.detach().cpu().contiguous().numpy().tobytes()


# ---------------------------------------------------------------------------
# export
# ---------------------------------------------------------------------------
def export_onnx(graph, fetches: Sequence, path: str = None) -> bytes:
    """Serialize the subgraph producing `fetches` to ONNX bytes (and write
    to `path` if given).  Placeholders become graph inputs; variables
    become initializers."""
    topo = graph.topo_sort(fetches)
    nodes: List[bytes] = []
    inits: List[bytes] = []
    g_inputs: List[bytes] = []

    def nm(t) -> str:
        n = t.name
        return n[:-2] if n.endswith(":0") else n.replace(":", "_")

    for op in topo:
        t = op.type
        o = nm(op.outputs[0]) if op.outputs else ""
        ins = [nm(x) for x in op.inputs]
        if t == "Placeholder":
            g_inputs.append(P.value_info(
                o, _TORCH2DT[op.outputs[0].dtype],
                list(op.outputs[0].shape)))
        elif t in ("Variable", "Constant"):
            data = op.outputs[0].get_data()
            if data is None and t == "Constant":
                data = torch.full(list(op.attrs["shape"]),
                                  op.attrs["value"],
                                  dtype=op.attrs.get("dtype",
                                                     torch.float32))
            inits.append(P.tensor_proto(o, list(data.shape),
                                        _TORCH2DT[data.dtype],
                                        _tensor_raw(data)))
        elif t == "Linear":
            # y = x @ W^T (+ b)  ->  Transpose(W) + MatMul + Add
            wt = o + "/WT"
            nodes.append(P.node_proto(
                "Transpose", [ins[1]], [wt], name=op.name + "/wt",
                attrs={"perm": list(range(op.inputs[1].ndim))[::-1]}))
            mm = o if len(ins) < 3 else o + "/mm"
            nodes.append(P.node_proto("MatMul", [ins[0], wt], [mm],
                                      name=op.name))
            if len(ins) > 2:
                nodes.append(P.node_proto("Add", [mm, ins[2]], [o],
                                          name=op.name + "/bias"))
        elif t == "MatMul":
            a, b = ins
            if op.attrs.get("trans_a"):
                at = o + "/AT"
                nodes.append(P.node_proto(
                    "Transpose", [a], [at], name=op.name + "/at",
                    attrs={"perm": list(range(op.inputs[0].ndim))[::-1]}))
                a = at
            if op.attrs.get("trans_b"):
                bt = o + "/BT"
                nodes.append(P.node_proto(
                    "Transpose", [b], [bt], name=op.name + "/bt",
                    attrs={"perm": list(range(op.inputs[1].ndim))[::-1]}))
                b = bt
            nodes.append(P.node_proto("MatMul", [a, b], [o], name=op.name))
        elif t in ("Add", "Sub", "Mul", "Div", "Relu", "Gelu", "Sigmoid",
                   "Tanh", "Exp", "Log", "Sqrt", "Abs", "Neg"):
            nodes.append(P.node_proto(t, ins, [o], name=op.name))
        elif t == "Concat":
            nodes.append(P.node_proto(
                "Concat", ins, [o], name=op.name,
                attrs={"axis": int(op.attrs.get("dim", 0))}))
        elif t == "Slice":
            # opset-10+ Slice takes starts/ends/axes as inputs
            start = int(op.attrs["start"])
            end = start + int(op.attrs["length"])
            axis = int(op.attrs["dim"])
            extra_ins = []
            for suffix, vals in (("/starts", [start]), ("/ends", [end]),
                                 ("/axes", [axis])):
                nmx = o + suffix
                inits.append(P.tensor_proto(
                    nmx, [1], P.DT_INT64,
                    np.asarray(vals, dtype=np.int64).tobytes()))
                extra_ins.append(nmx)
            nodes.append(P.node_proto("Slice", ins + extra_ins, [o],
                                      name=op.name))
        elif t == "Where":
            nodes.append(P.node_proto("Where", ins, [o], name=op.name))
        elif t == "Bool":
            nodes.append(P.node_proto(
                "Cast", ins, [o], name=op.name,
                attrs={"to": P.DT_BOOL}))
        elif t in ("AddScalar", "SubScalar", "MulScalar", "DivScalar",
                   "PowScalar"):
            # scalar operand becomes a rank-0 initializer
            sc = o + "/scalar"
            val = torch.tensor(float(op.attrs["value"]),
                               dtype=op.outputs[0].dtype)
            inits.append(P.tensor_proto(sc, [], _TORCH2DT[val.dtype],
                                        _tensor_raw(val)))
            onnx_op = {"AddScalar": "Add", "SubScalar": "Sub",
                       "MulScalar": "Mul", "DivScalar": "Div",
                       "PowScalar": "Pow"}[t]
            nodes.append(P.node_proto(onnx_op, [ins[0], sc], [o],
                                      name=op.name))
        elif t == "Softmax":
            nodes.append(P.node_proto("Softmax", ins, [o], name=op.name,
                                      attrs={"axis": op.attrs.get("dim",
                                                                  -1)}))
        elif t == "Reshape":
            shp = o + "/shape"
            shape = [int(s) for s in op.attrs["shape"]]
            inits.append(P.tensor_proto(
                shp, [len(shape)], P.DT_INT64,
                np.asarray(shape, dtype=np.int64).tobytes()))
            nodes.append(P.node_proto("Reshape", [ins[0], shp], [o],
                                      name=op.name))
        elif t == "Transpose":
            perm = list(range(op.inputs[0].ndim))
            d0, d1 = op.attrs["dim0"], op.attrs["dim1"]
            perm[d0], perm[d1] = perm[d1], perm[d0]
            nodes.append(P.node_proto("Transpose", ins, [o], name=op.name,
                                      attrs={"perm": perm}))
        elif t == "LayerNorm":
            nodes.append(P.node_proto(
                "LayerNormalization", ins, [o], name=op.name,
                attrs={"epsilon": float(op.attrs.get("eps", 1e-5)),
                       "axis": -1}))
        elif t == "Embedding":
            nodes.append(P.node_proto("Gather", [ins[0], ins[1]], [o],
                                      name=op.name, attrs={"axis": 0}))
        elif t == "Cast":
            nodes.append(P.node_proto(
                "Cast", ins, [o], name=op.name,
                attrs={"to": _TORCH2DT[op.attrs["dtype"]]}))
        elif t == "Reduce":
            onnx_op = {"sum": "ReduceSum", "mean": "ReduceMean",
                       "max": "ReduceMax", "min": "ReduceMin",
                       "prod": "ReduceProd"}[op.attrs["mode"]]
            attrs = {"keepdims": 1 if op.attrs.get("keepdim") else 0}
            dim = op.attrs.get("dim")
            if dim is not None:
                attrs["axes"] = ([dim] if isinstance(dim, int)
                                 else list(dim))
            nodes.append(P.node_proto(onnx_op, ins, [o], name=op.name,
                                      attrs=attrs))
        else:
            raise NotImplementedError(f"ONNX export: op {t}")

    g_outputs = [P.value_info(nm(f), _TORCH2DT[f.dtype], list(f.shape))
                 for f in fetches]
    g = P.graph_proto(nodes, graph.name or "hetu_amd", inits, g_inputs,
                      g_outputs)
    blob = P.model_proto(g)
    if path:
        with open(path, "wb") as fh:
            fh.write(blob)
    return blob


# ---------------------------------------------------------------------------
# import
# ---------------------------------------------------------------------------
def _parse_attr(buf: bytes):
    m = P.parse_message(buf)
    name = m[1][0].decode()
    at = m.get(20, [0])[0]
    if at == P.ATTR_FLOAT:
        return name, struct.unpack("<f", m[2][0])[0]
    if at == P.ATTR_INT:
        return name, _signed(m[3][0])
    if at == P.ATTR_STRING:
        return name, m[4][0].decode()
    if at == P.ATTR_INTS:
        return name, [_signed(v) for v in m.get(8, [])]
    if at == P.ATTR_FLOATS:
        return name, [struct.unpack("<f", v)[0] for v in m.get(7, [])]
    raise NotImplementedError(f"attr type {at}")


def _signed(v: int) -> int:
    return v - (1 << 64) if v >= (1 << 63) else v


def _parse_tensor(buf: bytes) -> Tuple[str, torch.Tensor]:
    m = P.parse_message(buf)
    dims = [_signed(v) for v in m.get(1, [])]
    dt = m.get(2, [P.DT_FLOAT])[0]
    name = m.get(8, [b""])[0].decode()
    npdt = {P.DT_FLOAT: np.float32, P.DT_INT64: np.int64,
            P.DT_INT32: np.int32}[dt]
    if 9 in m:
        arr = np.frombuffer(m[9][0], dtype=npdt).reshape(dims).copy()
    elif dt == P.DT_FLOAT and 4 in m:
        arr = np.array([struct.unpack("<f", v)[0] for v in m[4]],
                       dtype=np.float32).reshape(dims)
    else:
        arr = np.zeros(dims, dtype=npdt)
    return name, torch.from_numpy(arr)


def _parse_value_info(buf: bytes) -> Tuple[str, torch.dtype, List[int]]:
    m = P.parse_message(buf)
    name = m[1][0].decode()
    tt = P.parse_message(P.parse_message(m[2][0])[1][0])
    elem = tt.get(1, [P.DT_FLOAT])[0]
    shape = []
    if 2 in tt:
        for dim in P.parse_message(tt[2][0]).get(1, []):
            dm = P.parse_message(dim)
            shape.append(_signed(dm.get(1, [0])[0]))
    return name, _DT2TORCH[elem], shape


def import_onnx(blob) -> Tuple[object, Dict[str, object], List[object]]:
    """bytes or path -> (graph, {input_name: placeholder}, [outputs])."""
    from ..graph.graph import DefineAndRunGraph, pop_graph, push_graph
    from ..graph.ops import api as ht
    if isinstance(blob, str):
        with open(blob, "rb") as fh:
            blob = fh.read()
    model = P.parse_message(blob)
    gm = P.parse_message(model[7][0])
    g = DefineAndRunGraph(gm.get(2, [b"onnx"])[0].decode())
    push_graph(g)
    try:
        env: Dict[str, object] = {}
        const: Dict[str, torch.Tensor] = {}
        for t in gm.get(5, []):
            name, data = _parse_tensor(t)
            const[name] = data
        inputs: Dict[str, object] = {}
        for vi in gm.get(11, []):
            name, dt, shape = _parse_value_info(vi)
            if name in const:
                continue
            ph = ht.placeholder(tuple(shape), dtype=dt, name=name)
            env[name] = ph
            inputs[name] = ph

        def get(name):
            if name in env:
                return env[name]
            data = const[name]
            v = ht.variable(data, name=name,
                            requires_grad=data.dtype.is_floating_point)
            env[name] = v
            return v

        for nd in gm.get(1, []):
            m = P.parse_message(nd)
            ins = [x.decode() for x in m.get(1, [])]
            outs = [x.decode() for x in m.get(2, [])]
            op = m[4][0].decode()
            attrs = dict(_parse_attr(a) for a in m.get(5, []))
            if op == "MatMul":
                y = ht.matmul(get(ins[0]), get(ins[1]))
            elif op == "Gemm":
                x, w = get(ins[0]), get(ins[1])
                ta = bool(attrs.get("transA", 0))
                tb = bool(attrs.get("transB", 0))
                y = ht.matmul(x, w, trans_a=ta, trans_b=tb)
                if len(ins) > 2:
                    y = ht.add(y, get(ins[2]))
            elif op in ("Add", "Sub", "Mul", "Div"):
                f = {"Add": ht.add, "Sub": ht.sub, "Mul": ht.mul,
                     "Div": ht.div}[op]
                y = f(get(ins[0]), get(ins[1]))
            elif op in ("Relu", "Gelu", "Sigmoid", "Tanh", "Exp", "Log",
                        "Sqrt", "Neg"):
                y = getattr(ht, op.lower())(get(ins[0]))
            elif op == "Abs":
                y = ht.abs_(get(ins[0]))
            elif op == "Pow" and ins[1] in const \
                    and const[ins[1]].numel() == 1:
                y = ht.pow(get(ins[0]), float(const[ins[1]]))
            elif op == "Slice":
                starts = const[ins[1]].tolist()
                ends = const[ins[2]].tolist()
                axes = (const[ins[3]].tolist() if len(ins) > 3
                        else list(range(len(starts))))
                y = get(ins[0])
                for s, e, ax in zip(starts, ends, axes):
                    y = ht.slice_(y, int(ax), int(s), int(e) - int(s))
            elif op == "Where":
                y = ht.where(get(ins[0]), get(ins[1]), get(ins[2]))
            elif op == "Concat":
                y = ht.concat([get(i) for i in ins],
                              dim=int(attrs.get("axis", 0)))
            elif op == "Softmax":
                y = ht.softmax(get(ins[0]), dim=attrs.get("axis", -1))
            elif op == "Reshape":
                shape = const[ins[1]].tolist()
                y = ht.reshape(get(ins[0]), tuple(int(s) for s in shape))
            elif op == "Transpose":
                perm = attrs["perm"]
                swaps = [i for i, p in enumerate(perm) if p != i]
                if not swaps:
                    y = get(ins[0])
                elif len(swaps) == 2:
                    y = ht.transpose(get(ins[0]), swaps[0], swaps[1])
                else:
                    raise NotImplementedError(f"perm {perm}")
            elif op == "LayerNormalization":
                y = ht.layer_norm(get(ins[0]), get(ins[1]), get(ins[2]),
                                  eps=attrs.get("epsilon", 1e-5))
            elif op == "Gather":
                assert attrs.get("axis", 0) == 0
                y = ht.embedding(get(ins[0]), get(ins[1]))
            elif op == "Cast":
                y = ht.cast(get(ins[0]), _DT2TORCH[attrs["to"]])
            elif op in ("ReduceSum", "ReduceMean", "ReduceMax",
                        "ReduceMin", "ReduceProd"):
                f = {"ReduceSum": ht.reduce_sum,
                     "ReduceMean": ht.reduce_mean,
                     "ReduceMax": ht.reduce_max,
                     "ReduceMin": ht.reduce_min,
                     "ReduceProd": ht.reduce_prod}[op]
                axes = attrs.get("axes")
                dim = (None if axes is None
                       else axes[0] if len(axes) == 1 else list(axes))
                y = f(get(ins[0]), dim=dim,
                      keepdim=bool(attrs.get("keepdims", 0)))
            else:
                raise NotImplementedError(f"ONNX import: op {op}")
            env[outs[0]] = y
        outputs = [env[_parse_value_info(vi)[0]] for vi in gm.get(12, [])]
    finally:
        pop_graph()
    return g, inputs, outputs
