"""Minimal ONNX protobuf wire-format encoder/decoder (no onnx package).

Implements exactly the message subset the exporter/importer needs
(ModelProto / GraphProto / NodeProto / TensorProto / ValueInfoProto /
AttributeProto) directly at the protobuf wire level, so exported files are
readable by standard ONNX tooling and standard .onnx files with this op
subset import cleanly.  Reference parity target: hetu/v1/python/hetu/onnx.
"""
from __future__ import annotations

import struct
from typing import Dict, List, Tuple


# ---------------------------------------------------------------------------
# wire primitives
# ---------------------------------------------------------------------------
def _varint(v: int) -> bytes:
    out = bytearray()
    v &= (1 << 64) - 1
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _tag(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def f_varint(field: int, v: int) -> bytes:
    return _tag(field, 0) + _varint(v)


def f_bytes(field: int, v: bytes) -> bytes:
    return _tag(field, 2) + _varint(len(v)) + v


def f_str(field: int, v: str) -> bytes:
    return f_bytes(field, v.encode())


def read_varint(buf: bytes, i: int) -> Tuple[int, int]:
    v = 0
    shift = 0
    while True:
        b = buf[i]
        i += 1
        v |= (b & 0x7F) << shift
        if not b & 0x80:
            return v, i
        shift += 7


def parse_message(buf: bytes) -> Dict[int, List]:
    """Decode one message level: {field: [values]} where a value is int
    (varint), bytes (length-delimited) or raw 8/4-byte chunks."""
    out: Dict[int, List] = {}
    i = 0
    n = len(buf)
    while i < n:
        key, i = read_varint(buf, i)
        field, wire = key >> 3, key & 7
        if wire == 0:
            v, i = read_varint(buf, i)
        elif wire == 2:
            ln, i = read_varint(buf, i)
            v = buf[i:i + ln]
            i += ln
        elif wire == 5:
            v = buf[i:i + 4]
            i += 4
        elif wire == 1:
            v = buf[i:i + 8]
            i += 8
        else:
            raise ValueError(f"unsupported wire type {wire}")
        out.setdefault(field, []).append(v)
    return out


# ---------------------------------------------------------------------------
# ONNX message builders
# ---------------------------------------------------------------------------
DT_FLOAT, DT_INT64, DT_INT32, DT_BOOL = 1, 7, 6, 9
ATTR_FLOAT, ATTR_INT, ATTR_STRING, ATTR_TENSOR = 1, 2, 3, 4
ATTR_FLOATS, ATTR_INTS = 6, 7


def tensor_proto(name: str, dims: List[int], data_type: int,
                 raw: bytes) -> bytes:
    out = b""
    for d in dims:
        out += f_varint(1, d)
    out += f_varint(2, data_type)
    out += f_str(8, name)
    out += f_bytes(9, raw)
    return out


def attr(name: str, value) -> bytes:
    out = f_str(1, name)
    if isinstance(value, float):
        out += _tag(2, 5) + struct.pack("<f", value) + f_varint(20, ATTR_FLOAT)
    elif isinstance(value, bool) or isinstance(value, int):
        out += f_varint(3, int(value)) + f_varint(20, ATTR_INT)
    elif isinstance(value, str):
        out += f_str(4, value) + f_varint(20, ATTR_STRING)
    elif isinstance(value, (list, tuple)):
        if value and isinstance(value[0], float):
            for v in value:
                out += _tag(7, 5) + struct.pack("<f", v)
            out += f_varint(20, ATTR_FLOATS)
        else:
            for v in value:
                out += f_varint(8, int(v))
            out += f_varint(20, ATTR_INTS)
    else:
        raise TypeError(f"attr {name}: {type(value)}")
    return out


def node_proto(op_type: str, inputs: List[str], outputs: List[str],
               name: str = "", attrs: Dict = None) -> bytes:
    out = b""
    for x in inputs:
        out += f_str(1, x)
    for x in outputs:
        out += f_str(2, x)
    if name:
        out += f_str(3, name)
    out += f_str(4, op_type)
    for k, v in (attrs or {}).items():
        out += f_bytes(5, attr(k, v))
    return out


def value_info(name: str, elem_type: int, shape: List[int]) -> bytes:
    dims = b""
    for d in shape:
        dims += f_bytes(1, f_varint(1, d))          # Dimension.dim_value
    tshape = f_bytes(2, dims)                        # Tensor.shape
    ttype = f_varint(1, elem_type) + tshape          # Tensor.elem_type
    tp = f_bytes(1, ttype)                           # TypeProto.tensor_type
    return f_str(1, name) + f_bytes(2, tp)


def graph_proto(nodes: List[bytes], name: str, initializers: List[bytes],
                inputs: List[bytes], outputs: List[bytes]) -> bytes:
    out = b""
    for nd in nodes:
        out += f_bytes(1, nd)
    out += f_str(2, name)
    for t in initializers:
        out += f_bytes(5, t)
    for vi in inputs:
        out += f_bytes(11, vi)
    for vi in outputs:
        out += f_bytes(12, vi)
    return out


def model_proto(graph: bytes, opset: int = 17) -> bytes:
    opset_msg = f_str(1, "") + f_varint(2, opset)
    return (f_varint(1, 8)                 # ir_version
            + f_str(2, "hetu_amd")         # producer
            + f_bytes(7, graph)
            + f_bytes(8, opset_msg))
