"""ONNX export/import for hetu_amd graphs (reference v1/python/hetu/onnx).

Self-contained: the protobuf wire format is encoded/decoded directly
(hetu_amd/onnx/proto.py) so no onnx package is required; files interop
with standard ONNX tooling for the supported op subset.
"""
from .convert import export_onnx, import_onnx

__all__ = ["export_onnx", "import_onnx"]
