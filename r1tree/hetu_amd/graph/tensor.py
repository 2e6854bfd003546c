"""Graph tensor (edge) — symbolic handle produced/consumed by ops.

Mirrors the role of the reference's Tensor/TensorDef
(/root/reference/hetu/graph/tensor.h:21-583) but is a thin Python object:
actual storage is a torch.Tensor owned by the executing graph (PyTorch-ROCm's
caching allocator replaces the reference's CUDACachingMemoryPool).
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch

from ..core.symbol import IntSymbol, resolve_shape, shape_has_symbol
from ..parallel.dstates import DistributedStates


class TensorMeta:
    __slots__ = ("shape", "dtype")

    def __init__(self, shape: Sequence, dtype: torch.dtype):
        self.shape = tuple(shape)
        self.dtype = dtype

    @property
    def ndim(self):
        return len(self.shape)

    def resolved_shape(self):
        return resolve_shape(self.shape)

    def has_symbol(self):
        return shape_has_symbol(self.shape)

    def __repr__(self):
        return f"TensorMeta({list(self.shape)}, {self.dtype})"


class Tensor:
    """Edge in the dataflow graph."""
    _next_id = 0

    __slots__ = ("id", "producer", "output_index", "meta", "name", "graph",
                 "ds", "device_group", "shard_sections", "is_parameter", "requires_grad",
                 "_data")

    def __init__(self, producer, output_index: int, meta: TensorMeta,
                 name: str = "", graph=None,
                 ds: Optional[DistributedStates] = None,
                 device_group=None, requires_grad: bool = False):
        self.id = Tensor._next_id
        Tensor._next_id += 1
        self.producer = producer
        self.output_index = output_index
        self.meta = meta
        self.name = name or (f"{producer.name}:{output_index}" if producer else f"t{self.id}")
        self.graph = graph
        self.ds = ds
        self.device_group = device_group
        self.is_parameter = False
        self.requires_grad = requires_grad
        self._data: Optional[torch.Tensor] = None  # eager value / param storage

    # ---- meta ------------------------------------------------------------
    @property
    def shape(self):
        return self.meta.shape

    @property
    def dtype(self):
        return self.meta.dtype

    @property
    def ndim(self):
        return self.meta.ndim

    def global_shape(self):
        if self.ds is None:
            return self.meta.resolved_shape()
        return self.ds.global_shape(self.meta.resolved_shape())

    # ---- data (eager / parameter storage) --------------------------------
    def get_data(self) -> Optional[torch.Tensor]:
        return self._data

    def set_data(self, value: torch.Tensor):
        self._data = value

    def numpy(self):
        if self._data is None:
            raise RuntimeError(f"tensor {self.name} has no materialized data")
        return self._data.detach().cpu().numpy()

    def item(self):
        if self._data is None:
            raise RuntimeError(f"tensor {self.name} has no materialized data")
        return self._data.item()

    # ---- operator sugar (builds graph ops) -------------------------------
    def _f(self):
        from .ops import api as F
        return F

    def __add__(self, other):
        return self._f().add(self, other)

    __radd__ = __add__

    def __sub__(self, other):
        return self._f().sub(self, other)

    def __mul__(self, other):
        return self._f().mul(self, other)

    __rmul__ = __mul__

    def __truediv__(self, other):
        return self._f().div(self, other)

    def __neg__(self):
        return self._f().neg(self)

    def __matmul__(self, other):
        return self._f().matmul(self, other)

    def reshape(self, shape):
        return self._f().reshape(self, shape)

    def transpose(self, d0, d1):
        return self._f().transpose(self, d0, d1)

    def sum(self, dim=None, keepdim=False):
        return self._f().reduce_sum(self, dim, keepdim)

    def mean(self, dim=None, keepdim=False):
        return self._f().reduce_mean(self, dim, keepdim)

    def __repr__(self):
        return (f"Tensor({self.name}, shape={list(self.shape)}, "
                f"dtype={self.dtype}, ds={self.ds})")
