"""Symbolic integers for dynamic shapes (sequence lengths, micro-batch sizes).

MI355X-native analogue of the reference's IntSymbol/SyShape
(/root/reference/hetu/core/symbol.h:1-178): a symbol is a named mutable
integer; shapes may mix ints and symbols and are resolved against the
currently-set symbol values at execution time.
"""
from __future__ import annotations


class IntSymbol:
    __slots__ = ("name", "_value")

    def __init__(self, value: int | None = None, name: str = "sym"):
        self.name = name
        self._value = value

    @property
    def value(self) -> int:
        if self._value is None:
            raise RuntimeError(f"IntSymbol {self.name} has no value set")
        return self._value

    def set(self, value: int) -> None:
        self._value = int(value)

    def is_set(self) -> bool:
        return self._value is not None

    def __int__(self) -> int:
        return self.value

    def __repr__(self):
        return f"IntSymbol({self.name}={self._value})"


def resolve_dim(d):
    """Resolve a shape dimension that may be an int or an IntSymbol."""
    if isinstance(d, IntSymbol):
        return d.value
    return int(d)


def resolve_shape(shape):
    return tuple(resolve_dim(d) for d in shape)


def shape_has_symbol(shape) -> bool:
    return any(isinstance(d, IntSymbol) for d in shape)
