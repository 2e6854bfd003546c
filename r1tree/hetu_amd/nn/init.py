"""Initializers (reference hetu/graph/init/initializer.h: constant/uniform/
normal/xavier/he/lecun).  Deterministic per-parameter: the seed is derived
from a global seed and the parameter name, so every rank materializes the
same global weight and slices its own shard."""
from __future__ import annotations

import hashlib
import math

import torch

_GLOBAL_SEED = [1234]


def set_seed(seed: int):
    _GLOBAL_SEED[0] = seed


def _gen(name: str) -> torch.Generator:
    h = hashlib.sha256(f"{_GLOBAL_SEED[0]}:{name}".encode()).digest()
    g = torch.Generator()
    g.manual_seed(int.from_bytes(h[:8], "little") & 0x7FFFFFFFFFFFFFFF)
    return g


def normal(shape, std=0.02, mean=0.0, dtype=torch.float32, name="w"):
    return (torch.randn(*shape, generator=_gen(name)) * std + mean).to(dtype)


def uniform(shape, a=-0.1, b=0.1, dtype=torch.float32, name="w"):
    return (torch.rand(*shape, generator=_gen(name)) * (b - a) + a).to(dtype)


def constant(shape, value=0.0, dtype=torch.float32):
    return torch.full(shape, float(value)).to(dtype)


def zeros(shape, dtype=torch.float32):
    return torch.zeros(*shape, dtype=dtype)


def ones(shape, dtype=torch.float32):
    return torch.ones(*shape, dtype=dtype)


def xavier_uniform(shape, gain=1.0, dtype=torch.float32, name="w"):
    fan_out, fan_in = shape[0], shape[-1]
    a = gain * math.sqrt(6.0 / (fan_in + fan_out))
    return uniform(shape, -a, a, dtype, name)


def xavier_normal(shape, gain=1.0, dtype=torch.float32, name="w"):
    fan_out, fan_in = shape[0], shape[-1]
    std = gain * math.sqrt(2.0 / (fan_in + fan_out))
    return normal(shape, std, 0.0, dtype, name)


def he_normal(shape, dtype=torch.float32, name="w"):
    fan_in = shape[-1]
    return normal(shape, math.sqrt(2.0 / fan_in), 0.0, dtype, name)


def lecun_normal(shape, dtype=torch.float32, name="w"):
    fan_in = shape[-1]
    return normal(shape, math.sqrt(1.0 / fan_in), 0.0, dtype, name)
