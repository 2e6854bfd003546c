"""Single-device nn layers over the graph API.

Reference parity: python/hetu/nn/modules/ (Linear/Embedding/Norm/Dropout/
activations/losses).  Each layer creates graph `variable` parameters at
construction and emits ops in forward().
"""
from __future__ import annotations

from typing import Optional

import torch

from ..graph.ops import api as ht
from . import init
from .module import Module


class Linear(Module):
    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 dtype=torch.float32, name: str = "linear"):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        w = init.xavier_normal((out_features, in_features), dtype=dtype,
                               name=f"{name}.weight")
        self.weight = ht.variable(w, name=f"{name}.weight")
        if bias:
            self.bias = ht.variable(init.zeros((out_features,), dtype),
                                    name=f"{name}.bias")
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        return ht.linear(x, self.weight, self.bias)


class Embedding(Module):
    def __init__(self, num_embeddings: int, embedding_dim: int,
                 dtype=torch.float32, name: str = "embedding"):
        super().__init__()
        w = init.normal((num_embeddings, embedding_dim), std=0.02,
                        dtype=dtype, name=f"{name}.weight")
        self.weight = ht.variable(w, name=f"{name}.weight")

    def forward(self, ids):
        return ht.embedding(self.weight, ids)


class LayerNorm(Module):
    def __init__(self, dim: int, eps: float = 1e-5, dtype=torch.float32,
                 name: str = "ln"):
        super().__init__()
        self.eps = eps
        self.weight = ht.variable(init.ones((dim,), dtype),
                                  name=f"{name}.weight")
        self.bias = ht.variable(init.zeros((dim,), dtype),
                                name=f"{name}.bias")

    def forward(self, x):
        return ht.layer_norm(x, self.weight, self.bias, self.eps)


class RMSNorm(Module):
    def __init__(self, dim: int, eps: float = 1e-6, dtype=torch.float32,
                 name: str = "rms"):
        super().__init__()
        self.eps = eps
        self.weight = ht.variable(init.ones((dim,), dtype),
                                  name=f"{name}.weight")

    def forward(self, x):
        return ht.rms_norm(x, self.weight, self.eps)


class Dropout(Module):
    def __init__(self, p: float = 0.1):
        super().__init__()
        self.p = p

    def forward(self, x):
        if self.p <= 0.0 or not self.training:
            return x
        return ht.dropout(x, self.p)


class GELU(Module):
    def forward(self, x):
        return ht.gelu(x)


class SiLU(Module):
    def forward(self, x):
        return ht.silu(x)


class ReLU(Module):
    def forward(self, x):
        return ht.relu(x)


class Tanh(Module):
    def forward(self, x):
        return ht.tanh(x)


class Sigmoid(Module):
    def forward(self, x):
        return ht.sigmoid(x)
