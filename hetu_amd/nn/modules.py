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


class Conv2d(Module):
    """(reference nn/modules/conv.py)"""

    def __init__(self, in_channels: int, out_channels: int, kernel_size,
                 stride=1, padding=0, dilation=1, groups: int = 1,
                 bias: bool = True, dtype=torch.float32,
                 name: str = "conv"):
        super().__init__()
        ks = (kernel_size, kernel_size) if isinstance(kernel_size, int) \
            else tuple(kernel_size)
        w = init.he_normal((out_channels, in_channels // groups, *ks),
                           dtype=dtype, name=f"{name}.weight")
        self.weight = ht.variable(w, name=f"{name}.weight")
        if bias:
            self.bias = ht.variable(init.zeros((out_channels,), dtype),
                                    name=f"{name}.bias")
        else:
            self.register_parameter("bias", None)
        self.stride, self.padding = stride, padding
        self.dilation, self.groups = dilation, groups

    def forward(self, x):
        return ht.conv2d(x, self.weight, self.bias, self.stride,
                         self.padding, self.dilation, self.groups)


class MaxPool2d(Module):
    def __init__(self, kernel_size, stride=None, padding=0):
        super().__init__()
        self.k, self.s, self.p = kernel_size, stride, padding

    def forward(self, x):
        return ht.max_pool2d(x, self.k, self.s, self.p)


class AvgPool2d(Module):
    def __init__(self, kernel_size, stride=None, padding=0):
        super().__init__()
        self.k, self.s, self.p = kernel_size, stride, padding

    def forward(self, x):
        return ht.avg_pool2d(x, self.k, self.s, self.p)


class BatchNorm2d(Module):
    """(reference nn/modules/batchnorm.py)"""

    def __init__(self, num_features: int, eps: float = 1e-5,
                 dtype=torch.float32, name: str = "bn"):
        super().__init__()
        self.eps = eps
        self.weight = ht.variable(init.ones((num_features,), dtype),
                                  name=f"{name}.weight")
        self.bias = ht.variable(init.zeros((num_features,), dtype),
                                name=f"{name}.bias")

    def forward(self, x):
        return ht.batch_norm(x, self.weight, self.bias, self.eps)


class InstanceNorm2d(Module):
    def __init__(self, eps: float = 1e-5):
        super().__init__()
        self.eps = eps

    def forward(self, x):
        return ht.instance_norm(x, self.eps)


class ZeroPad2d(Module):
    """(reference nn/modules/padding.py ZeroPad2d)"""

    def __init__(self, padding):
        super().__init__()
        p = (padding,) * 4 if isinstance(padding, int) else tuple(padding)
        self.pad = p          # (left, right, top, bottom)

    def forward(self, x):
        l, r, t, b = self.pad
        # torch pad convention: last dim first -> (left, right, top, bottom)
        return ht.pad(x, [l, r, t, b], 0.0)


class MSELoss(Module):
    def forward(self, x, target):
        return ht.mse_loss(x, target)


class NLLLoss(Module):
    def __init__(self, reduction="mean", ignore_index=-100):
        super().__init__()
        self.reduction, self.ignore_index = reduction, ignore_index

    def forward(self, x, target):
        return ht.nll_loss(x, target, self.reduction, self.ignore_index)


class BCELoss(Module):
    def __init__(self, reduction="mean"):
        super().__init__()
        self.reduction = reduction

    def forward(self, x, target):
        return ht.binary_cross_entropy(x, target, self.reduction)


class KLDivLoss(Module):
    def __init__(self, reduction="batchmean"):
        super().__init__()
        self.reduction = reduction

    def forward(self, x, target):
        return ht.kl_div(x, target, self.reduction)


class CrossEntropyLoss(Module):
    """Dense soft-label or sparse integer-label CE by target dtype."""

    def __init__(self, reduction="mean"):
        super().__init__()
        self.reduction = reduction

    def forward(self, x, target):
        if target.dtype in (torch.int64, torch.int32):
            per_tok = ht.softmax_cross_entropy_sparse(x, target)
            return ht.reduce_mean(per_tok) if self.reduction == "mean" \
                else ht.reduce_sum(per_tok)
        return ht.softmax_cross_entropy(x, target, self.reduction)
