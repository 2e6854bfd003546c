"""LoRA adapters over the parallel linear layers.

Reference parity: python/hetu/peft/lora/layer.py:83-184 (column/row
parallel LoRA adapters: frozen base weight + trainable low-rank A/B pair,
scaling alpha/r, merge into the base weight for inference).
"""
from __future__ import annotations

from typing import Optional

import torch

from ..graph.ops import api as ht
from ..nn import init
from ..nn.module import Module
from ..nn.parallel import (ColumnParallelLinear, ParallelSpec,
                           RowParallelLinear, _shard)


class LoRALinear(Module):
    """y = base(x) + (alpha/r) * (x @ A^T) @ B^T.

    base: a ColumnParallelLinear or RowParallelLinear whose weight is
    FROZEN (removed from graph.parameters); A [r, in] follows the base's
    input sharding, B [out, r] the output sharding, so the adapter path
    needs no extra collectives beyond the base layer's own."""

    def __init__(self, base: Module, r: int = 8, alpha: float = 16.0,
                 name: str = "lora"):
        super().__init__()
        self.base = base
        self.r = r
        self.scaling = alpha / r
        spec: ParallelSpec = base.spec
        w = base.weight
        g = w.graph
        # freeze the base weight (and bias): LoRA trains adapters only
        for p in (base.weight, getattr(base, "bias", None)):
            if p is not None and p in g.parameters:
                g.parameters.remove(p)
                p.is_parameter = False
        out_f_local, in_f_local = tuple(w.shape)
        tp, ti = spec.tp, spec.my_tp_index()
        if isinstance(base, ColumnParallelLinear):
            # A replicated [r, in]; B col-sharded [out/tp, r]
            a = init.normal((r, in_f_local), std=0.01, name=f"{name}.A"
                            ).to(w.dtype)
            b = torch.zeros(out_f_local * tp, r)
            self.A = ht.variable(a, name=f"{name}.A",
                                 ds=spec.ds_weight_dup(),
                                 device_group=spec.device_group)
            self.B = ht.variable(_shard(b, 0, tp, ti).to(w.dtype),
                                 name=f"{name}.B", ds=spec.ds_weight_col(0),
                                 device_group=spec.device_group)
        elif isinstance(base, RowParallelLinear):
            # A row-sharded [r, in/tp]; B replicated [out, r]
            a = init.normal((r, in_f_local * tp), std=0.01,
                            name=f"{name}.A").to(w.dtype)
            self.A = ht.variable(_shard(a, 1, tp, ti),
                                 name=f"{name}.A", ds=spec.ds_weight_row(1),
                                 device_group=spec.device_group)
            self.B = ht.variable(torch.zeros(out_f_local, r, dtype=w.dtype),
                                 name=f"{name}.B", ds=spec.ds_weight_dup(),
                                 device_group=spec.device_group)
        else:
            raise TypeError("LoRALinear wraps Column/RowParallelLinear")

    def forward(self, x):
        y = self.base(x)
        h = ht.linear(x, self.A)                 # [.., r]
        d = ht.linear(h, self.B)                 # [.., out]
        return ht.add(y, ht.mul(d, self.scaling))

    @torch.no_grad()
    def merge(self):
        """Fold the adapter into the frozen base weight (inference)."""
        a = self.A.get_data().float()
        b = self.B.get_data().float()
        w = self.base.weight.get_data()
        w += (self.scaling * (b @ a)).to(w.dtype)

    @torch.no_grad()
    def unmerge(self):
        a = self.A.get_data().float()
        b = self.B.get_data().float()
        w = self.base.weight.get_data()
        w -= (self.scaling * (b @ a)).to(w.dtype)


class QLinear(torch.nn.Module):
    """4-bit-quantized frozen linear (QLoRA base layer): weight stored as
    packed nf4/fp4/int8 blocks, dequantized through the blockwise kernel
    into the GEMM (reference graph/ops/Quantization.h matmul4bit).  Pair
    with LoRALinear for QLoRA fine-tuning."""

    def __init__(self, weight: torch.Tensor, bias=None, qtype: str = "nf4",
                 blocksize: int = 64):
        super().__init__()
        from ..ops import functional as F
        self.out_features, self.in_features = weight.shape
        self.qtype, self.blocksize = qtype, blocksize
        q, amax = F.quantize_blockwise(
            weight.reshape(-1).contiguous(), qtype, blocksize)
        self.register_buffer("qweight", q)
        self.register_buffer("absmax", amax)
        if bias is not None:
            self.register_buffer("bias", bias.detach().clone())
        else:
            self.bias = None

    def forward(self, x):
        from ..ops import functional as F
        y = F.matmul_4bit(x, self.qweight, self.absmax, self.qtype,
                          self.blocksize,
                          (self.out_features, self.in_features))
        return y + self.bias if self.bias is not None else y
