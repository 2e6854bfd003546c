"""Compressed embedding methods (hash/QR, TT, low-rank, quantized, DHE).

Reference behaviors re-created (not copied) from
/root/reference/tools/EmbeddingMemoryCompression/methods/:
  - `compo.py`   (compositional / QR trick)  -> HashEmbedding
  - `tensortrain.py`                          -> TTEmbedding
  - `mde.py` (mixed-dimension / low-rank)     -> LowRankEmbedding
  - `quantize.py` (post-training blockwise)   -> QuantizedEmbedding
  - `dhe.py` (deep hash embeddings)           -> DeepHashEmbedding

All are torch modules usable standalone or as the storage behind
hetu_amd.ps.CachedEmbedding's table (frozen quantized serving path).
"""
from __future__ import annotations

import math
from typing import Optional

import torch
from torch import nn


class CompressedEmbedding(nn.Module):
    num: int
    dim: int

    def memory_bytes(self) -> int:
        return sum(p.numel() * p.element_size() for p in self.parameters()) \
            + sum(b.numel() * b.element_size() for b in self.buffers())

    def compression_ratio(self) -> float:
        return (self.num * self.dim * 4) / max(self.memory_bytes(), 1)


class HashEmbedding(CompressedEmbedding):
    """QR / compositional embedding: id -> quotient and remainder rows from
    two small tables, combined by `op` (sum or mult)."""

    def __init__(self, num: int, dim: int, ratio: float = 0.125,
                 op: str = "sum"):
        super().__init__()
        self.num, self.dim = num, dim
        q = max(2, int(math.sqrt(num * ratio)))
        self.q = math.ceil(num / q)
        self.quotient = nn.Embedding(math.ceil(num / self.q), dim)
        self.remainder = nn.Embedding(self.q, dim)
        self.op = op
        nn.init.normal_(self.quotient.weight, std=0.01)
        nn.init.normal_(self.remainder.weight, std=0.01)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        a = self.quotient(ids // self.q)
        b = self.remainder(ids % self.q)
        return a + b if self.op == "sum" else a * b


class TTEmbedding(CompressedEmbedding):
    """Tensor-train factorized table: id is unfolded into 3 indices over
    factor shapes (n1,n2,n3), the row is the TT contraction of 3 cores."""

    def __init__(self, num: int, dim: int, rank: int = 8):
        super().__init__()
        self.num, self.dim = num, dim
        n1 = max(2, round(num ** (1 / 3)))
        n2 = max(2, round(math.sqrt(num / n1)))
        n3 = math.ceil(num / (n1 * n2))
        self.shape = (n1, n2, n3)
        d1 = max(1, round(dim ** (1 / 3)))
        while dim % d1:
            d1 -= 1
        d2 = max(1, round(math.sqrt(dim // d1)))
        while (dim // d1) % d2:
            d2 -= 1
        d3 = dim // (d1 * d2)
        self.dims = (d1, d2, d3)
        r = rank
        self.g1 = nn.Parameter(torch.randn(n1, d1 * r) * 0.1)
        self.g2 = nn.Parameter(torch.randn(n2, r, d2 * r) * 0.1)
        self.g3 = nn.Parameter(torch.randn(n3, r, d3) * 0.1)
        self.rank = r

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        n1, n2, n3 = self.shape
        d1, d2, d3 = self.dims
        r = self.rank
        flat = ids.reshape(-1)
        i1 = flat // (n2 * n3)
        rem = flat % (n2 * n3)
        i2, i3 = rem // n3, rem % n3
        a = self.g1[i1].reshape(-1, d1, r)                 # [b, d1, r]
        b = self.g2[i2]                                    # [b, r, d2*r]
        c = self.g3[i3]                                    # [b, r, d3]
        ab = torch.bmm(a, b).reshape(-1, d1 * d2, r)       # [b, d1*d2, r]
        out = torch.bmm(ab, c)                             # [b, d1*d2, d3]
        return out.reshape(*ids.shape, self.dim)


class LowRankEmbedding(CompressedEmbedding):
    """Mixed-dimension / low-rank: E = A @ P with A [num, r], P [r, dim]."""

    def __init__(self, num: int, dim: int, rank: Optional[int] = None):
        super().__init__()
        self.num, self.dim = num, dim
        r = rank or max(1, dim // 8)
        self.a = nn.Embedding(num, r)
        self.proj = nn.Linear(r, dim, bias=False)
        nn.init.normal_(self.a.weight, std=0.01)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        return self.proj(self.a(ids))


class QuantizedEmbedding(CompressedEmbedding):
    """Post-training blockwise-quantized table (int8 / fp4 / nf4) for the
    frozen/serving path; rows are dequantized on lookup.  On GPU this uses
    the quant.hip kernels; on CPU a torch fallback of the same math."""

    def __init__(self, weight: torch.Tensor, qtype: str = "int8",
                 blocksize: int = 64):
        super().__init__()
        self.num, self.dim = weight.shape
        self.qtype, self.blocksize = qtype, blocksize
        from ..ops import functional as F
        if weight.is_cuda and F.has_ext():
            q, amax = F.ext().quantize_blockwise(
                weight.reshape(-1).contiguous(), qtype, blocksize)
        else:
            q, amax = self._quant_cpu(weight.reshape(-1), qtype, blocksize)
        self.register_buffer("q", q)
        self.register_buffer("absmax", amax)
        self.out_dtype = weight.dtype

    @staticmethod
    def _quant_cpu(x, qtype, bs):
        n = x.numel()
        pad = (-n) % bs
        xb = torch.cat([x.float(), x.new_zeros(pad).float()]).reshape(-1, bs)
        amax = xb.abs().amax(1)
        if qtype == "int8":
            qv = torch.round(xb / amax.clamp(min=1e-30)[:, None] * 127)
            q = (qv + 128).to(torch.uint8).reshape(-1)[:n]
            return q, amax
        code = _codebook(qtype)
        norm = xb / amax.clamp(min=1e-30)[:, None]
        idx = (norm.reshape(-1, 1) - code[None, :]).abs().argmin(1)
        idx = idx.reshape(-1)[:n].to(torch.uint8)
        q = (idx[0::2] << 4) | idx[1::2]
        return q, amax

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        from ..ops import functional as F
        if self.q.is_cuda and F.has_ext():
            w = F.ext().dequantize_blockwise(
                self.q, self.absmax, self.qtype, self.blocksize,
                self.num * self.dim, self.out_dtype)
        else:
            w = self._dequant_cpu()
        return w.reshape(self.num, self.dim)[ids]

    def _dequant_cpu(self):
        n = self.num * self.dim
        if self.qtype == "int8":
            v = self.q.float() - 128
            scale = self.absmax.repeat_interleave(self.blocksize)[:n] / 127
            return (v * scale).to(self.out_dtype)
        code = _codebook(self.qtype)
        hi, lo = self.q >> 4, self.q & 15
        idx = torch.stack([hi, lo], 1).reshape(-1)[:n].long()
        scale = self.absmax.repeat_interleave(self.blocksize)[:n]
        return (code[idx] * scale).to(self.out_dtype)


def _codebook(qtype):
    if qtype == "fp4":
        return torch.tensor([0.0, 0.0052083333, 0.6666667, 1.0, 0.3333333,
                             0.5, 0.1666667, 0.25, -0.0, -0.0052083333,
                             -0.6666667, -1.0, -0.3333333, -0.5, -0.1666667,
                             -0.25])
    return torch.tensor([-1.0, -0.6961928009986877, -0.5250730514526367,
                         -0.39491748809814453, -0.28444138169288635,
                         -0.18477343022823334, -0.09105003625154495, 0.0,
                         0.07958029955625534, 0.16093020141124725,
                         0.24611230194568634, 0.33791524171829224,
                         0.44070982933044434, 0.5626170039176941,
                         0.7229568362236023, 1.0])


class DeepHashEmbedding(CompressedEmbedding):
    """DHE: k universal hashes of the id -> dense feature vector -> MLP.
    No O(num) table at all."""

    def __init__(self, num: int, dim: int, k: int = 64, hidden: int = 128,
                 layers: int = 2, seed: int = 17):
        super().__init__()
        self.num, self.dim, self.k = num, dim, k
        g = torch.Generator().manual_seed(seed)
        prime = 2147483647
        self.register_buffer(
            "ha", torch.randint(1, prime, (k,), generator=g))
        self.register_buffer(
            "hb", torch.randint(0, prime, (k,), generator=g))
        self.prime = prime
        mods = []
        d_in = k
        for _ in range(layers - 1):
            mods += [nn.Linear(d_in, hidden), nn.SiLU()]
            d_in = hidden
        mods.append(nn.Linear(d_in, dim))
        self.mlp = nn.Sequential(*mods)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        f = (ids.reshape(-1, 1) * self.ha[None, :] + self.hb[None, :]) \
            % self.prime
        f = f.float() / self.prime * 2 - 1          # uniform [-1, 1)
        return self.mlp(f).reshape(*ids.shape, self.dim)


_METHODS = {"hash": HashEmbedding, "tt": TTEmbedding,
            "lowrank": LowRankEmbedding, "dhe": DeepHashEmbedding}


def make_compressed_embedding(method: str, num: int, dim: int, **kw
                              ) -> CompressedEmbedding:
    if method == "quantize":
        return QuantizedEmbedding(kw.pop("weight"), **kw)
    return _METHODS[method](num, dim, **kw)
