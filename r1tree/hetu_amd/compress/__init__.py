"""Embedding memory compression methods.

Re-creation of the reference's EmbeddingMemoryCompression toolkit
(/root/reference/tools/EmbeddingMemoryCompression/methods — VLDB'24
survey implementation: hash/QR, tensor-train, low-rank, quantization,
deep-hash and friends over a common `Embedding` interface).  Each method
here is a drop-in replacement for a [num, dim] embedding with a
`memory_bytes()` report; on GPU the quantized variant uses the blockwise
int8/fp4/nf4 kernels in hetu_amd/ops/hip/quant.hip.
"""
from .embed import (CompressedEmbedding, DeepHashEmbedding, HashEmbedding,
                    LowRankEmbedding, QuantizedEmbedding, TTEmbedding,
                    make_compressed_embedding)

__all__ = ["CompressedEmbedding", "HashEmbedding", "TTEmbedding",
           "LowRankEmbedding", "QuantizedEmbedding", "DeepHashEmbedding",
           "make_compressed_embedding"]
