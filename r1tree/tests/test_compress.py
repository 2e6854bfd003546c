"""Embedding compression methods (reference tools/EmbeddingMemoryCompression)."""
import pytest
import torch

from hetu_amd.compress import (DeepHashEmbedding, HashEmbedding,
                               LowRankEmbedding, QuantizedEmbedding,
                               TTEmbedding, make_compressed_embedding)


@pytest.mark.parametrize("method,kw", [
    ("hash", {}), ("tt", {"rank": 8}), ("lowrank", {"rank": 8}),
    ("dhe", {"k": 32, "hidden": 64}),
])
def test_trainable_methods_compress_and_learn(method, kw):
    torch.manual_seed(0)
    num, dim = 1000, 16
    emb = make_compressed_embedding(method, num, dim, **kw)
    assert emb.compression_ratio() > 1.5, emb.memory_bytes()
    ids = torch.randint(0, num, (64,))
    tgt = torch.randn(64, dim)
    opt = torch.optim.Adam(emb.parameters(), lr=1e-2)
    first = last = None
    for _ in range(60):
        loss = ((emb(ids) - tgt) ** 2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        first = first if first is not None else loss.item()
        last = loss.item()
    assert last < 0.7 * first, (first, last)
    # arbitrary-shape ids
    assert emb(torch.randint(0, num, (3, 5))).shape == (3, 5, dim)


@pytest.mark.parametrize("qtype,tol", [("int8", 0.02), ("nf4", 0.2),
                                       ("fp4", 0.4)])
def test_quantized_embedding_roundtrip(qtype, tol):
    torch.manual_seed(1)
    w = torch.randn(64, 32)
    emb = QuantizedEmbedding(w, qtype=qtype, blocksize=64)
    ids = torch.arange(64)
    err = (emb(ids) - w).abs().max().item()
    scale = w.abs().max().item()
    assert err < tol * scale, err
    assert emb.memory_bytes() < w.numel() * 4


def test_hash_collision_free_within_tables():
    emb = HashEmbedding(1000, 8)
    ids = torch.arange(1000)
    pairs = torch.stack([ids // emb.q, ids % emb.q], 1)
    assert torch.unique(pairs, dim=0).shape[0] == 1000  # (q, r) is unique
