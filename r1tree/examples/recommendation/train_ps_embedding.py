#!/usr/bin/env python3
"""Recommendation-style training with the parameter server + HET cache
(reference: hetu/v1 PS examples — wide&deep style sparse embeddings).

Each rank owns a shard of the embedding table (id % world) and looks rows
up through an LRU GPU cache with bounded staleness; dense layers train
normally with torch.

Run: python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
       --master-addr 127.0.0.1 examples/recommendation/train_ps_embedding.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.parallel.comm import comm_backend  # noqa: E402
from hetu_amd.ps import CachedEmbedding, ShardedEmbeddingTable  # noqa: E402

NUM_IDS, DIM, FIELDS = 100_000, 16, 8


def main():
    comm = comm_backend()
    device = comm.device
    torch.manual_seed(11 + comm.rank)
    table = ShardedEmbeddingTable(NUM_IDS, DIM, comm=comm, lr=0.05)
    emb = CachedEmbedding(table, capacity=8192, policy="lfu", staleness=4,
                          device=device)
    mlp = torch.nn.Sequential(
        torch.nn.Linear(FIELDS * DIM, 64), torch.nn.ReLU(),
        torch.nn.Linear(64, 1)).to(device)
    opt = torch.optim.Adam(mlp.parameters(), lr=1e-3)
    for step in range(50):
        ids = torch.randint(0, NUM_IDS, (256, FIELDS), device=device)
        y = torch.rand(256, 1, device=device)
        e = emb(ids).reshape(256, -1)
        pred = mlp(e)
        loss = torch.nn.functional.binary_cross_entropy_with_logits(pred, y)
        opt.zero_grad()
        loss.backward()        # pushes sparse grads through the cache
        opt.step()
        if comm.rank == 0 and step % 10 == 0:
            print(f"step {step} loss {loss.item():.4f} "
                  f"cache_hit {emb.hit_rate:.2%}")
    emb.flush()


if __name__ == "__main__":
    main()
