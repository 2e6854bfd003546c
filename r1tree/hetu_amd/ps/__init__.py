"""Parameter-server subsystem: sharded embedding tables + GPU caches.

MI355X-native re-design of the reference's v1 PS stack
(/root/reference/hetu/v1/ps-lite — zmq/ibverbs vans, PSFunc pull/push —
and /root/reference/hetu/v1/src/hetu_cache — HET VLDB'22 LRU/LFU/LFUOpt
embedding caches): instead of dedicated server processes over a zmq
transport, every worker OWNS a shard of each table (id % world) and
pull/push are all-to-all exchanges over RCCL/xGMI (gloo on CPU) —
the xGMI mesh makes every worker equidistant, so dedicated servers would
only add a hop.  The HET client cache (bounded-staleness versioned sync)
sits in front of the sharded table with the index in C++
(hetu_amd/ops/hip/embed_cache.cpp).
"""
from .table import ShardedEmbeddingTable
from .cache import CachedEmbedding

__all__ = ["ShardedEmbeddingTable", "CachedEmbedding"]
