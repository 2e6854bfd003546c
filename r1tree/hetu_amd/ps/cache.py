"""HET-style cached embedding: GPU cache in front of the sharded table.

Re-creation of the reference's hetu_cache client
(/root/reference/hetu/v1/src/hetu_cache/include/cache.h — embed lookup
through an LRU/LFU/LFUOpt cache with versioned bounded-staleness sync,
HET VLDB'22).  Row storage is a [capacity, dim] tensor on the compute
device so a cached lookup is one GPU gather; the id->slot index lives in
C++ (hetu_amd/ops/hip/embed_cache.cpp).

Write policy: gradients update the cached rows immediately (local SGD on
the cache copy) and are ACCUMULATED into a pending buffer that is flushed
to the owner shard (`table.push`) every `staleness+1` steps — staleness 0
is write-through and exactly matches uncached training on one worker.
Rows evicted while dirty flush first; rows whose cached copy is older than
the staleness bound are re-pulled after each flush (versioned sync).
"""
from __future__ import annotations

from typing import Dict, Optional

import torch

from .table import ShardedEmbeddingTable


def _cache_index(capacity: int, policy: str):
    from ..ops.functional import ext
    return ext().EmbedCache(capacity, policy)


class _CachedLookup(torch.autograd.Function):
    @staticmethod
    def forward(ctx, rows_flat, uniq, inv, module, hook):
        # `hook` is a dummy requires-grad leaf: rows_flat itself is a plain
        # buffer, and autograd drops Functions with no differentiable input
        ctx.save_for_backward(uniq, inv)
        ctx.module = module
        return rows_flat[inv]

    @staticmethod
    def backward(ctx, grad_out):
        uniq, inv = ctx.saved_tensors
        g = torch.zeros(uniq.numel(), grad_out.shape[-1],
                        dtype=grad_out.dtype, device=grad_out.device)
        g.index_add_(0, inv.reshape(-1),
                     grad_out.reshape(-1, grad_out.shape[-1]))
        ctx.module._apply_grads(uniq, g)
        return None, None, None, None, None


class CachedEmbedding:
    def __init__(self, table: ShardedEmbeddingTable, capacity: int,
                 policy: str = "lru", staleness: int = 0,
                 device: Optional[torch.device] = None,
                 dtype: Optional[torch.dtype] = None):
        self.table = table
        self.capacity = capacity
        self.device = device or table.device
        self.dtype = dtype or table.dtype
        self.index = _cache_index(capacity, policy)
        self.storage = torch.zeros(capacity, table.dim, device=self.device,
                                   dtype=self.dtype)
        self.staleness = staleness
        self._pending: Dict[int, torch.Tensor] = {}
        self._hook = torch.zeros(1, requires_grad=True)
        self._step = 0

    # ---- forward ---------------------------------------------------------
    def __call__(self, ids: torch.Tensor) -> torch.Tensor:
        shape = ids.shape
        uniq, inv = torch.unique(ids.reshape(-1).long(),
                                 return_inverse=True)
        slots, hit = self.index.query(uniq.cpu())
        n_miss = int((~hit).sum())
        if n_miss:
            miss_ids = uniq.cpu()[~hit]
            rows = self.table.pull(miss_ids).to(self.device, self.dtype)
            mslots, ev_ids, ev_slots = self.index.admit(miss_ids)
            if ev_ids.numel():
                self._flush_ids(ev_ids)
            admitted = mslots >= 0
            if admitted.any():
                self.storage[mslots[admitted].to(self.device)] = \
                    rows[admitted.to(self.device)]
            slots[~hit] = mslots
            # refused ids (LFUOpt admission): serve from the pulled rows
            flat = self.storage[slots.clamp(min=0).to(self.device)]
            if (~admitted).any():
                pos = torch.nonzero(~hit).reshape(-1)[~admitted]
                flat[pos.to(self.device)] = rows[(~admitted).to(self.device)]
        else:
            flat = self.storage[slots.to(self.device)]
        self._uniq_slots = slots  # for gradient write-back
        out = _CachedLookup.apply(flat, uniq, inv, self, self._hook)
        return out.reshape(*shape, self.table.dim)

    # ---- gradient path ---------------------------------------------------
    def _apply_grads(self, uniq: torch.Tensor, g: torch.Tensor):
        """Update cached copies now; stage grads for the owner shard."""
        slots = self._uniq_slots.to(self.device)
        cached = slots >= 0
        if cached.any():
            self.storage.index_add_(0, slots[cached],
                                    g[cached].to(self.dtype),
                                    alpha=-self.table.lr)
        gc = g.cpu()
        for i, ident in enumerate(uniq.cpu().tolist()):
            if ident in self._pending:
                self._pending[ident] += gc[i]
            else:
                self._pending[ident] = gc[i].clone()
        self._step += 1
        if self._step % (self.staleness + 1) == 0:
            self.flush()

    def _flush_ids(self, ids: torch.Tensor):
        sel = [i for i in ids.tolist() if i in self._pending]
        if not sel:
            return
        g = torch.stack([self._pending.pop(i) for i in sel])
        self.table.push(torch.tensor(sel, dtype=torch.int64), g)

    def flush(self):
        """Push all pending grads; re-pull rows beyond the staleness bound."""
        if self._pending:
            ids = torch.tensor(list(self._pending.keys()), dtype=torch.int64)
            g = torch.stack([self._pending[i] for i in ids.tolist()])
            self._pending.clear()
            self.table.push(ids, g)
        self.index.bump_version()
        sids, sslots = self.index.stale(self.staleness)
        if sids.numel():
            rows = self.table.pull(sids).to(self.device, self.dtype)
            self.storage[sslots.to(self.device)] = rows
            self.index.refresh(sslots)

    @property
    def hit_rate(self) -> float:
        return self.index.hit_rate()
