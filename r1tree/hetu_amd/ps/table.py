"""Sharded embedding table: serverless parameter server over all-to-all.

Replaces the reference's ps-lite pull/push PSFunc path
(/root/reference/hetu/v1/ps-lite/include/ps/kv_app.h — `Pull`/`Push` on a
server shard) with an owner-sharded design: row id is owned by rank
`id % world`; `pull(ids)` and `push(ids, grads)` route requests with ONE
all-to-all pair each (counts, then payload), which on MI355X is an RCCL
all-to-all over the xGMI mesh.  Server-side sparse optimizers (sgd /
adagrad, reference PSFhandle semantics) are applied by the owner on push.
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist

from ..parallel.comm import CommBackend, comm_backend


class ShardedEmbeddingTable:
    def __init__(self, num_embeddings: int, dim: int,
                 comm: Optional[CommBackend] = None,
                 device: Optional[torch.device] = None,
                 dtype: torch.dtype = torch.float32,
                 init_std: float = 0.01, optimizer: str = "sgd",
                 lr: float = 0.1, eps: float = 1e-10, seed: int = 0):
        self.comm = comm or comm_backend()
        self.num = num_embeddings
        self.dim = dim
        self.world = self.comm.world_size
        self.rank = self.comm.rank
        self.device = device or torch.device("cpu")
        self.dtype = dtype
        assert optimizer in ("sgd", "adagrad")
        self.optimizer = optimizer
        self.lr = lr
        self.eps = eps
        # rows owned by this rank: ids with id % world == rank,
        # stored at local index id // world
        n_local = int(math.ceil((num_embeddings - self.rank) / self.world)) \
            if num_embeddings > self.rank else 0
        g = torch.Generator().manual_seed(seed + self.rank)
        self.local = torch.randn(n_local, dim, generator=g).to(
            self.device, dtype) * init_std
        if optimizer == "adagrad":
            self.state = torch.zeros(n_local, dim, device=self.device,
                                     dtype=torch.float32)

    # ---- routing ---------------------------------------------------------
    def _route(self, ids: torch.Tensor):
        """Sort ids by owner; returns (sorted_ids, perm, counts_per_owner)."""
        owner = ids % self.world
        perm = torch.argsort(owner, stable=True)
        counts = torch.bincount(owner, minlength=self.world)
        return ids[perm], perm, counts

    def _exchange_counts(self, counts: torch.Tensor) -> torch.Tensor:
        dev = self.comm.device
        g = self.comm.group(list(range(self.world)))
        out = torch.empty(self.world, dtype=torch.int64, device=dev)
        dist.all_to_all_single(out, counts.to(dev).contiguous(), group=g)
        return out.cpu()

    def _exchange(self, flat, send_counts, recv_counts, row_shape, dtype):
        """all-to-all with per-peer variable sizes (alltoall_base — the
        one collective gloo and RCCL both provide)."""
        dev = self.comm.device
        g = self.comm.group(list(range(self.world)))
        out = torch.empty((int(sum(recv_counts)),) + row_shape,
                          dtype=dtype, device=dev)
        dist.all_to_all_single(out, flat.to(dev).contiguous(),
                               output_split_sizes=[int(c) for c in
                                                   recv_counts],
                               input_split_sizes=[int(c) for c in
                                                  send_counts], group=g)
        return list(out.split([int(c) for c in recv_counts]))

    # ---- pull / push -----------------------------------------------------
    def pull(self, ids: torch.Tensor) -> torch.Tensor:
        """Fetch rows for (possibly duplicated) ids; returns [n, dim] on
        ids.device."""
        out_device = ids.device
        ids = ids.reshape(-1).long().cpu()
        if self.world == 1 or not dist.is_initialized():
            return self.local[ids // self.world].to(out_device)
        sids, perm, counts = self._route(ids)
        rcounts = self._exchange_counts(counts)
        req = self._exchange(sids, counts.tolist(), rcounts.tolist(),
                             (), torch.int64)
        # serve: gather owned rows for each requester
        replies = torch.cat([self.local[(r.cpu() // self.world)]
                             for r in req]) if req else \
            torch.empty(0, self.dim, dtype=self.dtype)
        rows = self._exchange(replies, rcounts.tolist(), counts.tolist(),
                              (self.dim,), self.dtype)
        rows = torch.cat([r.cpu() for r in rows]) if rows else \
            torch.empty(0, self.dim, dtype=self.dtype)
        out = torch.empty_like(rows)
        out[perm] = rows
        return out.to(out_device)

    def push(self, ids: torch.Tensor, grads: torch.Tensor):
        """Send gradients to owners; owner applies the sparse optimizer.
        Duplicate ids are accumulated (sum) before the update."""
        ids = ids.reshape(-1).long().cpu()
        grads = grads.reshape(-1, self.dim).to(self.dtype).cpu()
        # local dedup first: fewer bytes on the wire
        uniq, inv = torch.unique(ids, return_inverse=True)
        acc = torch.zeros(uniq.numel(), self.dim, dtype=self.dtype)
        acc.index_add_(0, inv, grads)
        if self.world == 1 or not dist.is_initialized():
            self._apply(uniq, acc)
            return
        sids, perm, counts = self._route(uniq)
        rcounts = self._exchange_counts(counts)
        req = self._exchange(sids, counts.tolist(), rcounts.tolist(),
                             (), torch.int64)
        gparts = self._exchange(acc[perm], counts.tolist(),
                                rcounts.tolist(), (self.dim,), self.dtype)
        for rids, rg in zip(req, gparts):
            if rids.numel():
                self._apply(rids.cpu(), rg.cpu())

    def _apply(self, ids: torch.Tensor, grads: torch.Tensor):
        li = ids // self.world
        # cross-worker duplicates: accumulate again before the update
        uniq, inv = torch.unique(li, return_inverse=True)
        acc = torch.zeros(uniq.numel(), self.dim, dtype=self.dtype,
                          device=self.device)
        acc.index_add_(0, inv, grads.to(self.device))
        if self.optimizer == "sgd":
            self.local.index_add_(0, uniq.to(self.device), acc,
                                  alpha=-self.lr)
        else:  # adagrad
            st = self.state[uniq] + acc.float() ** 2
            self.state[uniq] = st
            self.local[uniq] -= (self.lr * acc.float()
                                 / (st.sqrt() + self.eps)).to(self.dtype)

    # ---- debug/test helpers ---------------------------------------------
    def gather_full(self) -> torch.Tensor:
        """All ranks: materialize the full [num, dim] table (tests only)."""
        if self.world == 1 or not dist.is_initialized():
            full = torch.zeros(self.num, self.dim, dtype=self.dtype)
            full[torch.arange(self.num)] = self.local.cpu()
            return full
        full = torch.zeros(self.num, self.dim, dtype=self.dtype,
                           device=self.comm.device)
        mine = torch.arange(self.rank, self.num, self.world)
        full[mine] = self.local.to(self.comm.device, self.dtype)
        self.comm.allreduce(full, list(range(self.world)))
        return full.cpu()
