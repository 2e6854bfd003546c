"""MoE collectives: flat and hierarchical all-to-all over RCCL.

Reference parity: HetuMoE (arXiv:2203.14685) — v1/python/hetu/gpu_ops/
HAllToAll.py:9-99 and v1/src/ops/H_A2A_LayoutTransform.cu: the hierarchical
all-to-all gathers each node's shards per destination node, runs ONE
inter-node exchange per node pair, and scatters intra-node — turning P^2
small cross-node messages into node_count^2 large ones.

MI355X re-staging: inside one node the 8 GPUs are a fully-connected xGMI
mesh, so the intra-node phases are single-hop RCCL collectives on the node
subgroup; the inter-node phase (one buffer per node pair) rides the NIC.
On a single node it degenerates to the flat a2a.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from .comm import CommBackend


def alltoall(comm: Optional[CommBackend], ranks: List[int],
             x: torch.Tensor) -> torch.Tensor:
    """Flat all-to-all: x [P*chunk, ...] row-blocks; block i goes to
    ranks[i], which returns its block for me."""
    P = len(ranks)
    if P <= 1 or comm is None or not dist.is_initialized():
        return x
    g = comm.group(ranks)
    out = torch.empty_like(x)
    dist.all_to_all_single(out, x.contiguous(), group=g)
    return out


def hierarchical_alltoall(comm: Optional[CommBackend], ranks: List[int],
                          x: torch.Tensor, node_size: int) -> torch.Tensor:
    """Equivalent to alltoall(comm, ranks, x) for P = n_nodes * node_size
    ranks with rank (a, j) = ranks[a*node_size + j].

    Phase 1 (intra a2a): local slot m of node a collects node a's rows
    destined to node m -> [src_local j][dst_local l][chunk].
    Phase 2 (inter p2p): rank (a, m) <-> rank (m, a) swap buffers; after
    the swap (a, m) holds node m's rows for node a.
    Phase 3 (intra a2a over dst_local): distribute to final owners.

    Requires n_nodes <= node_size (pad slots carry zeros otherwise fall
    back to the flat a2a).
    """
    P = len(ranks)
    if P <= 1 or comm is None or not dist.is_initialized():
        return x
    if node_size <= 1 or P % node_size != 0:
        return alltoall(comm, ranks, x)
    n_nodes = P // node_size
    if n_nodes == 1:
        return alltoall(comm, ranks, x)
    if n_nodes > node_size:
        return alltoall(comm, ranks, x)

    me = ranks.index(comm.rank)
    a, j = me // node_size, me % node_size
    node_ranks = ranks[a * node_size:(a + 1) * node_size]
    chunk = x.shape[0] // P
    tail = x.shape[1:]

    # ---- phase 1: [dst_node, dst_local, chunk] -> slot dst_node ----------
    xr = x.reshape(n_nodes, node_size * chunk, *tail)
    if n_nodes < node_size:
        xr = torch.cat([xr, xr.new_zeros(node_size - n_nodes,
                                         node_size * chunk, *tail)], 0)
    p1 = alltoall(comm, node_ranks, xr.reshape(-1, *tail))
    # p1 on rank (a, m): [src_local j, dst_local l, chunk] for dst node m
    m = j

    # ---- phase 2: swap (a, m) <-> (m, a) ---------------------------------
    if m < n_nodes and a != m:
        partner = ranks[m * node_size + a]
        recv = torch.empty_like(p1)
        reqs = dist.batch_isend_irecv([
            dist.P2POp(dist.isend, p1.contiguous(), partner),
            dist.P2POp(dist.irecv, recv, partner)])
        for r in reqs:
            r.wait()
        p2 = recv
    elif m < n_nodes:
        p2 = p1
    else:
        p2 = torch.zeros_like(p1)
    # p2 on rank (a, m): node m's rows destined to node a:
    # [src_local s, dst_local l, chunk]

    # ---- phase 3: intra a2a over dst_local -------------------------------
    p2 = p2.reshape(node_size, node_size, chunk, *tail)     # [s][l][c]
    p3 = p2.transpose(0, 1).contiguous()                    # [l][s][c]
    got = alltoall(comm, node_ranks, p3.reshape(-1, *tail))
    # got on rank (a, l): slot m = [src_local s, chunk] from src node m
    got = got.reshape(node_size, node_size, chunk, *tail)
    out = got[:n_nodes].reshape(P * chunk, *tail)
    return out.contiguous()
