"""Context parallelism: ring attention over RCCL batched p2p.

Reference parity: hetu/graph/ops/ParallelAttention.cc (AttnCommRing :165,
GenerateAttnInfo :212, ExecFlashAttn/ExecCorr .h:411-424, KV rotation on
kP2PStream, online-softmax merge, piggybacked dKV accumulation in the
backward ring).

MI355X-native: each CP rank holds a contiguous seq chunk (NORMAL split);
KV blocks rotate around the ring with ONE batched isend/irecv per step
(single-hop on the fully-connected xGMI mesh); the local compute is the
hand-written fa2 flash-attention kernel; partial outputs merge with the
standard log-sum-exp correction.  The backward rotates KV again and
piggybacks the accumulated dKV block around the full ring so it arrives
back at its owner (reference piggyback_grad).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..ops import functional as F
from .comm import CommBackend


def _merge(o1, lse1, o2, lse2):
    """Online-softmax merge of two attention partials (ExecCorr)."""
    lse = torch.logaddexp(lse1, lse2)
    w1 = torch.exp(lse1 - lse).nan_to_num(0.0).unsqueeze(-1)
    w2 = torch.exp(lse2 - lse).nan_to_num(0.0).unsqueeze(-1)
    o = o1.float() * w1 + o2.float() * w2
    return o.to(o1.dtype), lse


def _ring_exchange(comm: CommBackend, ranks: List[int], my_pos: int,
                   tensors: List[torch.Tensor]
                   ) -> List[torch.Tensor]:
    """Send `tensors` to the next ring member, receive same-shaped ones
    from the previous, as one batched p2p group call."""
    n = len(ranks)
    nxt = ranks[(my_pos + 1) % n]
    prv = ranks[(my_pos - 1) % n]
    import torch.distributed as dist
    recvs = [torch.empty_like(t) for t in tensors]
    ops = []
    # tag-free ordering: every rank posts sends before recvs; pairing is by
    # (src, dst) program order which is identical ring-wide
    for t in tensors:
        ops.append(dist.P2POp(dist.isend, t.contiguous(), nxt))
    for t in recvs:
        ops.append(dist.P2POp(dist.irecv, t, prv))
    for r in dist.batch_isend_irecv(ops):
        r.wait()
    return recvs


def ring_attn_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  comm: Optional[CommBackend], ranks: List[int],
                  causal: bool = True, scale: Optional[float] = None
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """q,k,v: local seq chunk [B, H, S_loc, D] (NORMAL split: rank at ring
    position p owns global seq [p*S_loc, (p+1)*S_loc)).  Returns local
    (o, lse)."""
    n = len(ranks)
    if n <= 1 or comm is None:
        return F.flash_attn_fwd(q, k, v, causal, scale)
    my_pos = ranks.index(comm.rank)
    o, lse = None, None
    kv_k, kv_v = k, v
    src_pos = my_pos                      # owner of the current kv block
    for step in range(n):
        if causal and src_pos > my_pos:
            pass                          # fully masked block
        else:
            blk_causal = causal and (src_pos == my_pos)
            ob, lseb = F.flash_attn_fwd(q, kv_k, kv_v, blk_causal, scale)
            if o is None:
                o, lse = ob, lseb
            else:
                o, lse = _merge(o, lse, ob, lseb)
        if step < n - 1:
            kv_k, kv_v = _ring_exchange(comm, ranks, my_pos, [kv_k, kv_v])
            src_pos = (src_pos - 1) % n
    if o is None:                         # degenerate: everything masked
        o = torch.zeros_like(q)
        lse = torch.full(q.shape[:-1], float("-inf"), dtype=torch.float32,
                         device=q.device)
    return o, lse


def ring_attn_fwd_hetero(q, k, v, comm, ranks, seq_lens, causal=True,
                         scale=None):
    """Heterogeneous context parallelism (reference hetero CP,
    trainer.py:255-259 `[32k, 32k, 48k...]`): NORMAL split with UNEQUAL
    per-rank seq shards — rank at ring position p owns global rows
    [sum(seq_lens[:p]), +seq_lens[p]).  KV blocks of differing sizes
    rotate the ring; recv shapes come from the static seq_lens."""
    n = len(ranks)
    if n <= 1 or comm is None:
        return F.flash_attn_fwd(q, k, v, causal, scale)
    my_pos = ranks.index(comm.rank)
    offs = [0]
    for s in seq_lens:
        offs.append(offs[-1] + s)
    my_off = offs[my_pos]
    my_len = seq_lens[my_pos]
    o, lse = None, None
    kv_k, kv_v = k, v
    src_pos = my_pos
    B, H, _, D = q.shape
    for step in range(n):
        src_off = offs[src_pos]
        src_len = seq_lens[src_pos]
        skip = causal and src_off >= my_off + my_len
        if not skip and src_len > 0 and my_len > 0:
            blk_causal = causal and (src_pos == my_pos)
            ob, lseb = F.flash_attn_fwd(q, kv_k, kv_v, blk_causal, scale)
            if o is None:
                o, lse = ob, lseb
            else:
                o, lse = _merge(o, lse, ob, lseb)
        if step < n - 1:
            nxt_src = (src_pos - 1) % n
            shp = (B, H, seq_lens[nxt_src], D)
            kv_k, kv_v = _ring_exchange_shaped(
                comm, ranks, my_pos, [kv_k, kv_v], [shp, shp])
            src_pos = nxt_src
    if o is None:
        o = torch.zeros_like(q)
        lse = torch.full(q.shape[:-1], float("-inf"), dtype=torch.float32,
                         device=q.device)
    return o, lse


def ring_attn_bwd_hetero(dout, q, k, v, o, lse, comm, ranks, seq_lens,
                         causal=True, scale=None):
    n = len(ranks)
    if n <= 1 or comm is None:
        return F.flash_attn_bwd(dout, q, k, v, o, lse, causal, scale)
    my_pos = ranks.index(comm.rank)
    offs = [0]
    for s in seq_lens:
        offs.append(offs[-1] + s)
    my_off, my_len = offs[my_pos], seq_lens[my_pos]
    dq = torch.zeros_like(q, dtype=torch.float32)
    dk_acc = torch.zeros_like(k, dtype=torch.float32)
    dv_acc = torch.zeros_like(v, dtype=torch.float32)
    kv_k, kv_v = k, v
    src_pos = my_pos
    B, H, _, D = q.shape
    for step in range(n):
        src_off, src_len = offs[src_pos], seq_lens[src_pos]
        skip = causal and src_off >= my_off + my_len
        if not skip and src_len > 0 and my_len > 0:
            blk_causal = causal and (src_pos == my_pos)
            dqb, dkb, dvb = F.flash_attn_bwd(dout, q, kv_k, kv_v, o, lse,
                                             blk_causal, scale)
            dq += dqb.float()
            dk_acc += dkb.float()
            dv_acc += dvb.float()
        if step < n - 1:
            nxt_src = (src_pos - 1) % n
            shp = (B, H, seq_lens[nxt_src], D)
            kv_k, kv_v, dk_acc, dv_acc = _ring_exchange_shaped(
                comm, ranks, my_pos, [kv_k, kv_v, dk_acc, dv_acc],
                [shp, shp, shp, shp])
            src_pos = nxt_src
    # after n-1 hops the block here is owned by my_pos+1; one more
    # exchange delivers the my_pos-owned accumulator home
    home = (B, H, my_len, D)
    dk_acc, dv_acc = _ring_exchange_shaped(
        comm, ranks, my_pos, [dk_acc, dv_acc], [home, home])
    return dq.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype)


def _ring_exchange_shaped(comm, ranks, my_pos, tensors, recv_shapes):
    """Ring exchange where the incoming block's shape differs from the
    outgoing one (hetero CP)."""
    n = len(ranks)
    nxt = ranks[(my_pos + 1) % n]
    prv = ranks[(my_pos - 1) % n]
    import torch.distributed as dist
    recvs = [torch.empty(s, dtype=t.dtype, device=t.device)
             for t, s in zip(tensors, recv_shapes)]
    ops = []
    for t in tensors:
        ops.append(dist.P2POp(dist.isend, t.contiguous(), nxt))
    for r in recvs:
        ops.append(dist.P2POp(dist.irecv, r, prv))
    for r in dist.batch_isend_irecv(ops):
        r.wait()
    return recvs


def _sym_subblocks(my_pos: int, src_pos: int, n: int):
    """Sub-block schedule for the SYM (zigzag) split: rank p owns global
    chunks (p, 2n-1-p) as its [head | tail] halves.  Yields
    (q_half, kv_half, causal_diag) for every visible pair — causal work is
    (2n+1) chunk-pairs for EVERY rank (reference STRIPE/SYM split,
    ParallelAttention.cc:196-204; data side: bucket.generate_cp_pack_data).
    """
    mine = (my_pos, 2 * n - 1 - my_pos)
    src = (src_pos, 2 * n - 1 - src_pos)
    for qh, qc in enumerate(mine):
        for kh, kc in enumerate(src):
            if kc < qc:
                yield qh, kh, False
            elif kc == qc:
                yield qh, kh, True


def ring_attn_fwd_sym(q, k, v, comm, ranks, scale=None):
    """Causal ring attention with the SYM split: local seq = [head|tail]
    halves of the global sequence (chunks p and 2n-1-p)."""
    n = len(ranks)
    half = q.shape[2] // 2
    my_pos = ranks.index(comm.rank)
    B, H, S, D = q.shape
    o = torch.zeros(B, H, S, D, dtype=torch.float32, device=q.device)
    lse = torch.full((B, H, S), float("-inf"), dtype=torch.float32,
                     device=q.device)
    kv_k, kv_v = k, v
    src_pos = my_pos
    for step in range(n):
        for qh, kh, diag in _sym_subblocks(my_pos, src_pos, n):
            qs = slice(qh * half, (qh + 1) * half)
            ks = slice(kh * half, (kh + 1) * half)
            ob, lb = F.flash_attn_fwd(q[:, :, qs], kv_k[:, :, ks],
                                      kv_v[:, :, ks], diag, scale)
            om, lm = _merge(o[:, :, qs].to(q.dtype), lse[:, :, qs], ob, lb)
            o[:, :, qs] = om.float()
            lse[:, :, qs] = lm
        if step < n - 1:
            kv_k, kv_v = _ring_exchange(comm, ranks, my_pos, [kv_k, kv_v])
            src_pos = (src_pos - 1) % n
    return o.to(q.dtype), lse


def ring_attn_bwd_sym(dout, q, k, v, o, lse, comm, ranks, scale=None):
    n = len(ranks)
    half = q.shape[2] // 2
    my_pos = ranks.index(comm.rank)
    dq = torch.zeros_like(q, dtype=torch.float32)
    dk_acc = torch.zeros_like(k, dtype=torch.float32)
    dv_acc = torch.zeros_like(v, dtype=torch.float32)
    kv_k, kv_v = k, v
    src_pos = my_pos
    for step in range(n):
        for qh, kh, diag in _sym_subblocks(my_pos, src_pos, n):
            qs = slice(qh * half, (qh + 1) * half)
            ks = slice(kh * half, (kh + 1) * half)
            dqb, dkb, dvb = F.flash_attn_bwd(
                dout[:, :, qs], q[:, :, qs], kv_k[:, :, ks],
                kv_v[:, :, ks], o[:, :, qs], lse[:, :, qs], diag, scale)
            dq[:, :, qs] += dqb.float()
            dk_acc[:, :, ks] += dkb.float()
            dv_acc[:, :, ks] += dvb.float()
        if step < n - 1:
            kv_k, kv_v, dk_acc, dv_acc = _ring_exchange(
                comm, ranks, my_pos, [kv_k, kv_v, dk_acc, dv_acc])
            src_pos = (src_pos - 1) % n
    dk_acc, dv_acc = _ring_exchange(comm, ranks, my_pos, [dk_acc, dv_acc])
    return dq.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype)


def ring_attn_bwd(dout: torch.Tensor, q: torch.Tensor, k: torch.Tensor,
                  v: torch.Tensor, o: torch.Tensor, lse: torch.Tensor,
                  comm: Optional[CommBackend], ranks: List[int],
                  causal: bool = True, scale: Optional[float] = None):
    """Backward ring: rotate (k, v, dk_acc, dv_acc) a full cycle; each rank
    adds its (q-chunk x current-kv-block) contribution; after n steps the
    accumulated dKV block returns to its owner.  dq accumulates locally.
    The per-pair backward uses the GLOBAL lse/o (p = exp(s - lse_global))."""
    n = len(ranks)
    if n <= 1 or comm is None:
        return F.flash_attn_bwd(dout, q, k, v, o, lse, causal, scale)
    my_pos = ranks.index(comm.rank)
    dq = torch.zeros_like(q, dtype=torch.float32)
    dk_acc = torch.zeros_like(k, dtype=torch.float32)
    dv_acc = torch.zeros_like(v, dtype=torch.float32)
    kv_k, kv_v = k, v
    src_pos = my_pos
    for step in range(n):
        if not (causal and src_pos > my_pos):
            blk_causal = causal and (src_pos == my_pos)
            dqb, dkb, dvb = F.flash_attn_bwd(dout, q, kv_k, kv_v, o, lse,
                                             blk_causal, scale)
            dq += dqb.float()
            dk_acc += dkb.float()
            dv_acc += dvb.float()
        if step < n - 1:
            kv_k, kv_v, dk_acc, dv_acc = _ring_exchange(
                comm, ranks, my_pos, [kv_k, kv_v, dk_acc, dv_acc])
            src_pos = (src_pos - 1) % n
    # after n-1 rotations the block that started at my_pos+1 is here; one
    # more exchange returns each accumulated dKV to its owner
    dk_acc, dv_acc = _ring_exchange(comm, ranks, my_pos, [dk_acc, dv_acc])
    return dq.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype)
