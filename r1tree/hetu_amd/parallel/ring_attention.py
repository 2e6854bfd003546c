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
