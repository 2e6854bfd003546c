"""Expert-parallel MoE ops: capacity-based dispatch / combine over RCCL a2a.

Reference parity: HetuMoE — v1/python/hetu/layers/moe_layer.py (top-k gate,
capacity dispatch), gpu_ops/AllToAll.py and HAllToAll.py (flat +
hierarchical a2a).  MI355X-native: static [E, C] routing buffers keep every
shape graph-static (hipGraph-capturable); the a2a rides RCCL over xGMI,
optionally hierarchical across nodes (HETU_AMD_MOE_NODE_SIZE).

Layouts (P = ep group size, E = total experts, El = E/P local experts,
C = per-(source rank, expert) capacity):
  dispatch:  x [N, h], probs [N, E] ->
    expert_in  [El, P*C, h]   (rows from every source rank)
    combine_w  [N, K] fp32    (gate weight per used slot; 0 if dropped)
    meta:      pos [E, C] int64 (token index per slot, -1 pad) — kept as a
               graph tensor so combine/grads replay the routing
  combine: expert_out [El, P*C, h], combine_w, pos -> y [N, h]
"""
from __future__ import annotations

import os

import torch

from ...parallel.dstates import DistributedStates
from ..op import OpInterface
from ..tensor import TensorMeta
from .basics import _g, _make
from .comm import _my_index, _ranks


def _ep_ranks(op, ctx):
    ranks = op.attrs.get("ep_ranks")
    if ranks is None or ctx.comm is None:
        return [0]
    return list(ranks)


def _node_size():
    return int(os.environ.get("HETU_AMD_MOE_NODE_SIZE", "0"))


def _a2a(comm, ranks, x):
    from ...parallel.moe import alltoall, hierarchical_alltoall
    ns = _node_size()
    if ns > 1:
        return hierarchical_alltoall(comm, ranks, x, ns)
    return alltoall(comm, ranks, x)


def _route(probs: torch.Tensor, E: int, C: int, K: int):
    """Greedy capacity routing.  Returns pos [E, C] int64 (-1 pad),
    combine scatter info: slot_of [N, K] int64 (flat index into E*C, -1 if
    dropped), topk idx [N, K], weights [N, K].

    Fully static shapes: no nonzero/boolean-mask gathers, so the routing
    is hipGraph-capturable and never syncs the host (dropped tokens
    scatter into a sacrificial slot E*C instead)."""
    N = probs.shape[0]
    dev = probs.device
    w, idx = probs.topk(K, dim=-1)                      # [N, K]
    # pos with one extra dummy slot at index E*C for dropped tokens
    pos_fl = torch.full((E * C + 1,), -1, dtype=torch.int64, device=dev)
    slot_of = torch.full((N, K), -1, dtype=torch.int64, device=dev)
    toks = torch.arange(N, dtype=torch.int64, device=dev)
    base = torch.zeros(E, dtype=torch.int64, device=dev)
    for k in range(K):
        e = idx[:, k]                                   # [N]
        onehot = torch.nn.functional.one_hot(e, E)      # [N, E]
        order = onehot.cumsum(0) * onehot               # 1-based rank
        q = (order.gather(1, e.unsqueeze(1)).squeeze(1) - 1) + base[e]
        keep = q < C
        flat = torch.where(keep, e * C + q,
                           torch.full_like(q, E * C))  # dummy if dropped
        pos_fl.scatter_(0, flat, toks)
        slot_of[:, k] = torch.where(keep, flat,
                                    torch.full_like(flat, -1))
        # recount filled slots per expert (robust to same-slot rewrites)
        base = (pos_fl[:E * C].reshape(E, C) >= 0).sum(-1)
    pos = pos_fl[:E * C].reshape(E, C)
    wk = torch.where(slot_of >= 0, w, torch.zeros_like(w))
    return pos, slot_of, idx, wk


class MoEDispatchOp(OpInterface):
    """inputs: x [N, h], probs [N, E]; outputs: expert_in [El, P*C, h],
    combine_w [N, K] (differentiable wrt probs), pos [E, C], slot_of
    [N, K]."""
    type = "MoEDispatch"

    def infer_meta(self, attrs, inputs):
        x, probs = inputs
        E, C, K = attrs["experts"], attrs["capacity"], attrs["k"]
        P = len(attrs.get("ep_ranks") or [0])
        El = E // P
        return [TensorMeta([El, P * C, x.shape[-1]], x.dtype),
                TensorMeta([x.shape[0], K], torch.float32),
                TensorMeta([E, C], torch.int64),
                TensorMeta([x.shape[0], K], torch.int64)]

    def deduce_states(self, op):
        x = op.inputs[0]
        for t in op.outputs:
            t.device_group = x.device_group
        # expert_in is sharded over the ep group (local experts); the rest
        # mirror x's token layout / are per-rank routing state
        if x.ds is not None:
            for t in op.outputs:
                t.ds = DistributedStates(x.ds.device_num, {-1: x.ds.device_num}
                                         if x.ds.device_num > 1 else {})

    def compute(self, op, inputs, ctx):
        x, probs = inputs
        a = op.attrs
        E, C, K = a["experts"], a["capacity"], a["k"]
        ranks = _ep_ranks(op, ctx)
        P = len(ranks)
        El = E // P
        pos, slot_of, idx, wk = _route(probs.float(), E, C, K)
        # static-shape gather: dropped slots read token 0 and are masked
        pf = pos.view(-1)
        send = x[pf.clamp(min=0)] * (pf >= 0).unsqueeze(-1).to(x.dtype)
        if P > 1:
            # [E*C, h] = [P, El*C, h] blocks by destination rank
            recv = _a2a(ctx.comm, ranks, send)
            expert_in = recv.reshape(P, El, C, -1).transpose(0, 1) \
                .reshape(El, P * C, -1).contiguous()
        else:
            expert_in = send.reshape(El, C, -1)
            expert_in = expert_in.reshape(El, P * C, -1)
        return [expert_in, wk, pos, slot_of]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        g_exp = g[0]
        g_w = g[1]
        grads = _make(gr, MoEDispatchGradOp(),
                      [t for t in [g_exp, g_w, op.outputs[2], op.outputs[3],
                                   op.inputs[0], op.inputs[1]]
                       if t is not None],
                      dict(op.attrs), name="moe_dispatch_grad")
        return [grads.output(0), grads.output(1)]


class MoEDispatchGradOp(OpInterface):
    """inputs: g_expert_in (may be zeros), g_combine_w, pos, slot_of, x,
    probs -> dx [N, h], dprobs [N, E]."""
    type = "MoEDispatchGrad"

    def infer_meta(self, attrs, inputs):
        x, probs = inputs[4], inputs[5]
        return [TensorMeta(x.shape, x.dtype),
                TensorMeta(probs.shape, probs.dtype)]

    def compute(self, op, inputs, ctx):
        g_exp, g_w, pos, slot_of, x, probs = inputs
        a = op.attrs
        E, C, K = a["experts"], a["capacity"], a["k"]
        ranks = _ep_ranks(op, ctx)
        P = len(ranks)
        El = E // P
        # reverse the a2a: expert_in grads back to source layout [E*C, h]
        if P > 1:
            back = g_exp.reshape(El, P, C, -1).transpose(0, 1) \
                .reshape(P * El * C, -1).contiguous()
            gsend = _a2a(ctx.comm, ranks, back)
        else:
            gsend = g_exp.reshape(E * C, -1)
        dx = torch.zeros_like(x)
        pf = pos.view(-1)
        dx.index_add_(0, pf.clamp(min=0),
                      (gsend * (pf >= 0).unsqueeze(-1)).to(x.dtype))
        # dprobs: combine_w = probs.gather(topk) masked -> scatter g_w
        dprobs = torch.zeros_like(probs)
        if g_w is not None:
            wmask = (slot_of >= 0).to(probs.dtype)
            # idx recomputed from probs (same topk order)
            _, idx = probs.float().topk(K, dim=-1)
            dprobs.scatter_(1, idx, (g_w.to(probs.dtype) * wmask))
        return [dx, dprobs]


class MoECombineOp(OpInterface):
    """inputs: expert_out [El, P*C, h], combine_w [N, K], pos [E, C],
    slot_of [N, K] -> y [N, h]."""
    type = "MoECombine"

    def infer_meta(self, attrs, inputs):
        eo, wk = inputs[0], inputs[1]
        return [TensorMeta([wk.shape[0], eo.shape[-1]], eo.dtype)]

    def deduce_states(self, op):
        src = op.attrs.get("out_ds")
        op.outputs[0].ds = src
        op.outputs[0].device_group = op.inputs[0].device_group

    @staticmethod
    def _gather_back(eo, pos, ranks, ctx, E, C, P, El):
        if P > 1:
            back = eo.reshape(El, P, C, -1).transpose(0, 1) \
                .reshape(P * El * C, -1).contiguous()
            recv = _a2a(ctx.comm, ranks, back)
        else:
            recv = eo.reshape(E * C, -1)
        return recv      # [E*C, h] rows in send-slot order

    def compute(self, op, inputs, ctx):
        eo, wk, pos, slot_of = inputs
        a = op.attrs
        E, C, K = a["experts"], a["capacity"], a["k"]
        ranks = _ep_ranks(op, ctx)
        P = len(ranks)
        El = E // P
        recv = self._gather_back(eo, pos, ranks, ctx, E, C, P, El)
        N = wk.shape[0]
        y = eo.new_zeros(N, eo.shape[-1])
        for k in range(K):
            sl = slot_of[:, k]
            rows = recv[sl.clamp(min=0)]
            # wk is already 0 for dropped slots (masked at routing)
            y += rows * wk[:, k].unsqueeze(-1).to(rows.dtype)
        return [y]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        bwd = _make(gr, MoECombineGradOp(),
                    [g[0], op.inputs[0], op.inputs[1], op.inputs[2],
                     op.inputs[3]], dict(op.attrs), name="moe_combine_grad")
        return [bwd.output(0), bwd.output(1), None, None]


class MoECombineGradOp(OpInterface):
    """inputs: gy [N, h], expert_out, combine_w, pos, slot_of ->
    d_expert_out [El, P*C, h], d_combine_w [N, K]."""
    type = "MoECombineGrad"

    def infer_meta(self, attrs, inputs):
        eo, wk = inputs[1], inputs[2]
        return [TensorMeta(eo.shape, eo.dtype),
                TensorMeta(wk.shape, wk.dtype)]

    def compute(self, op, inputs, ctx):
        gy, eo, wk, pos, slot_of = inputs
        a = op.attrs
        E, C, K = a["experts"], a["capacity"], a["k"]
        ranks = _ep_ranks(op, ctx)
        P = len(ranks)
        El = E // P
        # d_recv [E*C, h]: rows scattered from gy * w (slot E*C is the
        # sacrificial target for dropped entries — static shapes)
        h = gy.shape[-1]
        d_fl = gy.new_zeros(E * C + 1, h)
        for k in range(K):
            sl = slot_of[:, k]
            tgt = torch.where(sl >= 0, sl, torch.full_like(sl, E * C))
            d_fl.scatter_(0, tgt.unsqueeze(-1).expand(-1, h),
                          gy * wk[:, k].unsqueeze(-1).to(gy.dtype))
        d_recv = d_fl[:E * C]
        # d_combine_w: dot(gy[token], recv[slot]); 0 where dropped
        recv = MoECombineOp._gather_back(eo, pos, ranks, ctx, E, C, P, El)
        dwk = torch.zeros_like(wk)
        for k in range(K):
            sl = slot_of[:, k]
            dot = (gy.float() * recv[sl.clamp(min=0)].float()).sum(-1)
            dwk[:, k] = dot * (sl >= 0).to(dot.dtype)
        # forward a2a of d_recv to expert layout
        if P > 1:
            recv2 = _a2a(ctx.comm, ranks, d_recv)
            d_eo = recv2.reshape(P, El, C, -1).transpose(0, 1) \
                .reshape(El, P * C, -1).contiguous()
        else:
            d_eo = d_recv.reshape(El, P * C, -1)
        return [d_eo, dwk]
