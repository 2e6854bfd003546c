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
    dropped), topk idx [N, K], weights [N, K]."""
    N = probs.shape[0]
    w, idx = probs.topk(K, dim=-1)                      # [N, K]
    pos = torch.full((E, C), -1, dtype=torch.int64, device=probs.device)
    slot_of = torch.full((N, K), -1, dtype=torch.int64,
                         device=probs.device)
    # position within expert queue per (token, k): rank among tokens
    # choosing that expert (first-come order, GShard style)
    for k in range(K):
        e = idx[:, k]                                   # [N]
        onehot = torch.nn.functional.one_hot(e, E)      # [N, E]
        # priority: tokens already queued from earlier k slots
        base = (pos >= 0).sum(-1)                       # [E]
        order = onehot.cumsum(0) * onehot               # 1-based rank
        q = (order.gather(1, e.unsqueeze(1)).squeeze(1) - 1) + base[e]
        keep = q < C
        tok = torch.nonzero(keep, as_tuple=False).squeeze(1)
        if tok.numel():
            flat = e[tok] * C + q[tok]
            pos.view(-1)[flat] = tok
            slot_of[tok, k] = flat
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
        send = x.new_zeros(E * C, x.shape[-1])
        valid = pos.view(-1) >= 0
        send[valid] = x[pos.view(-1)[valid]]
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
        valid = pos.view(-1) >= 0
        dx.index_add_(0, pos.view(-1)[valid], gsend[valid].to(x.dtype))
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
            ok = slot_of[:, k] >= 0
            tok = torch.nonzero(ok, as_tuple=False).squeeze(1)
            if tok.numel():
                rows = recv[slot_of[tok, k]]
                y[tok] += rows * wk[tok, k].unsqueeze(-1).to(rows.dtype)
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
        # d_recv [E*C, h]: rows scattered from gy * w
        d_recv = gy.new_zeros(E * C, gy.shape[-1])
        for k in range(K):
            ok = slot_of[:, k] >= 0
            tok = torch.nonzero(ok, as_tuple=False).squeeze(1)
            if tok.numel():
                d_recv[slot_of[tok, k]] = \
                    gy[tok] * wk[tok, k].unsqueeze(-1).to(gy.dtype)
        # d_combine_w: dot(gy[token], recv[slot])
        recv = MoECombineOp._gather_back(eo, pos, ranks, ctx, E, C, P, El)
        dwk = torch.zeros_like(wk)
        for k in range(K):
            ok = slot_of[:, k] >= 0
            tok = torch.nonzero(ok, as_tuple=False).squeeze(1)
            if tok.numel():
                dwk[tok, k] = (gy[tok].float()
                               * recv[slot_of[tok, k]].float()).sum(-1)
        # forward a2a of d_recv to expert layout
        if P > 1:
            recv2 = _a2a(ctx.comm, ranks, d_recv)
            d_eo = recv2.reshape(P, El, C, -1).transpose(0, 1) \
                .reshape(El, P * C, -1).contiguous()
        else:
            d_eo = d_recv.reshape(El, P * C, -1)
        return [d_eo, dwk]
