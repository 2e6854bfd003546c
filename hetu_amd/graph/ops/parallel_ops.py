"""Tensor-parallel ops: vocab-parallel embedding + vocab-parallel CE loss.

Reference parity: hetu/graph/ops/VocabParallelCrossEntropyLoss.cc (TP-sharded
vocab loss with local max/sum + predicted-logit mask paired with cross-rank
allreduce) and the masked-lookup half of HtMultiVocabParallelEmbedding
(python/hetu/nn/modules/parallel_multi_ds.py:268).  MI355X-native: the three
allreduces ride RCCL via the CommBackend inside compute(); the forward saves
the global log-sum-exp so the backward needs no further collectives.
"""
from __future__ import annotations

import torch

from ...parallel.dstates import DistributedStates, ds_from_index_table
from ..op import OpInterface
from ..tensor import TensorMeta
from .basics import _g, _make
from .comm import _my_index, _ranks


def _shard_info(ds: DistributedStates, dim: int, my_index: int, size: int):
    """(n_shards, my_shard_index, shard_size) along tensor dim `dim`."""
    n = ds.get_dim(dim) if ds is not None else 1
    idx = ds.map_device_to_state_index(my_index).get(dim, 0) if ds else 0
    return n, idx, size // max(n, 1)


def _tp_ranks(ds, device_group, dim, my_index):
    if ds is None or ds.get_dim(dim) <= 1:
        return [0]
    return _ranks(device_group, ds.group_devices_along(dim), my_index)


class VocabParallelEmbeddingOp(OpInterface):
    """inputs: table [V/tp, D] (ds split dim0 over tp), ids [...] ->
    [..., D] partial over tp (caller comms to dup / seq-split)."""
    type = "VocabParallelEmbedding"

    def infer_meta(self, attrs, inputs):
        table, ids = inputs
        return [TensorMeta(list(ids.shape) + [table.shape[1]], table.dtype)]

    def deduce_states(self, op):
        table, ids = op.inputs
        out = op.outputs[0]
        tds = table.ds
        if tds is None or tds.get_dim(0) <= 1:
            # degenerate: plain lookup
            if ids.ds is not None:
                out.ds = DistributedStates(ids.ds.device_num,
                                           dict(ids.ds.states),
                                           list(ids.ds.order))
            out.device_group = ids.device_group or table.device_group
            return
        n = tds.device_num
        tp = tds.get_dim(0)
        table_idx = []
        ids_ds = ids.ds
        counts = {-2: tp}
        for i in range(n):
            ent = {-2: tds.map_device_to_state_index(i).get(0, 0)}
            if ids_ds is not None:
                st = ids_ds.map_device_to_state_index(i)
                for d in ids_ds.split_dims():
                    ent[d] = st.get(d, 0)
            table_idx.append(ent)
        if ids_ds is not None:
            for d in ids_ds.split_dims():
                counts[d] = ids_ds.get_dim(d)
        out.ds = ds_from_index_table(n, table_idx, counts)
        out.device_group = table.device_group or ids.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        table, ids = inputs
        tds = op.inputs[0].ds
        my = _my_index(ctx, op.inputs[0].device_group)
        tp, tp_idx, vlocal = _shard_info(tds, 0, my, op.attrs["vocab"])
        if tp <= 1:
            return [F.embedding_fwd(table, ids)]
        vstart = tp_idx * vlocal
        mask = (ids >= vstart) & (ids < vstart + vlocal)
        local_ids = torch.where(mask, ids - vstart,
                                torch.zeros_like(ids))
        out = F.embedding_fwd(table, local_ids)
        out = out * mask.unsqueeze(-1).to(out.dtype)
        return [out]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        dtable = _make(gr, VocabParallelEmbeddingGradOp(),
                       [g[0], op.inputs[1], op.inputs[0]],
                       dict(op.attrs)).output()
        return [dtable, None]


class VocabParallelEmbeddingGradOp(OpInterface):
    """inputs: gy [..., D] (dup over tp), ids, table -> dtable [V/tp, D]
    (partial over the token-split dims)."""
    type = "VocabParallelEmbeddingGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[2].shape, inputs[0].dtype)]

    def deduce_states(self, op):
        gy, ids, table = op.inputs
        out = op.outputs[0]
        tds = table.ds
        if tds is None:
            out.device_group = table.device_group
            return
        n = tds.device_num
        tp = tds.get_dim(0)
        ids_ds = ids.ds
        npart = 1
        if ids_ds is not None:
            for d in ids_ds.split_dims():
                npart *= ids_ds.get_dim(d)
        table_idx = []
        for i in range(n):
            ent = {}
            if tp > 1:
                ent[0] = tds.map_device_to_state_index(i).get(0, 0)
            if npart > 1:
                st = ids_ds.map_device_to_state_index(i)
                ip = 0
                for d in ids_ds.split_dims():
                    ip = ip * ids_ds.get_dim(d) + st.get(d, 0)
                ent[-2] = ip
            table_idx.append(ent)
        counts = {0: tp, -2: npart}
        out.ds = ds_from_index_table(n, table_idx, counts)
        out.device_group = table.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        gy, ids, table = inputs
        tds = op.inputs[2].ds
        my = _my_index(ctx, op.inputs[2].device_group)
        tp, tp_idx, vlocal = _shard_info(tds, 0, my, op.attrs["vocab"])
        if tp <= 1:
            return [F.embedding_bwd(gy, ids, table.shape[0])]
        vstart = tp_idx * vlocal
        mask = (ids >= vstart) & (ids < vstart + vlocal)
        local_ids = torch.where(mask, ids - vstart, torch.zeros_like(ids))
        gy = gy * mask.unsqueeze(-1).to(gy.dtype)
        return [F.embedding_bwd(gy, local_ids, vlocal)]


class RingAttentionOp(OpInterface):
    """Context-parallel flash attention (reference ParallelAttention.cc /
    AttnCommRing): q,k,v [B, H, S_loc, D] with the seq dim split over the
    cp ring; the op rotates KV blocks over RCCL batched p2p and merges
    partials with the log-sum-exp correction.  attrs: causal, scale,
    cp_ranks (global rank list of this rank's ring)."""
    type = "RingAttention"

    def infer_meta(self, attrs, inputs):
        q = inputs[0]
        rows = list(q.shape[:-1])
        return [TensorMeta(q.shape, q.dtype),
                TensorMeta(rows, torch.float32)]

    def deduce_states(self, op):
        q = op.inputs[0]
        for t in op.outputs:
            if q.ds is not None:
                t.ds = DistributedStates(q.ds.device_num, dict(q.ds.states),
                                         list(q.ds.order))
            t.device_group = q.device_group

    def compute(self, op, inputs, ctx):
        from ...parallel.ring_attention import (ring_attn_fwd,
                                                ring_attn_fwd_sym)
        q, k, v = inputs
        if op.attrs.get("seq_lens") and ctx.comm is not None \
                and len(op.attrs["cp_ranks"]) > 1:
            from ...parallel.ring_attention import ring_attn_fwd_hetero
            o, lse = ring_attn_fwd_hetero(q, k, v, ctx.comm,
                                          op.attrs["cp_ranks"],
                                          op.attrs["seq_lens"],
                                          op.attrs.get("causal", True),
                                          op.attrs.get("scale"))
        elif op.attrs.get("split", "NORMAL") == "SYM" \
                and op.attrs.get("causal", True) and ctx.comm is not None \
                and len(op.attrs["cp_ranks"]) > 1:
            o, lse = ring_attn_fwd_sym(q, k, v, ctx.comm,
                                       op.attrs["cp_ranks"],
                                       op.attrs.get("scale"))
        else:
            o, lse = ring_attn_fwd(q, k, v, ctx.comm, op.attrs["cp_ranks"],
                                   op.attrs["causal"],
                                   op.attrs.get("scale"))
        return [o, lse]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        bwd = _make(gr, RingAttentionGradOp(),
                    [g[0], op.inputs[0], op.inputs[1], op.inputs[2],
                     op.outputs[0], op.outputs[1]], dict(op.attrs))
        return [bwd.output(0), bwd.output(1), bwd.output(2)]


class RingAttentionGradOp(OpInterface):
    type = "RingAttentionGrad"

    def infer_meta(self, attrs, inputs):
        _, q, k, v = inputs[:4]
        return [TensorMeta(q.shape, q.dtype),
                TensorMeta(k.shape, k.dtype),
                TensorMeta(v.shape, v.dtype)]

    def deduce_states(self, op):
        for t, src in zip(op.outputs, op.inputs[1:4]):
            if src.ds is not None:
                t.ds = DistributedStates(src.ds.device_num,
                                         dict(src.ds.states),
                                         list(src.ds.order))
            t.device_group = src.device_group

    def compute(self, op, inputs, ctx):
        from ...parallel.ring_attention import (ring_attn_bwd,
                                                ring_attn_bwd_sym)
        dout, q, k, v, o, lse = inputs
        if op.attrs.get("seq_lens") and ctx.comm is not None \
                and len(op.attrs["cp_ranks"]) > 1:
            from ...parallel.ring_attention import ring_attn_bwd_hetero
            return list(ring_attn_bwd_hetero(
                dout, q, k, v, o, lse, ctx.comm, op.attrs["cp_ranks"],
                op.attrs["seq_lens"], op.attrs.get("causal", True),
                op.attrs.get("scale")))
        if op.attrs.get("split", "NORMAL") == "SYM" \
                and op.attrs.get("causal", True) and ctx.comm is not None \
                and len(op.attrs["cp_ranks"]) > 1:
            return list(ring_attn_bwd_sym(dout, q, k, v, o, lse, ctx.comm,
                                          op.attrs["cp_ranks"],
                                          op.attrs.get("scale")))
        return list(ring_attn_bwd(dout, q, k, v, o, lse, ctx.comm,
                                  op.attrs["cp_ranks"], op.attrs["causal"],
                                  op.attrs.get("scale")))


class VocabParallelCrossEntropyOp(OpInterface):
    """inputs: logits [N, V/tp] (ds split dim1 over tp), labels [N] ->
    per-token loss [N] fp32 (dup over tp) + saved global lse [N].

    Math (VocabParallelCrossEntropyLoss.cu:15-70 semantics):
      gmax  = allreduce_max(max_v logits)
      gsum  = allreduce_sum(sum_v exp(logits - gmax))
      lse   = log(gsum) + gmax
      pred  = allreduce_sum(logit[label] if label in my shard else 0)
      loss  = lse - pred      (0 where label == ignore_index)
    """
    type = "VocabParallelCrossEntropy"

    def infer_meta(self, attrs, inputs):
        rows = list(inputs[0].shape[:-1])
        return [TensorMeta(rows, torch.float32),
                TensorMeta(rows, torch.float32)]

    def deduce_states(self, op):
        logits, labels = op.inputs
        lds = logits.ds
        if lds is None:
            op.outputs[0].device_group = logits.device_group
            op.outputs[1].device_group = logits.device_group
            return
        assert lds.partial <= 1, "partial logits not supported"
        n = lds.device_num
        vdim = logits.ndim - 1
        tp = lds.get_dim(vdim)
        table = []
        counts = {}
        for d in lds.split_dims():
            if d != vdim:
                counts[d] = lds.get_dim(d)
        for i in range(n):
            st = lds.map_device_to_state_index(i)
            ent = {d: st.get(d, 0) for d in counts}
            table.append(ent)
        ds_out = ds_from_index_table(n, table, counts)
        for t in op.outputs:
            t.ds = ds_out
            t.device_group = logits.device_group

    def compute(self, op, inputs, ctx):
        logits, labels = inputs
        labels = labels.reshape(logits.shape[:-1])
        lds = op.inputs[0].ds
        dg = op.inputs[0].device_group
        my = _my_index(ctx, dg)
        vdim = logits.ndim - 1
        tp, tp_idx, vlocal = _shard_info(lds, vdim, my, op.attrs["vocab"])
        ranks = _tp_ranks(lds, dg, vdim, my)
        ignore = op.attrs.get("ignore_index", -100)
        vstart = tp_idx * vlocal
        if logits.is_cuda:
            # kernel path: never materializes an fp32 logits copy
            # (26 GB at the 7B bench shape)
            from ...ops import functional as F
            shape = logits.shape[:-1]
            gmax, pred = F.vocab_parallel_ce_local_stats(
                logits, labels.reshape(-1), vstart, vstart + vlocal,
                ignore if ignore is not None else -100)
            gmax = gmax.reshape(shape)
            pred = pred.reshape(shape)
            if tp > 1 and ctx.comm is not None:
                gmax = ctx.comm.allreduce(gmax, ranks, op="max")
            gsum = F.ext().vp_sumexp(logits.contiguous(),
                                     gmax.reshape(-1)).reshape(shape)
            if tp > 1 and ctx.comm is not None:
                both = torch.stack([gsum, pred], dim=0)
                both = ctx.comm.allreduce(both, ranks, op="sum")
                gsum, pred = both[0], both[1]
        else:
            lf = logits.float()
            gmax = lf.max(dim=-1).values
            if tp > 1 and ctx.comm is not None:
                gmax = ctx.comm.allreduce(gmax, ranks, op="max")
            gsum = torch.exp(lf - gmax.unsqueeze(-1)).sum(-1)
            in_range = (labels >= vstart) & (labels < vstart + vlocal)
            safe = torch.where(in_range, labels - vstart,
                               torch.zeros_like(labels))
            pred = lf.gather(-1, safe.unsqueeze(-1)).squeeze(-1)
            pred = pred * in_range.to(pred.dtype)
            if tp > 1 and ctx.comm is not None:
                both = torch.stack([gsum, pred], dim=0)
                both = ctx.comm.allreduce(both, ranks, op="sum")
                gsum, pred = both[0], both[1]
        lse = torch.log(gsum) + gmax
        loss = lse - pred
        if ignore is not None:
            loss = torch.where(labels == ignore, torch.zeros_like(loss),
                               loss)
        return [loss, lse]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        gin = _make(gr, VocabParallelCrossEntropyGradOp(),
                    [g[0], op.inputs[0], op.inputs[1], op.outputs[1]],
                    dict(op.attrs)).output()
        return [gin, None]


class VocabParallelCrossEntropyGradOp(OpInterface):
    """dlogits = gy * (exp(logits - lse) - onehot_local(label)); no
    collectives needed (lse is global)."""
    type = "VocabParallelCrossEntropyGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[1].dtype)]

    def deduce_states(self, op):
        logits = op.inputs[1]
        out = op.outputs[0]
        if logits.ds is not None:
            out.ds = DistributedStates(logits.ds.device_num,
                                       dict(logits.ds.states),
                                       list(logits.ds.order))
        out.device_group = logits.device_group

    def compute(self, op, inputs, ctx):
        gy, logits, labels, lse = inputs
        labels = labels.reshape(logits.shape[:-1])
        gy = gy.reshape(logits.shape[:-1])
        lds = op.inputs[1].ds
        my = _my_index(ctx, op.inputs[1].device_group)
        vdim = logits.ndim - 1
        tp, tp_idx, vlocal = _shard_info(lds, vdim, my, op.attrs["vocab"])
        ignore = op.attrs.get("ignore_index", -100)
        vstart = tp_idx * vlocal
        if logits.is_cuda:
            from ...ops import functional as F
            return [F.ext().vp_ce_bwd(
                gy.reshape(-1).contiguous(), logits.contiguous(),
                labels.reshape(-1).contiguous(),
                lse.reshape(-1).contiguous(), vstart, vstart + vlocal,
                ignore if ignore is not None else -(1 << 40))]
        sm = torch.exp(logits.float() - lse.unsqueeze(-1))
        in_range = (labels >= vstart) & (labels < vstart + vlocal)
        safe = torch.where(in_range, labels - vstart,
                           torch.zeros_like(labels))
        onehot = torch.zeros_like(sm)
        onehot.scatter_(-1, safe.unsqueeze(-1),
                        in_range.to(sm.dtype).unsqueeze(-1))
        scale = gy.float()
        if ignore is not None:
            scale = torch.where(labels == ignore, torch.zeros_like(scale),
                                scale)
        return [((sm - onehot) * scale.unsqueeze(-1)).to(logits.dtype)]
