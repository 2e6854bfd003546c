"""CommOp: abstract redistribution between DistributedStates layouts.

Reference parity: the CommOp family + comm-type deduction
(/root/reference/hetu/graph/ops/Communication.cc:109-280 get_comm_type,
Communication.h:63-786). In this MI355X-native design the deduction happens
at op-construction time (the graph is SPMD-annotated up front) and compute()
issues the corresponding RCCL collective through parallel.comm.CommBackend —
there is no separate SubstituteCommOp pass; the "substitution" is the cached
deduction stored in op.attrs["kind"].

Deduction order mirrors the reference: equal -> identity; check_scatter ->
local slice (keep my shard); check_split -> local slice; check_allreduce ->
ALL_REDUCE; check_allgather -> ALL_GATHER; check_reducescatter ->
REDUCE_SCATTER; split->partial -> local zero-pad (adjoint of slice);
otherwise the generic gather+reslice redistribution (reduce partials,
allgather every split dim, slice to the destination layout — correct for
arbitrary pairs; exercised by tests/test_distributed_cpu.py reshard fuzz).
"""
from __future__ import annotations

from typing import Optional

import torch

from ...parallel.dstates import DistributedStates
from ..op import OpInterface
from ..tensor import Tensor, TensorMeta
from .basics import _g, _make


def _stable_dims_match(src: DistributedStates,
                       dst: DistributedStates) -> bool:
    """Per-device shard indices must agree on every tensor dim whose split
    count is unchanged.  The count-based predicates (check_*) ignore
    `order`, but order encodes the device->shard mapping: e.g.
    {0:2,1:2}[0,1] -> {0:2,1:2}[1,0] has equal counts yet every device
    owns a different shard — that transition needs the generic path."""
    dims = [d for d in set(list(src.states) + list(dst.states))
            if d >= 0 and src.get_dim(d) == dst.get_dim(d)
            and src.get_dim(d) > 1]
    if not dims:
        return True
    for i in range(src.device_num):
        si = src.map_device_to_state_index(i)
        di = dst.map_device_to_state_index(i)
        if any(si.get(d, 0) != di.get(d, 0) for d in dims):
            return False
    return True


def _slice_contained(src: DistributedStates,
                     dst: DistributedStates) -> bool:
    """For the no-comm slice kind: each device's dst shard must lie inside
    its src shard on every refined dim."""
    for i in range(src.device_num):
        si = src.map_device_to_state_index(i)
        di = dst.map_device_to_state_index(i)
        for d in dst.split_dims():
            ns, nd = src.get_dim(d), dst.get_dim(d)
            if nd > ns:
                r = nd // ns
                if not (si.get(d, 0) * r <= di.get(d, 0)
                        < (si.get(d, 0) + 1) * r):
                    return False
    return True


def _rs_positions_ok(src: DistributedStates, dst: DistributedStates,
                     d: int) -> bool:
    """reduce_scatter_tensor hands chunk i to group rank i: each device's
    dst split index must equal its position in its partial group."""
    for grp in src.group_devices_along(-2):
        for pos, i in enumerate(grp):
            if dst.map_device_to_state_index(i).get(d, 0) != pos:
                return False
    return True


def deduce_comm_kind(src: DistributedStates, dst: DistributedStates):
    """Returns (kind, info) where kind in {identity, slice, allreduce,
    allgather, reducescatter, zeropad, generic}.  Every fast kind also
    verifies the per-device placement it assumes (order-aware); when the
    counts fit but the device->shard mapping does not, the transition
    falls through to the generic gather+reslice path, which is correct
    for arbitrary layout pairs."""
    if src.check_equal(dst):
        return "identity", None
    stable = _stable_dims_match(src, dst)
    if src.check_allreduce(dst) and stable:
        return "allreduce", None
    # allgather: some split dim in src becomes dup in dst
    for d in src.split_dims():
        if src.get_dim(d) > dst.get_dim(d) and src.check_allgather(dst, d) \
                and stable:
            return "allgather", d
    # reduce-scatter: partial becomes a split dim
    for d in dst.split_dims():
        if dst.get_dim(d) > src.get_dim(d) \
                and src.check_reducescatter(dst, d) and stable \
                and _rs_positions_ok(src, dst, d):
            return "reducescatter", d
    # scatter / split: dup becomes split — keep local shard, no comm
    if (src.check_split(dst) or any(
            src.check_scatter(dst, d) for d in dst.split_dims())) \
            and stable and _slice_contained(src, dst):
        return "slice", None
    # adjoint of slice: split dim in src becomes partial in dst
    for d in src.split_dims():
        if src.get_dim(d) > dst.get_dim(d) and src._same_but(dst, d, -2) \
                and stable:
            return "zeropad", d
    return "generic", None


class CommOp(OpInterface):
    type = "Comm"
    is_comm = True

    def infer_meta(self, attrs, inputs):
        src_ds: DistributedStates = inputs[0].ds
        dst_ds: DistributedStates = attrs["dst_ds"]
        shape = list(inputs[0].shape)
        if src_ds is not None:
            # local shape changes with the layout
            gshape = src_ds.global_shape(shape)
            shape = list(dst_ds.local_shape(gshape))
        return [TensorMeta(shape, inputs[0].dtype)]

    def deduce_states(self, op):
        src = op.inputs[0].ds
        dst = op.attrs["dst_ds"]
        if src is None:
            # single-device graph: comm is identity
            op.attrs["kind"] = ("identity", None)
        else:
            op.attrs["kind"] = deduce_comm_kind(src, dst)
        op.outputs[0].ds = dst
        op.outputs[0].device_group = op.inputs[0].device_group

    def compute(self, op, inputs, ctx):
        x = inputs[0]
        kind, info = op.attrs["kind"]
        src: DistributedStates = op.inputs[0].ds
        dst: DistributedStates = op.attrs["dst_ds"]
        if kind == "identity" or ctx.comm is None:
            if kind in ("identity", "allreduce", "allgather"):
                return [x]
            # world_size 1 but layout changes shape: slice/zeropad still run
        dg = op.inputs[0].device_group
        my = _my_index(ctx, dg)
        if kind == "identity":
            return [x]
        if kind == "allreduce":
            ranks = _ranks(dg, src.group_devices_along(-2), my)
            return [ctx.comm.allreduce(x, ranks)]
        if kind == "allgather":
            d = info
            ranks = _ranks(dg, src.group_devices_along(d), my)
            return [ctx.comm.allgather(x, ranks, dim=d)]
        if kind == "reducescatter":
            d = info
            ranks = _ranks(dg, src.group_devices_along(-2), my)
            idx = dst.map_device_to_state_index(my).get(d, 0)
            return [ctx.comm.reducescatter(x, ranks, dim=d, my_index=idx)]
        if kind == "slice":
            gshape = src.global_shape(x.shape)
            src_sl = src.local_slice(gshape, my)
            dst_sl = dst.local_slice(gshape, my)
            out = x
            for dim in range(x.ndim):
                ds_, de_ = dst_sl[dim].start, dst_sl[dim].stop
                if ds_ is None:
                    continue
                ss_ = src_sl[dim].start or 0
                out = out.narrow(dim, ds_ - ss_, de_ - ds_)
            return [out.contiguous()]
        if kind == "zeropad":
            d = info
            gshape = list(src.global_shape(x.shape))
            full = list(x.shape)
            full[d] = gshape[d] // dst.get_dim(d) if dst.get_dim(d) > 1 \
                else gshape[d]
            out = torch.zeros(full, dtype=x.dtype, device=x.device)
            idx = src.map_device_to_state_index(my).get(d, 0)
            out.narrow(d, idx * x.shape[d], x.shape[d]).copy_(x)
            return [out]
        if kind == "generic":
            # total fallback for arbitrary layout transitions (reference
            # Communication.h falls back to gather+redistribute too):
            # reduce partial, gather every split dim to the full tensor,
            # then slice down to the destination layout.  Costs one full
            # materialization — fine for the rare resharding edges the
            # faster kinds do not cover.
            out = x
            if src.partial > 1:
                ranks = _ranks(dg, src.group_devices_along(-2), my)
                out = ctx.comm.allreduce(out, ranks)
            for d in src.split_dims():
                ranks = _ranks(dg, src.group_devices_along(d), my)
                out = ctx.comm.allgather(out, ranks, dim=d)
            gshape = tuple(out.shape)
            out = out[dst.local_slice(gshape, my)].contiguous()
            if dst.partial > 1 and \
                    dst.map_device_to_state_index(my).get(-2, 0) != 0:
                # adjoint-of-slice convention (like the zeropad kind): only
                # the partial-index-0 rank carries the value, the rest hold
                # zeros, so a later partial-sum reduction is exact instead
                # of overcounting by the partial factor.
                out = torch.zeros_like(out)
            return [out]
        raise NotImplementedError(kind)

    def gradient(self, op, g):
        src = op.inputs[0].ds
        if src is None:
            return [g[0]]
        # adjoint target: src ds with dup <-> partial swapped
        states = {}
        for d, n in src.states.items():
            states[-1 if d == -2 else -2 if d == -1 else d] = n
        order = [-1 if d == -2 else -2 if d == -1 else d for d in src.order]
        tgt = DistributedStates(src.device_num, states, order)
        gr = _g(op.outputs[0])
        return [_make(gr, CommOp(), [g[0]], {"dst_ds": tgt},
                      name=f"grad_{op.name}").output()]


def _my_index(ctx, device_group) -> int:
    """This rank's index within the tensor's device group."""
    if ctx.comm is None:
        return 0
    if device_group is None:
        return ctx.comm.rank
    return list(device_group).index(ctx.comm.rank)


def _ranks(device_group, groups, my_index):
    """Global ranks of the subgroup containing my_index."""
    for grp in groups:
        if my_index in grp:
            if device_group is None:
                return list(grp)
            return [device_group[j] for j in grp]
    return [my_index]


def make_comm(graph, x: Tensor, dst_ds: DistributedStates,
              name: str = "comm") -> Tensor:
    if x.ds is not None and x.ds.check_equal(dst_ds):
        return x
    return _make(graph, CommOp(), [x], {"dst_ds": dst_ds}, name=name).output()


class AllReduceDirectOp(OpInterface):
    """Explicit allreduce over a fixed rank list (used by vocab-parallel CE
    internals and tests)."""
    type = "AllReduceDirect"
    is_comm = True

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        if ctx.comm is None:
            return [inputs[0]]
        return [ctx.comm.allreduce(inputs[0], op.attrs["ranks"],
                                   op.attrs.get("op", "sum"))]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, AllReduceDirectOp(), [g[0]],
                      dict(op.attrs)).output()]
