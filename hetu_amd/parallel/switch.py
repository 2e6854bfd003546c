"""Parallelism hot switching: live param/optimizer-state migration between
strategies (HotSPa, SOSP'24).

Reference parity: hetu/graph/switch_exec_graph.{h,cc} — SwitchParam pairs
src/dst shards by DistributedStates diff (:636), bucketizes into ParamBuffer
fragments and runs ONE batched isend/irecv group
(BufferBatchedIsendIrecvExec :628) on the switch stream; here the plan is
computed from the DS algebra's local_slice and executed as one RCCL batched
p2p group over xGMI (single-hop on the 8-GPU mesh), with local overlap
copied directly.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from .comm import CommBackend, comm_backend


def _overlap(a: Tuple[slice, ...], b: Tuple[slice, ...], shape
             ) -> Optional[Tuple[Tuple[slice, ...], ...]]:
    """Intersection of two shard slices; returns (global, in_a, in_b)."""
    g, ia, ib = [], [], []
    for sa, sb, n in zip(a, b, shape):
        lo_a, hi_a = sa.start or 0, sa.stop if sa.stop is not None else n
        lo_b, hi_b = sb.start or 0, sb.stop if sb.stop is not None else n
        lo, hi = max(lo_a, lo_b), min(hi_a, hi_b)
        if lo >= hi:
            return None
        g.append(slice(lo, hi))
        ia.append(slice(lo - lo_a, hi - lo_a))
        ib.append(slice(lo - lo_b, hi - lo_b))
    return tuple(g), tuple(ia), tuple(ib)


def _shard_slices(ds, device_group, global_shape):
    """Per-device-index local slice; dup replicas share a slice."""
    n = len(device_group) if device_group else 1
    out = []
    for i in range(n):
        if ds is None:
            out.append(tuple(slice(0, s) for s in global_shape))
        else:
            sl = ds.local_slice(global_shape, i)
            out.append(tuple(
                slice(s.start or 0, s.stop if s.stop is not None else dim)
                for s, dim in zip(sl, global_shape)))
    return out


def switch_params(plan: List[Dict], comm: Optional[CommBackend] = None):
    """Migrate a set of tensors between layouts with one batched p2p group.

    plan entries: {"src": torch.Tensor|None (my A shard), "dst":
    torch.Tensor (my B shard buffer, filled in place), "global_shape":
    tuple, "src_ds", "src_group", "dst_ds", "dst_group"}.

    Every rank calls this; for each (param, src shard s, dst shard d) with
    overlapping regions, the OWNER of s with dup index 0 sends to every
    owner of d (skipping self-copies, which run locally).
    """
    comm = comm or comm_backend()
    rank = comm.rank
    sends, recvs = [], []
    for ent in plan:
        gshape = tuple(ent["global_shape"])
        sg = ent.get("src_group") or [rank]
        dg = ent.get("dst_group") or [rank]
        s_sl = _shard_slices(ent.get("src_ds"), sg, gshape)
        d_sl = _shard_slices(ent.get("dst_ds"), dg, gshape)
        s_ds, d_ds = ent.get("src_ds"), ent.get("dst_ds")

        def dup_leader(ds, group, idx):
            """first device index holding the same shard (dup leader)."""
            if ds is None:
                return 0
            st = ds.map_device_to_state_index(idx)
            for j in range(len(group)):
                stj = ds.map_device_to_state_index(j)
                if all(stj.get(d, 0) == st.get(d, 0)
                       for d in ds.split_dims()):
                    return j
            return idx

        for di, drank in enumerate(dg):
            for si, srank in enumerate(sg):
                # only the dup leader of each src shard sends
                if dup_leader(s_ds, sg, si) != si:
                    continue
                ov = _overlap(s_sl[si], d_sl[di], gshape)
                if ov is None:
                    continue
                _, in_a, in_b = ov
                if srank == drank:
                    if srank == rank and ent["src"] is not None:
                        ent["dst"][in_b].copy_(ent["src"][in_a])
                    continue
                if srank == rank and ent["src"] is not None:
                    sends.append((ent["src"][in_a].contiguous(), drank))
                if drank == rank:
                    buf = torch.empty(
                        [s.stop - s.start for s in in_b],
                        dtype=ent["dst"].dtype, device=ent["dst"].device)
                    recvs.append((buf, srank, ent["dst"], in_b))
    # ---- ParamBuffer staging (reference switch_exec_graph.h:76-160):
    # all fragments headed to the same peer coalesce into ONE flat buffer
    # per direction, so the batched p2p moves few LARGE messages over the
    # xGMI links instead of one message per overlap region.  Fragment
    # order is the plan order, which is identical on both endpoints.
    # key by (peer, dtype): a mixed-precision plan (e.g. bf16 params +
    # fp32 master copies in one call) cannot share a flat buffer
    send_groups: Dict[Tuple[int, torch.dtype], List[torch.Tensor]] = {}
    for t, dst in sends:
        send_groups.setdefault((dst, t.dtype), []).append(t.reshape(-1))
    recv_groups: Dict[Tuple[int, torch.dtype], List] = {}
    for buf, src, dst_t, in_b in recvs:
        recv_groups.setdefault((src, buf.dtype), []).append(
            (buf, dst_t, in_b))
    ops = []
    flat_sends = {k: (torch.cat(ts) if len(ts) > 1 else ts[0])
                  for k, ts in send_groups.items()}
    flat_recvs = {}
    for k, items in recv_groups.items():
        n = sum(b.numel() for b, _, _ in items)
        flat_recvs[k] = torch.empty(
            n, dtype=items[0][0].dtype, device=items[0][0].device)
    for d, _ in sorted(flat_sends, key=lambda k: (k[0], str(k[1]))):
        ops.append(dist.P2POp(dist.isend,
                              flat_sends[(d, _)], d))
    for s, _ in sorted(flat_recvs, key=lambda k: (k[0], str(k[1]))):
        ops.append(dist.P2POp(dist.irecv, flat_recvs[(s, _)], s))
    if ops and dist.is_initialized():
        for r in dist.batch_isend_irecv(ops):
            r.wait()
    for k, items in recv_groups.items():
        flat = flat_recvs[k]
        off = 0
        for buf, dst_t, in_b in items:
            n = buf.numel()
            dst_t[in_b].copy_(flat[off:off + n].view(buf.shape))
            off += n


def plan_entries(pa, pb, src_t, dst_t):
    """Migration plan entries for one tensor pair — one entry, or one per
    fused section: a sectioned shard is [s0_loc|s1_loc|...], so each
    section is its own plainly-chunked dim-0 migration (reference qkv
    reorder, ht_safetensors.py:113)."""
    gshape = pb.ds.global_shape(tuple(pb.shape)) if pb.ds is not None \
        else tuple(pb.shape)
    secs = getattr(pb, "shard_sections", None) or \
        (getattr(pa, "shard_sections", None) if pa is not None else None)
    base = {
        "src_ds": pa.ds if pa is not None else None,
        "src_group": pa.device_group if pa is not None else None,
        "dst_ds": pb.ds, "dst_group": pb.device_group,
    }
    if not secs:
        return [dict(base, src=src_t, dst=dst_t, global_shape=gshape)]
    tp_a = pa.ds.get_dim(0) if (pa is not None and pa.ds) else 1
    tp_b = pb.ds.get_dim(0) if pb.ds is not None else 1
    out = []
    off_a = off_b = 0
    for s in secs:
        la, lb = s // tp_a, s // tp_b
        out.append(dict(
            base,
            src=(src_t[off_a:off_a + la] if src_t is not None else None),
            dst=dst_t[off_b:off_b + lb],
            global_shape=(s,) + gshape[1:]))
        off_a += la
        off_b += lb
    return out


def switch_graph_params(graph_a, graph_b,
                        comm: Optional[CommBackend] = None,
                        include_adam: bool = True):
    """Migrate all parameters (and Adam fp32 master/m/v when both sides
    use plain AdamStep) from graph A's layout to graph B's (the reference
    SWITCH_ORIGIN_PARAM_AND_OPTIMIZER level)."""
    comm = comm or comm_backend()
    by_name_a = {p.name.split(":")[0]: p for p in graph_a.parameters}
    plan = []
    entries = plan_entries

    for pb in graph_b.parameters:
        name = pb.name.split(":")[0]
        pa = by_name_a.get(name)
        plan.extend(entries(pa, pb,
                            pa.get_data() if pa is not None else None,
                            pb.get_data()))
    switch_params(plan, comm)
    if include_adam:
        def adam_states(g):
            out = {}
            for op in g.ops:
                if op.type == "AdamStep" and op.inputs:
                    out[op.inputs[0].name.split(":")[0]] = \
                        (op.inputs[0], op.interface)
            return out
        sa, sb = adam_states(graph_a), adam_states(graph_b)
        for name, (pb, ifb) in sb.items():
            if name not in sa:
                continue
            pa, ifa = sa[name]
            if "m" not in ifa.state:
                continue
            if "m" not in ifb.state:
                # initialize B's state buffers lazily at B's local shape
                master = pb.get_data().detach().float().clone()
                ifb.state["master"] = master
                ifb.state["m"] = torch.zeros_like(master)
                ifb.state["v"] = torch.zeros_like(master)
                ifb.state["step"] = 0
            st_plan = []
            for key in ("master", "m", "v"):
                st_plan.extend(entries(pa, pb, ifa.state[key],
                                       ifb.state[key]))
            switch_params(st_plan, comm)
            ifb.state["step"] = ifa.state["step"]
