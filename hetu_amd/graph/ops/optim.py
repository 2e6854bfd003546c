"""Optimizer update ops + Optimizer.minimize.

Reference parity: hetu/graph/optim/optimizer.h:13-118 (SGD, Adam ->
MakeAdamOp) and ops/optimizer_update.h:9-130; the fused Adam kernel is
ops/hip/optimizers.hip (reference Optimizers.cu:145 AdamCuda). Parameter
gradients arrive partial over the data-parallel dim and are reduced here via
a CommOp to the parameter's layout (the engine's fast path replaces this
with bucketed flat-buffer allreduce overlapped with backward).
"""
from __future__ import annotations

import weakref
from typing import Dict, List, Optional

import torch

from ..op import OpInterface
from ..tensor import Tensor, TensorMeta
from .basics import _make
from .comm import make_comm


_ZERO_TOKENS: Dict = {}


def _zero_token(device) -> torch.Tensor:
    """Shared dummy scalar used as the dependency token of update ops —
    allocating a fresh zeros(()) per op per step put hundreds of tiny
    fill kernels in every captured step."""
    t = _ZERO_TOKENS.get(device)
    if t is None:
        t = torch.zeros((), device=device)
        _ZERO_TOKENS[device] = t
    return t


class OptimizerUpdateOp(OpInterface):
    """Base: inputs [param, grad]; output: dummy scalar (dependency token)."""

    def infer_meta(self, attrs, inputs):
        return [TensorMeta([], torch.float32)]

    def deduce_states(self, op):
        op.outputs[0].ds = None
        op.outputs[0].device_group = op.inputs[0].device_group


def _bc_refresh(st: Dict, a: Dict):
    """Write + upload the shared pinned bias-correction buffer when this
    step's (step, lr_scale) differs from what the buffer holds.  Tagged
    by VALUE, not a monotonic step: the buffer is shared process-wide, so
    a second graph built later starts at step 1 again and must overwrite
    a stale higher-step value (the old `sh[step] < step` check silently
    reused it)."""
    if "bc" not in st:
        return None
    sh = st["bc"]
    tag = (st["step"], AdamStepOp._lr_scale)
    if sh.get("written") != tag:
        sh["host"][0] = (1.0 - a["beta1"] ** st["step"]) \
            / AdamStepOp._lr_scale
        sh["host"][1] = 1.0 - a["beta2"] ** st["step"]
        sh["dev"].copy_(sh["host"], non_blocking=True)
        sh["written"] = tag
    return sh["dev"]


# device-resident LR multipliers (one per GPU): pinned host source +
# device scalar; captured update ops read the device scalar, so
# set_replay_step's host write + one H2D reaches every replay
_LR_SCALE_BUFS: Dict[int, Dict] = {}


def _lr_scale_dev(device) -> torch.Tensor:
    sh = _LR_SCALE_BUFS.get(device.index)
    if sh is None:
        sh = {"host": torch.ones(1, dtype=torch.float32, pin_memory=True),
              "dev": torch.ones(1, dtype=torch.float32, device=device)}
        _LR_SCALE_BUFS[device.index] = sh
    return sh["dev"]


class SGDStepOp(OptimizerUpdateOp):
    type = "SGDStep"

    def __init__(self):
        self.state: Dict = {}

    def compute(self, op, inputs, ctx):
        param, grad = inputs
        momentum = op.attrs.get("momentum", 0.0)
        if momentum > 0.0:
            buf = self.state.get("momentum_buffer")
            if buf is None:
                buf = torch.zeros_like(param, dtype=torch.float32)
                self.state["momentum_buffer"] = buf
            buf.mul_(momentum).add_(grad.float())
            upd = buf
        else:
            upd = grad.float()
        if param.is_cuda:
            # device-scalar multiplier: a captured step re-reads it each
            # replay (set_replay_step updates the pinned source)
            scale = _lr_scale_dev(param.device)
            param -= (op.attrs["lr"] * upd * scale).to(param.dtype)
        else:
            lr = op.attrs["lr"] * AdamStepOp._lr_scale
            param -= (lr * upd).to(param.dtype)
        return [_zero_token(param.device)]


class AdamStepOp(OptimizerUpdateOp):
    """Fused Adam with fp32 master weights + m/v states; updates the
    variable's storage in place (bf16/fp16 params re-materialized from the
    fp32 master every step, as the reference's transfer params do).

    hipGraph capture support: bias corrections flow host-pinned -> device
    tensor -> kernel pointer; a captured step re-reads the pinned buffer,
    which `set_replay_step` updates between replays.  The (beta1, beta2)
    corrections are identical for every parameter, so ONE shared pinned
    buffer + ONE H2D memcpy per step serves all ~400 Adam ops (per-op
    copies put hundreds of tiny host-memcpy nodes in the hipGraph)."""
    type = "AdamStep"

    # weak registry: a discarded graph (hot-switch pool eviction, elastic
    # rebuild, tests) must release its op interfaces AND their fp32
    # master/m/v states — a strong list here leaked ~12 bytes/param per
    # dead graph build
    _instances: "weakref.WeakSet[AdamStepOp]" = weakref.WeakSet()
    # (beta1, beta2, device) -> {"host": pinned[2], "dev": cuda[2], "step"}
    _shared_bc: Dict = {}

    def __init__(self):
        self.state: Dict = {}
        AdamStepOp._instances.add(self)

    # global LR multiplier (schedules under hipGraph capture): the update
    # is lr * (m/bc1) / (sqrt(v/bc2)+eps), so scaling bc1 by 1/s scales
    # the WHOLE update by s exactly — the kernel's baked lr never changes,
    # the scale rides the same pinned bc buffer the replay re-reads.
    _lr_scale: float = 1.0

    @classmethod
    def set_lr_scale(cls, scale: float):
        cls._lr_scale = float(scale)
        for sh in _LR_SCALE_BUFS.values():
            sh["host"][0] = cls._lr_scale
            sh["dev"].copy_(sh["host"], non_blocking=True)

    @classmethod
    def set_replay_step(cls, step: int, lr_scale: Optional[float] = None):
        """Update the shared pinned bias-correction buffers for a graph
        replay at optimizer step `step` (1-based); lr_scale (if given)
        also updates the global LR multiplier."""
        if lr_scale is not None:
            cls.set_lr_scale(lr_scale)
        for (b1, b2, _dev), sh in cls._shared_bc.items():
            sh["host"][0] = (1.0 - b1 ** step) / cls._lr_scale
            sh["host"][1] = 1.0 - b2 ** step
            sh["written"] = (step, cls._lr_scale)
        for inst in cls._instances:
            if inst.state:
                inst.state["step"] = step

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        param, grad = inputs
        a = op.attrs
        st = self.state
        if "m" not in st:
            st["master"] = param.detach().float().clone()
            st["m"] = torch.zeros_like(st["master"])
            st["v"] = torch.zeros_like(st["master"])
            st["step"] = 0
            st["betas"] = (a["beta1"], a["beta2"])
            if param.is_cuda:
                key = (a["beta1"], a["beta2"], param.device.index)
                sh = AdamStepOp._shared_bc.get(key)
                if sh is None:
                    sh = {"host": torch.empty(2, dtype=torch.float32,
                                              pin_memory=True),
                          "dev": torch.empty(2, dtype=torch.float32,
                                             device=param.device),
                          "written": None}
                    AdamStepOp._shared_bc[key] = sh
                st["bc"] = sh
        st["step"] += 1
        bc_dev = _bc_refresh(st, a)
        out16 = param if param.dtype != torch.float32 else None
        lr = a["lr"] if bc_dev is not None \
            else a["lr"] * AdamStepOp._lr_scale
        F.adam_step(st["master"], grad, st["m"], st["v"],
                    lr, a["beta1"], a["beta2"], a["eps"],
                    a.get("weight_decay", 0.0), st["step"], out16, bc_dev)
        if out16 is None:
            param.copy_(st["master"])
        return [_zero_token(param.device)]


class ZeroAdamStepOp(OptimizerUpdateOp):
    """ZeRO-1/2 Adam: optimizer states sharded over the data-parallel group
    (reference: `zero` flag in ds configs + SplitReduceScatter /
    SplitAllGather bridge ops, hetu/graph/ops/Communication.h:660-786,
    subgraph.h:19-24 OPTIMIZE_COMPUTE_BRIDGE).

    compute(): grad -> reduce-scatter over the dp group -> fused Adam on the
    LOCAL shard (fp32 master/m/v only for 1/dp of the param) -> all-gather
    the updated bf16 shard back into the param storage.  Collectives ride
    RCCL; at world_size 1 it degrades to plain Adam."""
    type = "ZeroAdamStep"

    def __init__(self):
        self.state: Dict = {}
        AdamStepOp._instances.add(self)     # set_replay_step syncs "step"

    def _dp_ranks(self, op, ctx):
        p = op.inputs[0]
        if ctx.comm is None or p.ds is None or p.ds.dup <= 1:
            return [ctx.comm.rank if ctx.comm else 0]
        from .comm import _my_index, _ranks
        my = _my_index(ctx, p.device_group)
        return _ranks(p.device_group, p.ds.group_devices_along(-1), my)

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        param, grad = inputs
        a = op.attrs
        st = self.state
        ranks = self._dp_ranks(op, ctx)
        n = len(ranks)
        numel = param.numel()
        pad = (-numel) % n
        shard_elems = (numel + pad) // n
        my_idx = sorted(ranks).index(ctx.comm.rank) if ctx.comm and n > 1 \
            else 0
        if "m" not in st:
            flat = param.detach().float().reshape(-1)
            if pad:
                flat = torch.cat([flat, flat.new_zeros(pad)])
            st["master"] = flat[my_idx * shard_elems:(my_idx + 1)
                                * shard_elems].clone()
            st["m"] = torch.zeros_like(st["master"])
            st["v"] = torch.zeros_like(st["master"])
            st["step"] = 0
            st["pad"] = pad
            if param.is_cuda:
                # same pinned bias-correction buffer as AdamStepOp: a
                # hipGraph-captured ZeRO step re-reads corrections (and
                # the LR multiplier) each replay instead of baking them
                key = (a["beta1"], a["beta2"], param.device.index)
                sh = AdamStepOp._shared_bc.get(key)
                if sh is None:
                    sh = {"host": torch.empty(2, dtype=torch.float32,
                                              pin_memory=True),
                          "dev": torch.empty(2, dtype=torch.float32,
                                             device=param.device),
                          "written": None}
                    AdamStepOp._shared_bc[key] = sh
                st["bc"] = sh
        st["step"] += 1
        bc_dev = _bc_refresh(st, a)
        gflat = grad.reshape(-1)
        if pad:
            gflat = torch.cat([gflat, gflat.new_zeros(pad)])
        if n > 1:
            gshard = ctx.comm.reducescatter(gflat, ranks, dim=0,
                                            my_index=my_idx)
        else:
            gshard = gflat
        out16 = torch.empty(shard_elems, dtype=param.dtype,
                            device=param.device)
        lr = a["lr"] if bc_dev is not None \
            else a["lr"] * AdamStepOp._lr_scale
        F.adam_step(st["master"], gshard, st["m"], st["v"],
                    lr, a["beta1"], a["beta2"], a["eps"],
                    a.get("weight_decay", 0.0), st["step"], out16, bc_dev)
        if n > 1:
            full = ctx.comm.allgather(out16, ranks, dim=0)
        else:
            full = out16
        param.reshape(-1).copy_(full[:numel])
        return [_zero_token(param.device)]


class GradBucketOp(OpInterface):
    """Coalesced gradient all-reduce: N partial grads -> one flat-buffer
    RCCL all-reduce -> N reduced grads (reference AllReduceCoalesce,
    impl/communication/nccl_comm_group.cu:273-301).

    xGMI note: ring collectives are per-link bound (7 x ~153 GB/s), so few
    LARGE collectives beat many small ones; buckets are sized by
    HETU_AMD_BUCKET_MB (default 100).  The op carries attrs["_sched_key"]
    so topo_sort emits it right after the bucket's last gradient and the
    executor's comm-stream path overlaps it with the rest of backward."""
    type = "GradAllReduceBucket"
    is_comm = True

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(list(t.shape), t.dtype) for t in inputs]

    def deduce_states(self, op):
        for out, dst in zip(op.outputs, op.attrs["dst_dss"]):
            out.ds = dst
            out.device_group = op.inputs[0].device_group

    def compute(self, op, inputs, ctx):
        if ctx.comm is None:
            return list(inputs)
        from .comm import _my_index, _ranks
        src = op.inputs[0].ds
        dg = op.inputs[0].device_group
        my = _my_index(ctx, dg)
        ranks = _ranks(dg, src.group_devices_along(-2), my)
        if len(ranks) <= 1:
            return list(inputs)
        if len(inputs) == 1:
            return [ctx.comm.allreduce(inputs[0], ranks)]
        flat = torch._utils._flatten_dense_tensors(tuple(inputs))
        flat = ctx.comm.allreduce(flat, ranks)
        return list(torch._utils._unflatten_dense_tensors(flat,
                                                          tuple(inputs)))


def make_grad_buckets(graph, pairs, bucket_bytes: Optional[int] = None,
                      name_prefix: str = "grad_allreduce_bucket"):
    """pairs: [(param, partial_grad)] needing reduction to the param layout.
    Groups them into flat-buffer buckets (compatible comm group + dtype),
    ordered by backward readiness (reverse creation order), and returns
    {param_id: reduced_grad}."""
    import os as _os
    if bucket_bytes is None:
        bucket_bytes = int(_os.environ.get("HETU_AMD_BUCKET_MB", "100")) << 20
    # backward computes late-layer grads first; pairs arrive in forward
    # (parameter) order, so reversed order approximates readiness order
    ordered = list(reversed(pairs))
    buckets: Dict = {}
    for p, g in ordered:
        key = (str(g.ds), tuple(p.device_group or ()), g.dtype)
        nb = 1
        for s in g.shape:
            nb *= int(s)
        nb *= g.dtype.itemsize if hasattr(g.dtype, "itemsize") else \
            torch.empty(0, dtype=g.dtype).element_size()
        lst = buckets.setdefault(key, [[]])
        cur = lst[-1]
        cur_bytes = sum(b for _, _, b in cur)
        if cur and cur_bytes + nb > bucket_bytes:
            cur = []
            lst.append(cur)
        cur.append((p, g, nb))
    out: Dict[int, Tensor] = {}
    bi = 0
    for key, lst in buckets.items():
        for items in lst:
            ps = [p for p, _, _ in items]
            gs = [g for _, g, _ in items]
            attrs = {"dst_dss": [p.ds for p in ps],
                     "_sched_key": max(
                         (g.producer.id for g in gs
                          if g.producer is not None), default=0) + 0.5}
            op = _make(graph, GradBucketOp(), gs, attrs,
                       name=f"{name_prefix}{bi}")
            for p, t in zip(ps, op.outputs):
                out[p.id] = t
            bi += 1
    return out


class HeteroSyncOp(OpInterface):
    """Cross-pipeline split-allreduce of one parameter gradient (Malleus
    hetero DP; reference SplitAllReduce, Communication.h:660-786).  The
    shared HeteroGradSync plan lives in attrs; grads mutate in place."""
    type = "HeteroSync"
    is_comm = True

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(list(inputs[0].shape), inputs[0].dtype)]

    def deduce_states(self, op):
        op.outputs[0].ds = op.inputs[0].ds
        op.outputs[0].device_group = op.inputs[0].device_group

    def compute(self, op, inputs, ctx):
        hs = op.attrs["sync"]
        return [hs.sync_one(op.attrs["idx"], inputs[0])]


class GroupOp(OpInterface):
    """Join node over update ops (reference ops/group.cc)."""
    type = "Group"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta([], torch.float32)]

    def deduce_states(self, op):
        op.outputs[0].ds = None

    def compute(self, op, inputs, ctx):
        dev = inputs[0].device if inputs else torch.device("cpu")
        return [_zero_token(dev)]


class Optimizer:
    def __init__(self, lr: float, zero: bool = False, hetero=None):
        self.lr = lr
        self.zero = zero     # ZeRO: shard optimizer states over dp
        self.hetero = hetero  # parallel.hetero.HeteroSpec or None
        self.update_ops: List = []

    def _make_update(self, graph, param: Tensor, grad: Tensor) -> Tensor:
        raise NotImplementedError

    def minimize(self, loss: Tensor, params: Optional[List[Tensor]] = None
                 ) -> Tensor:
        graph = loss.graph
        params = params if params is not None else list(graph.parameters)
        watermark = len(graph.ops)
        has_scopes = any("_rc_scope" in op.attrs for op in graph.ops)
        grads = graph.gradients([loss], params)
        # parameter-grad reduction: partial (over dp) -> param layout, via
        # coalesced flat-buffer buckets overlapped with backward.  Under
        # ZeRO the update op itself reduce-scatters the partial grad
        # (COMPUTE_OPTIMIZE_BRIDGE semantics), so no comm here.
        pend = []
        for p, g in zip(params, grads):
            if (g is not None and not self.zero and g.ds is not None
                    and p.ds is not None and not g.ds.check_equal(p.ds)
                    and g.ds.check_allreduce(p.ds)):
                pend.append((p, g))
        reduced = make_grad_buckets(graph, pend) if pend else {}
        hs = None
        if self.hetero is not None and len(self.hetero.pipelines) > 1:
            from ...parallel.hetero import HeteroGradSync
            live = [p for p, g in zip(params, grads) if g is not None]
            hs = HeteroGradSync(self.hetero, live)
            hidx = {p.id: i for i, p in enumerate(live)}
        updates = []
        for p, g in zip(params, grads):
            if g is None:
                continue
            if p.id in reduced:
                g = reduced[p.id]
            elif (not self.zero and g.ds is not None and p.ds is not None
                    and not g.ds.check_equal(p.ds)):
                # non-allreduce reshard (rare): keep the per-tensor CommOp
                g = make_comm(graph, g, p.ds,
                              name=f"grad_allreduce_{p.name}")
            if hs is not None:
                sop = _make(graph, HeteroSyncOp(), [g],
                            {"sync": hs, "idx": hidx[p.id]},
                            name=f"hetero_sync_{p.name}")
                # chain syncs so every rank issues the cross-pipeline
                # collectives in the same (params) order — differently
                # shaped per-pipeline graphs would otherwise be free to
                # interleave them differently and deadlock
                if getattr(self, "_last_sync", None) is not None:
                    sop.in_deps.append(self._last_sync)
                self._last_sync = sop
                g = sop.output()
            updates.append(self._make_update(graph, p, g))
        self.update_ops = updates
        out = _make(graph, GroupOp(), updates, name="train_op").output()
        if has_scopes:
            # duplicate the marked forward subgraphs into the backward
            # (reference recompute.cc semantics, op granularity)
            graph.apply_recompute(watermark)
        return out


class SGD(Optimizer):
    def __init__(self, lr: float = 0.01, momentum: float = 0.0):
        super().__init__(lr)
        self.momentum = momentum

    def _make_update(self, graph, param, grad):
        return _make(graph, SGDStepOp(), [param, grad],
                     {"lr": self.lr, "momentum": self.momentum},
                     name=f"sgd_{param.name}").output()


class Adam(Optimizer):
    def __init__(self, lr: float = 1e-3, beta1: float = 0.9,
                 beta2: float = 0.999, eps: float = 1e-8,
                 weight_decay: float = 0.0, zero: bool = False,
                 hetero=None):
        super().__init__(lr, zero=zero, hetero=hetero)
        self.beta1, self.beta2 = beta1, beta2
        self.eps = eps
        self.weight_decay = weight_decay

    def _make_update(self, graph, param, grad):
        cls = ZeroAdamStepOp if self.zero else AdamStepOp
        return _make(graph, cls(), [param, grad],
                     {"lr": self.lr, "beta1": self.beta1,
                      "beta2": self.beta2, "eps": self.eps,
                      "weight_decay": self.weight_decay},
                     name=f"adam_{param.name}").output()
