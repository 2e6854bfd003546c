"""Optimizer update ops + Optimizer.minimize.

Reference parity: hetu/graph/optim/optimizer.h:13-118 (SGD, Adam ->
MakeAdamOp) and ops/optimizer_update.h:9-130; the fused Adam kernel is
ops/hip/optimizers.hip (reference Optimizers.cu:145 AdamCuda). Parameter
gradients arrive partial over the data-parallel dim and are reduced here via
a CommOp to the parameter's layout (the engine's fast path replaces this
with bucketed flat-buffer allreduce overlapped with backward).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from ..op import OpInterface
from ..tensor import Tensor, TensorMeta
from .basics import _make
from .comm import make_comm


class OptimizerUpdateOp(OpInterface):
    """Base: inputs [param, grad]; output: dummy scalar (dependency token)."""

    def infer_meta(self, attrs, inputs):
        return [TensorMeta([], torch.float32)]

    def deduce_states(self, op):
        op.outputs[0].ds = None
        op.outputs[0].device_group = op.inputs[0].device_group


class SGDStepOp(OptimizerUpdateOp):
    type = "SGDStep"

    def __init__(self):
        self.state: Dict = {}

    def compute(self, op, inputs, ctx):
        param, grad = inputs
        lr = op.attrs["lr"]
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
        param -= (lr * upd).to(param.dtype)
        return [torch.zeros((), device=param.device)]


class AdamStepOp(OptimizerUpdateOp):
    """Fused Adam with fp32 master weights + m/v states; updates the
    variable's storage in place (bf16/fp16 params re-materialized from the
    fp32 master every step, as the reference's transfer params do).

    hipGraph capture support: bias corrections flow host-pinned -> device
    tensor -> kernel pointer; a captured step re-reads the pinned buffer,
    which `set_replay_step` updates between replays."""
    type = "AdamStep"

    _instances: List["AdamStepOp"] = []

    def __init__(self):
        self.state: Dict = {}
        AdamStepOp._instances.append(self)

    @classmethod
    def set_replay_step(cls, step: int):
        """Update every instance's pinned bias-correction buffer for a
        graph replay at optimizer step `step` (1-based)."""
        for inst in cls._instances:
            st = inst.state
            if "bc_host" in st:
                b1, b2 = st["betas"]
                st["bc_host"][0] = 1.0 - b1 ** step
                st["bc_host"][1] = 1.0 - b2 ** step
                st["step"] = step

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
                st["bc_host"] = torch.empty(2, dtype=torch.float32,
                                            pin_memory=True)
                st["bc_dev"] = torch.empty(2, dtype=torch.float32,
                                           device=param.device)
        st["step"] += 1
        bc_dev = None
        if "bc_host" in st:
            st["bc_host"][0] = 1.0 - a["beta1"] ** st["step"]
            st["bc_host"][1] = 1.0 - a["beta2"] ** st["step"]
            st["bc_dev"].copy_(st["bc_host"], non_blocking=True)
            bc_dev = st["bc_dev"]
        out16 = param if param.dtype != torch.float32 else None
        F.adam_step(st["master"], grad, st["m"], st["v"],
                    a["lr"], a["beta1"], a["beta2"], a["eps"],
                    a.get("weight_decay", 0.0), st["step"], out16, bc_dev)
        if out16 is None:
            param.copy_(st["master"])
        return [torch.zeros((), device=param.device)]


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
        st["step"] += 1
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
        F.adam_step(st["master"], gshard, st["m"], st["v"],
                    a["lr"], a["beta1"], a["beta2"], a["eps"],
                    a.get("weight_decay", 0.0), st["step"], out16, None)
        if n > 1:
            full = ctx.comm.allgather(out16, ranks, dim=0)
        else:
            full = out16
        param.reshape(-1).copy_(full[:numel])
        return [torch.zeros((), device=param.device)]


class GroupOp(OpInterface):
    """Join node over update ops (reference ops/group.cc)."""
    type = "Group"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta([], torch.float32)]

    def deduce_states(self, op):
        op.outputs[0].ds = None

    def compute(self, op, inputs, ctx):
        dev = inputs[0].device if inputs else "cpu"
        return [torch.zeros((), device=dev)]


class Optimizer:
    def __init__(self, lr: float, zero: bool = False):
        self.lr = lr
        self.zero = zero     # ZeRO: shard optimizer states over dp
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
        updates = []
        for p, g in zip(params, grads):
            if g is None:
                continue
            # parameter-grad reduction: partial (over dp) -> param layout.
            # Under ZeRO the update op itself reduce-scatters the partial
            # grad (COMPUTE_OPTIMIZE_BRIDGE semantics), so no comm here.
            if (not self.zero and g.ds is not None and p.ds is not None
                    and not g.ds.check_equal(p.ds)):
                g = make_comm(graph, g, p.ds, name=f"grad_allreduce_{p.name}")
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
                 weight_decay: float = 0.0, zero: bool = False):
        super().__init__(lr, zero=zero)
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
