"""Operator protocol.

MI355X-native analogue of the reference's Operator/OpDef/OpInterface
(/root/reference/hetu/graph/operator.h:406-700). An OpInterface implements:

  * infer_meta(attrs, inputs)  -> list[TensorMeta]      (DoInferMeta)
  * deduce_states(op)          -> None (sets op.outputs[i].ds)  (DoDeduceStates)
  * compute(op, inputs, ctx)   -> list[torch.Tensor]    (DoCompute)
  * gradient(op, grad_outputs) -> list[Tensor|None]     (DoGradient)

compute() receives torch.Tensors and returns torch.Tensors; kernels dispatch
through hetu_amd.ops.functional which routes to the hand-written HIP/CDNA4
extension on GPU and to plain torch reference implementations on CPU.
"""
from __future__ import annotations

from typing import Dict, List, Optional

from .tensor import Tensor, TensorMeta


class OpInterface:
    type: str = "Op"
    # ops that communicate across ranks (used by scheduling / substitution)
    is_comm: bool = False

    def infer_meta(self, attrs: Dict, inputs: List[Tensor]) -> List[TensorMeta]:
        raise NotImplementedError

    def deduce_states(self, op: "Op") -> None:
        """Default SPMD propagation: outputs inherit the common ds of the
        non-pure-duplicate inputs (weights that are replicated don't change
        an activation's layout); ops with nontrivial layouts override."""
        dss = [t.ds for t in op.inputs if t.ds is not None]
        nontrivial = [d for d in dss if not d.is_pure_dup()]
        ds = None
        pick = nontrivial or dss
        if pick:
            ds = pick[0]
            for other in pick[1:]:
                if not other.check_equal(ds):
                    ds = None
                    break
        for out in op.outputs:
            if out.ds is None:
                out.ds = ds
        dgs = [t.device_group for t in op.inputs if t.device_group is not None]
        if dgs:
            for out in op.outputs:
                if out.device_group is None:
                    out.device_group = dgs[0]

    def compute(self, op: "Op", inputs, ctx):
        raise NotImplementedError

    def gradient(self, op: "Op", grad_outputs: List[Optional[Tensor]]
                 ) -> List[Optional[Tensor]]:
        return [None] * len(op.inputs)


class Op:
    _next_id = 0

    __slots__ = ("id", "interface", "inputs", "outputs", "attrs", "name",
                 "graph", "in_deps")

    def __init__(self, interface: OpInterface, inputs: List[Tensor],
                 attrs: Dict, name: str = "", graph=None):
        self.id = Op._next_id
        Op._next_id += 1
        self.interface = interface
        self.inputs = list(inputs)
        self.attrs = dict(attrs)
        self.name = name or f"{interface.type}_{self.id}"
        self.graph = graph
        self.outputs: List[Tensor] = []
        self.in_deps: List[Op] = []   # extra control dependencies

    @property
    def type(self):
        return self.interface.type

    def output(self, i: int = 0) -> Tensor:
        return self.outputs[i]

    def __repr__(self):
        return (f"Op({self.name}, inputs={[t.name for t in self.inputs]}, "
                f"outputs={[t.name for t in self.outputs]})")
