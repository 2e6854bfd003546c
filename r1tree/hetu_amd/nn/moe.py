"""Mixture-of-Experts layer with expert parallelism.

Reference parity: HetuMoE — v1/python/hetu/layers/moe_layer.py (top-k gate
+ capacity dispatch + a2a), gates in v1/python/hetu/layers/gates/.  The
a2a (flat or hierarchical, HETU_AMD_MOE_NODE_SIZE) rides RCCL over xGMI;
experts run as batched GEMMs (hipBLASLt) over [El, P*C, h] buffers.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..graph.ops import api as ht
from ..graph.ops.moe_ops import MoECombineOp, MoEDispatchOp
from ..graph.ops.basics import _make
from . import init
from .module import Module
from .parallel import ParallelSpec


class MoEMLP(Module):
    """Gated expert FFN (gelu).  Experts sharded over the ep group
    (= the spec's full device group for now); gate weights replicated.

    gate_type (reference v1/python/hetu/layers/gates/ — 5 gate families):
      "topk"   — learned softmax top-k (k=2 == GShard top-2)
      "switch" — learned top-1 (Switch Transformer)
      "hash"   — static modulo-hash routing by token position (no gate
                 params; HashGate semantics for static-shape graphs)
      "random" — static pseudo-random token->expert assignment (BASE-like
                 balanced random routing, fixed at build time)
    """

    def __init__(self, hidden: int, ffn_hidden: int, num_experts: int,
                 spec: Optional[ParallelSpec] = None, k: int = 2,
                 capacity_factor: float = 1.25, dtype=torch.float32,
                 gate_type: str = "topk", name: str = "moe"):
        super().__init__()
        spec = spec or ParallelSpec()
        self.spec = spec
        self.E = num_experts
        assert gate_type in ("topk", "switch", "hash", "random")
        self.gate_type = gate_type
        if gate_type == "switch":
            k = 1
        elif gate_type in ("hash", "random"):
            k = 1
        self.k = k
        self.dtype = dtype
        self.name = name
        self.capacity_factor = capacity_factor
        self.hidden, self.ffn = hidden, ffn_hidden
        P = spec.num_devices
        assert num_experts % P == 0, "experts must divide ep group"
        El = num_experts // P
        me = spec.my_index()
        self.gate = None
        if gate_type in ("topk", "switch"):
            w_gate = init.normal((num_experts, hidden), std=0.02,
                                 dtype=dtype, name=f"{name}.gate")
            self.gate = ht.variable(w_gate, name=f"{name}.gate.weight",
                                    ds=spec.ds_weight_dup(),
                                    device_group=spec.device_group)
        self._static_probs = {}   # N -> probs variable (hash/random)
        w1 = init.normal((num_experts, hidden, ffn_hidden), std=0.02,
                         dtype=dtype, name=f"{name}.w1")
        w2 = init.normal((num_experts, ffn_hidden, hidden), std=0.02,
                         dtype=dtype, name=f"{name}.w2")
        # per-rank expert shards; labeled dup so the optimizer applies the
        # local grads without a comm (each rank owns its experts)
        self.w1 = ht.variable(w1[me * El:(me + 1) * El].contiguous(),
                              name=f"{name}.w1", ds=spec.ds_weight_dup(),
                              device_group=spec.device_group)
        self.w2 = ht.variable(w2[me * El:(me + 1) * El].contiguous(),
                              name=f"{name}.w2", ds=spec.ds_weight_dup(),
                              device_group=spec.device_group)

    def _static_gate(self, N: int):
        if N not in self._static_probs:
            if self.gate_type == "hash":
                assign = torch.arange(N) % self.E
            else:                                  # random (fixed at build)
                g = torch.Generator().manual_seed(hash((self.name, N))
                                                  & 0x7FFFFFFF)
                assign = torch.randperm(N, generator=g) % self.E
            probs = torch.zeros(N, self.E, dtype=self.dtype)
            probs[torch.arange(N), assign] = 1.0
            self._static_probs[N] = ht.variable(
                probs, name=f"{self.name}.static_gate_{N}",
                requires_grad=False, ds=self.spec.ds_weight_dup(),
                device_group=self.spec.device_group)
        return self._static_probs[N]

    def forward(self, x):
        """x: [N, h] tokens -> [N, h]."""
        spec = self.spec
        N = x.shape[0]
        P = spec.num_devices
        C = max(1, int(self.capacity_factor * self.k * N / self.E))
        if self.gate_type in ("topk", "switch"):
            logits = ht.linear(x, self.gate)      # [N, E]
            probs = ht.softmax(logits, dim=-1)
        else:
            probs = self._static_gate(int(N))
        g = x.graph
        attrs = {"experts": self.E, "capacity": C, "k": self.k,
                 "ep_ranks": list(spec.device_group) if P > 1 else None}
        disp = _make(g, MoEDispatchOp(), [x, probs], dict(attrs),
                     name="moe_dispatch")
        expert_in, wk, pos, slot_of = disp.outputs
        h1 = ht.bmm(expert_in, self.w1)           # [El, P*C, f]
        h1 = ht.gelu(h1)
        eo = ht.bmm(h1, self.w2)                  # [El, P*C, h]
        attrs["out_ds"] = x.ds
        y = _make(g, MoECombineOp(), [eo, wk, pos, slot_of], dict(attrs),
                  name="moe_combine").output()
        return y
