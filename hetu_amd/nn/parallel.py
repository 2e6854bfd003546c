"""Tensor-/sequence-parallel layers expressed as DistributedStates + comm.

Reference parity: python/hetu/nn/modules/parallel_multi_ds.py
(HtMultiColumnParallelLinear:328, HtMultiRowParallelLinear:411,
HtMultiParallelLayerNorm:163, HtMultiParallelRMSNorm:89,
HtMultiVocabParallelEmbedding:268): TP/SP is expressed purely as layout
annotations plus `comm(tensor, dst_ds)`; the CommOp deduction inserts the
allgather/allreduce/reduce-scatter collectives, which ride RCCL over xGMI.

MI355X notes: with 8 fully-connected GPUs per node, TP allreduces are
single-hop; sequence parallelism replaces the two allreduces per block with
allgather + reduce-scatter of the same volume but sharded activations.
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional

import torch

from ..graph.ops import api as ht
from ..parallel.comm import comm_backend
from ..parallel.dstates import DistributedStates
from . import init
from .module import Module


@dataclasses.dataclass
class ParallelSpec:
    """A dp x cp x tp mesh over a flat device group (tp fastest-varying,
    then cp, then dp).

    device_group: global ranks, len == dp*cp*tp.  Encodes the reference's
    ds_parallel_config device_group + split/dup layout generation
    (utils/parallel/generate_ds.py:253); cp is the "dcp" context-parallel
    dim driving ring attention (engine/trainer.py:251-260)."""
    dp: int = 1
    tp: int = 1
    cp: int = 1
    device_group: Optional[List[int]] = None
    sequence_parallel: bool = False

    def __post_init__(self):
        if self.device_group is None:
            self.device_group = list(range(self.dp * self.cp * self.tp))
        assert len(self.device_group) == self.dp * self.cp * self.tp

    @property
    def num_devices(self) -> int:
        return self.dp * self.cp * self.tp

    def my_index(self) -> int:
        rank = comm_backend().rank
        if rank in self.device_group:
            return self.device_group.index(rank)
        return 0

    def my_tp_index(self) -> int:
        return self.my_index() % self.tp

    def my_cp_index(self) -> int:
        return (self.my_index() // self.tp) % self.cp

    def my_dp_index(self) -> int:
        return self.my_index() // (self.tp * self.cp)

    # ---- layouts ---------------------------------------------------------
    def _ds(self, states, order):
        n = self.num_devices
        if n == 1:
            return None
        states = {d: c for d, c in states.items() if c > 1}
        order = [d for d in order if d in states]
        return DistributedStates(n, states, order)

    def ds_activation(self, batch_dim: int = 0, seq_dim: int = 1):
        """split(batch) over dp, split(seq) over cp, dup over tp."""
        if self.cp > 1:
            return self._ds({batch_dim: self.dp, seq_dim: self.cp,
                             -1: self.tp}, [batch_dim, seq_dim, -1])
        return self._ds({batch_dim: self.dp, -1: self.tp}, [batch_dim, -1])

    def ds_tokens(self, tok_dim: int = 0):
        """Layout of a flattened [B*S, ...] tensor: dp x cp fuse into one
        token split (order-insensitive for reductions), dup over tp."""
        return self._ds({tok_dim: self.dp * self.cp, -1: self.tp},
                        [tok_dim, -1])

    def ds_activation_sp(self, batch_dim: int = 0, seq_dim: int = 1):
        """split(batch) over dp, split(seq) over tp (sequence parallel)."""
        assert self.cp == 1, "sequence_parallel + cp not supported yet"
        return self._ds({batch_dim: self.dp, seq_dim: self.tp},
                        [batch_dim, seq_dim])

    def ds_weight_dup(self):
        return self._ds({-1: self.num_devices}, [-1])

    def ds_weight_col(self, split_dim: int = 0):
        """dup over dp x cp, split(out_features) over tp."""
        return self._ds({-1: self.dp * self.cp, split_dim: self.tp},
                        [-1, split_dim])

    def ds_weight_row(self, split_dim: int = 1):
        """dup over dp x cp, split(in_features) over tp."""
        return self._ds({-1: self.dp * self.cp, split_dim: self.tp},
                        [-1, split_dim])

    def ds_partial_tp(self, batch_dim: int = 0):
        """split(batch) over dp, partial over tp (row-parallel output)."""
        return self._ds({batch_dim: self.dp * self.cp, -2: self.tp},
                        [batch_dim, -2])

    def cp_ranks(self) -> List[int]:
        """Global ranks of this rank's cp ring (same dp and tp coords)."""
        di, ti = self.my_dp_index(), self.my_tp_index()
        return [self.device_group[(di * self.cp + c) * self.tp + ti]
                for c in range(self.cp)]


def _shard(data: torch.Tensor, dim: int, n: int, idx: int,
           sections=None) -> torch.Tensor:
    """Shard along `dim`.  For fused weights (qkv, gate|up) `sections`
    gives the sizes of the logically-separate blocks: each block is sharded
    independently and the local shards concatenated, so the local layout is
    [q_loc|k_loc|v_loc] / [gate_loc|up_loc] as the downstream fused kernels
    (attention reshape, swiglu) expect (Megatron 'stride' sharding)."""
    if n <= 1:
        return data
    if sections is None:
        return data.chunk(n, dim=dim)[idx].contiguous()
    parts = torch.split(data, list(sections), dim=dim)
    return torch.cat([p.chunk(n, dim=dim)[idx] for p in parts],
                     dim=dim).contiguous()


class ColumnParallelLinear(Module):
    """y = x @ W^T + b, W [out, in] split on out over tp.

    gather_output=False leaves y split on the last dim (feeds a row-parallel
    layer).  With sequence_parallel the input arrives seq-split and is
    allgathered here (parallel_multi_ds.py:389 comm-to-dup)."""

    def __init__(self, in_features: int, out_features: int,
                 spec: ParallelSpec, bias: bool = True,
                 gather_output: bool = False, dtype=torch.float32,
                 name: str = "col_linear", init_std: Optional[float] = None,
                 sections=None):
        super().__init__()
        self.spec = spec
        tp, ti = spec.tp, spec.my_tp_index()
        assert out_features % tp == 0
        if init_std is None:
            w = init.xavier_normal((out_features, in_features), dtype=dtype,
                                   name=f"{name}.weight")
        else:
            w = init.normal((out_features, in_features), std=init_std,
                            dtype=dtype, name=f"{name}.weight")
        self.sections = sections
        self.weight = ht.variable(_shard(w, 0, tp, ti, sections),
                                  name=f"{name}.weight",
                                  ds=spec.ds_weight_col(0),
                                  device_group=spec.device_group)
        if sections is not None and tp > 1:
            # checkpoint de-interleave metadata (utils/checkpoint)
            self.weight.shard_sections = list(sections)
        # cross-pipeline hetero grad sync metadata (parallel/hetero.py)
        self.weight.hetero_split = (0, list(sections) if sections else None)
        if bias:
            b = init.zeros((out_features,), dtype)
            self.bias = ht.variable(_shard(b, 0, tp, ti, sections),
                                    name=f"{name}.bias",
                                    ds=spec.ds_weight_col(0),
                                    device_group=spec.device_group)
            if sections is not None and tp > 1:
                self.bias.shard_sections = list(sections)
            self.bias.hetero_split = (0, list(sections) if sections
                                      else None)
        else:
            self.register_parameter("bias", None)
        self.gather_output = gather_output

    def forward(self, x):
        spec = self.spec
        if spec.sequence_parallel and spec.tp > 1 and x.ds is not None \
                and not x.ds.check_equal(spec.ds_activation(0)):
            # SP: seq-split -> dup over tp (allgather on the seq dim)
            x = ht.comm(x, spec.ds_activation(0), name="sp_allgather")
        y = ht.linear(x, self.weight, self.bias)
        if self.gather_output and spec.tp > 1:
            y = ht.comm(y, spec.ds_activation(0), name="col_gather")
        return y


class RowParallelLinear(Module):
    """y = x @ W^T + b, W [out, in] split on in over tp; x arrives split on
    the last dim; the partial output is allreduced (or reduce-scattered to
    seq-split under sequence parallelism — parallel_multi_ds.py:425)."""

    def __init__(self, in_features: int, out_features: int,
                 spec: ParallelSpec, bias: bool = True,
                 dtype=torch.float32, name: str = "row_linear",
                 init_std: Optional[float] = None):
        super().__init__()
        self.spec = spec
        tp, ti = spec.tp, spec.my_tp_index()
        assert in_features % tp == 0
        if init_std is None:
            w = init.xavier_normal((out_features, in_features), dtype=dtype,
                                   name=f"{name}.weight")
        else:
            w = init.normal((out_features, in_features), std=init_std,
                            dtype=dtype, name=f"{name}.weight")
        self.weight = ht.variable(_shard(w, 1, tp, ti),
                                  name=f"{name}.weight",
                                  ds=spec.ds_weight_row(1),
                                  device_group=spec.device_group)
        self.weight.hetero_split = (1, None)
        if bias:
            # bias is added AFTER the reduction; duplicated
            self.bias = ht.variable(init.zeros((out_features,), dtype),
                                    name=f"{name}.bias",
                                    ds=spec.ds_weight_dup(),
                                    device_group=spec.device_group)
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        spec = self.spec
        y = ht.linear(x, self.weight)       # partial over tp
        if spec.tp > 1:
            dst = (spec.ds_activation_sp(0, 1) if spec.sequence_parallel
                   else spec.ds_activation(0))
            y = ht.comm(y, dst, name="row_reduce")
        if self.bias is not None:
            y = ht.add(y, self.bias)
        return y


class VocabParallelEmbedding(Module):
    """Embedding with the vocab dim split over tp; the partial lookup is
    allreduced to dup (or reduce-scattered to seq-split under SP)."""

    def __init__(self, num_embeddings: int, embedding_dim: int,
                 spec: ParallelSpec, dtype=torch.float32,
                 name: str = "wte", init_std: float = 0.02):
        super().__init__()
        self.spec = spec
        self.vocab = num_embeddings
        tp, ti = spec.tp, spec.my_tp_index()
        assert num_embeddings % tp == 0
        w = init.normal((num_embeddings, embedding_dim), std=init_std,
                        dtype=dtype, name=f"{name}.weight")
        self.weight = ht.variable(_shard(w, 0, tp, ti),
                                  name=f"{name}.weight",
                                  ds=spec.ds_weight_col(0),
                                  device_group=spec.device_group)
        self.weight.hetero_split = (0, None)

    def forward(self, ids):
        spec = self.spec
        y = ht.vocab_parallel_embedding(self.weight, ids, self.vocab)
        if spec.tp > 1:
            dst = (spec.ds_activation_sp(0, 1) if spec.sequence_parallel
                   else spec.ds_activation(0))
            y = ht.comm(y, dst, name="vpe_reduce")
        return y


class ParallelLayerNorm(Module):
    """LayerNorm whose weights are duplicated across the mesh; operates on
    dup-over-tp or seq-split activations alike (row-wise op)."""

    def __init__(self, dim: int, spec: ParallelSpec, eps: float = 1e-5,
                 dtype=torch.float32, name: str = "ln"):
        super().__init__()
        self.eps = eps
        ds = spec.ds_weight_dup()
        self.weight = ht.variable(init.ones((dim,), dtype),
                                  name=f"{name}.weight", ds=ds,
                                  device_group=spec.device_group)
        self.bias = ht.variable(init.zeros((dim,), dtype),
                                name=f"{name}.bias", ds=ds,
                                device_group=spec.device_group)

    def forward(self, x):
        return ht.layer_norm(x, self.weight, self.bias, self.eps)


class ParallelRMSNorm(Module):
    def __init__(self, dim: int, spec: ParallelSpec, eps: float = 1e-6,
                 dtype=torch.float32, name: str = "rms"):
        super().__init__()
        self.eps = eps
        self.weight = ht.variable(init.ones((dim,), dtype),
                                  name=f"{name}.weight",
                                  ds=spec.ds_weight_dup(),
                                  device_group=spec.device_group)

    def forward(self, x):
        return ht.rms_norm(x, self.weight, self.eps)


def vocab_parallel_cross_entropy(logits, labels, vocab: int,
                                 ignore_index: int = -100):
    """Per-token loss from tp-sharded logits (reference
    VocabParallelCrossEntropyLoss.cc)."""
    return ht.vocab_parallel_cross_entropy(logits, labels, vocab,
                                           ignore_index)
