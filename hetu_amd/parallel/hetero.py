"""Heterogeneous data parallelism across differently-parallelized pipelines
(Malleus).

Reference parity: DistributedStatesUnion + hetero_dim
(/root/reference/hetu/graph/distributed_states.h:158-233), pipeline
deduction over unions (define_and_run_graph.cc:638 DeducePipeline), and the
Split* grouped collectives (graph/ops/Communication.h:660-786
SplitAllReduce): each pipeline trains the SAME model under its OWN
ParallelSpec (e.g. tp2 on fast devices, tp1 on a straggler), processes a
speed-weighted share of the global batch, and parameter gradients reduce
ACROSS pipelines with split-allreduce groups that pair up the overlapping
shard regions of the different layouts.

MI355X-native design: no graph surgery — a HeteroSpec describes the
pipelines; `hetero_grad_sync` runs after each pipeline's own (bucketed)
grad reduction and before the optimizer, issuing one RCCL collective per
distinct shard-overlap region with one representative device per pipeline,
then broadcasting within each pipeline's duplicate group.
"""
from __future__ import annotations

import dataclasses
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from ..nn.parallel import ParallelSpec
from .comm import CommBackend, comm_backend
from .dstates import DistributedStatesUnion


@dataclasses.dataclass
class HeteroSpec:
    """K pipelines over disjoint device groups; weights are each pipeline's
    share of the global batch (Malleus assigns them from straggler speeds:
    galvatron/searchers.py hetero_pipeline_partition)."""
    pipelines: List[ParallelSpec]
    weights: Optional[List[float]] = None      # default: uniform

    def __post_init__(self):
        if self.weights is None:
            self.weights = [1.0 / len(self.pipelines)] * len(self.pipelines)
        assert len(self.weights) == len(self.pipelines)
        groups = [tuple(p.device_group) for p in self.pipelines]
        flat = [r for g in groups for r in g]
        assert len(flat) == len(set(flat)), "pipelines must be disjoint"

    def my_pipeline(self, rank: Optional[int] = None) -> int:
        rank = comm_backend().rank if rank is None else rank
        for i, p in enumerate(self.pipelines):
            if rank in p.device_group:
                return i
        raise ValueError(f"rank {rank} not in any pipeline")

    def micro_batches(self, global_batch: int) -> List[int]:
        """Integer split of the global batch by weight (largest remainder)."""
        raw = [w * global_batch for w in self.weights]
        base = [int(r) for r in raw]
        rem = global_batch - sum(base)
        order = sorted(range(len(raw)), key=lambda i: base[i] - raw[i])
        for i in order[:rem]:
            base[i] += 1
        return base

    def param_union(self, p) -> DistributedStatesUnion:
        """The DS union describing parameter `p` across pipelines: entry i
        is p's layout under pipelines[i] (derived from the layer's
        hetero_split tag), lifted to the union's hetero (dup) dim."""
        from ..parallel.dstates import DistributedStates
        split = getattr(p, "hetero_split", None)
        dss = []
        for spec in self.pipelines:
            n = spec.num_devices
            if split is None:
                ds = DistributedStates(n, {-1: n} if n > 1 else {})
            else:
                dim, _ = split
                states = {-1: n // spec.tp, dim: spec.tp}
                ds = DistributedStates(
                    n, {d: c for d, c in states.items() if c > 1},
                    [-1, dim])
            dss.append(ds)
        if len(dss) == 1:
            return DistributedStatesUnion(dss)
        return DistributedStatesUnion(dss, hetero_dim=-1)


def _shard_regions(length: int, n: int, idx: int,
                   sections: Optional[Sequence[int]]
                   ) -> List[Tuple[int, int]]:
    """Global [start, stop) regions along the split dim owned by shard
    `idx` of `n`.  With `sections` (fused qkv / gate|up weights) every
    section is sharded independently (Megatron stride sharding), so the
    shard owns one region per section."""
    if sections is None:
        blk = length // n
        return [(idx * blk, (idx + 1) * blk)]
    out = []
    base = 0
    for s in sections:
        blk = s // n
        out.append((base + idx * blk, base + (idx + 1) * blk))
        base += s
    return out


def _local_offset(regions: List[Tuple[int, int]], g: int) -> int:
    """Offset of global index g within the concatenated local shard."""
    off = 0
    for a, b in regions:
        if a <= g < b:
            return off + (g - a)
        off += b - a
    raise ValueError(f"{g} not in {regions}")


class HeteroGradSync:
    """Cross-pipeline split-allreduce of parameter gradients.

    For each param, the global extent of its split dim is cut at every
    pipeline's shard boundary; each resulting atom is owned by exactly one
    (representative) device per pipeline.  Atoms sharing an owner set
    reduce in one collective; afterwards each pipeline broadcasts from its
    representative to its duplicate replicas.  Duplicated params skip the
    atom algebra: one allreduce over the K representatives."""

    def __init__(self, spec: HeteroSpec, params: Sequence,
                 comm: Optional[CommBackend] = None):
        self.spec = spec
        self.comm = comm or comm_backend()
        self.rank = self.comm.rank
        self.pi = spec.my_pipeline(self.rank)
        self.my_spec = spec.pipelines[self.pi]
        self.plans = [self._plan(p) for p in params]
        # eagerly create every comm group ANY rank's sync will use, in the
        # same order on every rank (see CommBackend.ensure_groups) — the
        # union must be computed from the spec, not from this rank's view
        groups = []
        for plan in self.plans:
            if plan is None:
                continue
            if plan["kind"] == "dup":
                groups.append(plan["reps"])
            else:
                groups.extend(grp["reps"] for grp in plan["groups"])
        for sp in spec.pipelines:
            groups.append(list(sp.device_group))       # dup-param broadcast
            for t in range(sp.tp):                     # split-param dups
                _, dups = self._rep_and_dup(sp, t)
                groups.append(dups)
        self.comm.ensure_groups(groups)

    # ---- plan construction (once) ---------------------------------------
    def _rep_and_dup(self, spec: ParallelSpec, tp_idx: int):
        """Representative rank (dp index 0) and its dup group for one tp
        shard of a pipeline."""
        dg = spec.device_group
        tp = spec.tp
        reps = dg[tp_idx]                       # dp index 0, cp index 0
        dups = [dg[d * tp + tp_idx] for d in range(len(dg) // tp)]
        return reps, dups

    def _plan(self, p):
        split = getattr(p, "hetero_split", None)
        spec = self.spec
        K = len(spec.pipelines)
        if K == 1:
            return None
        w = [float(x) for x in spec.weights]
        if split is None:
            # duplicated param: reduce across one representative per
            # pipeline, broadcast within my pipeline's replicas
            reps = [sp.device_group[0] for sp in spec.pipelines]
            dups = list(self.my_spec.device_group)
            return {"kind": "dup", "reps": reps, "dups": dups, "w": w}
        dim, sections = split
        # p.shape is the LOCAL shard shape; global length along the split
        # dim = local * tp (sections, when present, are the GLOBAL fused
        # block sizes, e.g. [h, h, h] for qkv)
        if sections is not None:
            secs = list(sections)
            glen = sum(secs)
        else:
            secs = None
            glen = int(p.shape[dim]) * self.my_spec.tp
        # atom boundaries: every pipeline's shard boundaries
        cuts = {0, glen}
        per_pipe_regions = []
        for sp in spec.pipelines:
            regs = [_shard_regions(glen, sp.tp, t, secs)
                    for t in range(sp.tp)]
            per_pipe_regions.append(regs)
            for shard in regs:
                for a, b in shard:
                    cuts.add(a)
                    cuts.add(b)
        cuts = sorted(cuts)
        my_tp_idx = self.my_spec.my_tp_index()
        my_regions = per_pipe_regions[self.pi][my_tp_idx]
        # group atoms by owner set
        groups: Dict[tuple, List[Tuple[int, int]]] = {}
        for a, b in zip(cuts[:-1], cuts[1:]):
            owners = []
            for k, sp in enumerate(spec.pipelines):
                for t in range(sp.tp):
                    if any(x <= a and b <= y
                           for x, y in per_pipe_regions[k][t]):
                        owners.append((k, t))
                        break
            key = tuple(owners)
            groups.setdefault(key, []).append((a, b))
        plan_groups = []
        for owners, atoms in groups.items():
            reps = []
            for k, t in owners:
                r, _ = self._rep_and_dup(spec.pipelines[k], t)
                reps.append(r)
            mine = next(((k, t) for k, t in owners if k == self.pi
                         and t == my_tp_idx), None)
            local_atoms = None
            if mine is not None:
                local_atoms = [(_local_offset(my_regions, a), b - a)
                               for a, b in atoms]
            plan_groups.append({"reps": sorted(reps),
                                "local_atoms": local_atoms})
        _, my_dups = self._rep_and_dup(self.my_spec, my_tp_idx)
        return {"kind": "split", "dim": dim, "groups": plan_groups,
                "dups": my_dups, "w": w}

    # ---- execution (per step) -------------------------------------------
    def sync(self, grads: Sequence[torch.Tensor]):
        """In-place: grads become the weighted cross-pipeline sum.  Each
        pipeline's grad is its LOCAL batch mean; the result is the global
        batch mean: sum_k w_k * grad_k."""
        for i, g in enumerate(grads):
            self.sync_one(i, g)

    def sync_one(self, i: int, g: Optional[torch.Tensor]):
        plan = self.plans[i]
        if plan is None or g is None:
            return g
        wk = plan["w"][self.pi]
        g.mul_(wk)
        if plan["kind"] == "dup":
            if self.rank in plan["reps"] and len(plan["reps"]) > 1:
                self.comm.allreduce(g, plan["reps"])
            if len(plan["dups"]) > 1:
                self.comm.broadcast(g, plan["dups"][0], plan["dups"])
            return g
        dim = plan["dim"]
        gm = g.movedim(dim, 0).contiguous() if dim != 0 else g
        for grp in plan["groups"]:
            if grp["local_atoms"] is None or self.rank not in grp["reps"]:
                continue
            if len(grp["reps"]) > 1:
                for off, ln in grp["local_atoms"]:
                    piece = gm.narrow(0, off, ln).contiguous()
                    red = self.comm.allreduce(piece, grp["reps"])
                    gm.narrow(0, off, ln).copy_(red)
        if dim != 0:
            g.copy_(gm.movedim(0, dim))
        if len(plan["dups"]) > 1:
            self.comm.broadcast(g, plan["dups"][0], plan["dups"])
        return g
