"""Layered JSON ds_parallel_config system.

Reference parity: python/hetu/utils/parallel/generate_ds.py:253
(generate_ds_parallel_config — per-module split/dup/device_group_union/type
entries, per-block recompute lists, top-level zero), read_ds.py:26
(config2ds — entry → DistributedStatesUnion + device groups) and ds_config.py.
The reference drives EVERY run off these files: a strategy is a
`tp_pp_list` (one (tp, pp) per data-parallel pipeline, possibly
heterogeneous) that is placed onto GPUs and expanded into a per-layer JSON
tree the graph builder consumes.

MI355X-native shape: the same JSON schema (so hand-written or
reference-generated configs interop) but the reader targets OUR layout
objects — `DistributedStates`/`DistributedStatesUnion` for tensor entries,
and `PipelineSpec`/`HeteroSpec` for the engine.  Placement assumes 8 GPUs
per MI355X node and keeps each tp group inside one node so its collectives
ride xGMI, letting pp cross nodes (p2p activations are the cheap,
infrequent traffic).
"""
from __future__ import annotations

import json
from typing import Dict, List, Optional, Sequence, Tuple, Union

from ..parallel.dstates import (NULL_HETERO_DIM, DistributedStates,
                                DistributedStatesUnion)

GPUS_PER_NODE = 8


# --------------------------------------------------------------------------
# strategy placement: tp_pp_list -> per-layer tp groups
# --------------------------------------------------------------------------
def convert_strategy(tp_pp_list: Sequence[Tuple[int, int]], ngpus: int,
                     layers: int, gpus_per_node: int = GPUS_PER_NODE,
                     hetero_layers: Optional[Sequence[Sequence[int]]] = None,
                     rank_map: Optional[Dict[int, int]] = None
                     ) -> Tuple[List[List[List[int]]], Dict[int, Tuple[int, int]]]:
    """Place K=(len list) pipelines onto `ngpus` devices.

    Returns (layers_tp_groups, gpu_pos): layers_tp_groups[l] is the list of
    tp rank-groups (one per pipeline) that own layer l; gpu_pos maps a
    global rank -> (pipeline index, stage index).

    Placement rule: each stage's tp group must sit inside one node
    (tp<=gpus_per_node, or tp a multiple of whole nodes); stages are
    packed best-fit-decreasing so big tp groups claim fresh nodes first.

    hetero_layers (reference generate_gpt_3d_config hetero_layers):
    pipeline k's per-stage layer COUNTS (sum == layers) for non-uniform
    splits — the Malleus hetero-pp partitioner's output.  rank_map
    (reference rank_to_device_mapping/unused_rank): logical->physical
    rank relabeling applied to every group (physical ranks absent from
    the map's values are simply unused).
    """
    assert ngpus >= sum(tp * pp for tp, pp in tp_pp_list), \
        f"need {sum(tp * pp for tp, pp in tp_pp_list)} gpus, have {ngpus}"
    nnodes = max(1, (ngpus + gpus_per_node - 1) // gpus_per_node)
    free = {n: list(range(n * gpus_per_node,
                          min((n + 1) * gpus_per_node, ngpus)))
            for n in range(nnodes)}
    layers_tp_groups: List[List[List[int]]] = [[] for _ in range(layers)]
    gpu_pos: Dict[int, Tuple[int, int]] = {}
    # place big-tp pipelines first so whole nodes are still available
    order = sorted(range(len(tp_pp_list)),
                   key=lambda i: -tp_pp_list[i][0])
    stage_groups: Dict[int, List[List[int]]] = {}
    for k in order:
        tp, pp = tp_pp_list[k]
        groups = []
        for _ in range(pp):
            if tp > gpus_per_node:
                assert tp % gpus_per_node == 0, \
                    f"tp {tp} must be a multiple of {gpus_per_node}"
                grp: List[int] = []
                for n in sorted(free, key=lambda n: -len(free[n])):
                    if len(free[n]) == gpus_per_node:
                        grp.extend(free[n])
                        free[n] = []
                        if len(grp) == tp:
                            break
                assert len(grp) == tp, f"cannot place tp {tp}"
            else:
                # best-fit: fullest node that still has room
                cands = [n for n in free if len(free[n]) >= tp]
                assert cands, f"cannot place tp {tp}"
                n = min(cands, key=lambda n: (len(free[n]), n))
                grp, free[n] = free[n][:tp], free[n][tp:]
            groups.append(grp)
        stage_groups[k] = groups
    if hetero_layers is not None:
        assert len(hetero_layers) == len(tp_pp_list)
        for k, (tp, pp) in enumerate(tp_pp_list):
            assert len(hetero_layers[k]) == pp \
                and sum(hetero_layers[k]) == layers, \
                f"pipeline {k}: stage layer counts must sum to {layers}"
    for k, (tp, pp) in enumerate(tp_pp_list):
        groups = stage_groups[k]
        for s, grp in enumerate(groups):
            for r in grp:
                gpu_pos[r] = (k, s)
        if hetero_layers is not None:
            counts = list(hetero_layers[k])
        else:
            per = layers // pp
            rem = layers % pp
            counts = [per + (1 if s < rem else 0) for s in range(pp)]
        lo = 0
        for s in range(pp):
            hi = lo + counts[s]
            for l in range(lo, hi):
                layers_tp_groups[l].append(groups[s])
            lo = hi
    if rank_map is not None:
        layers_tp_groups = [[[rank_map.get(r, r) for r in grp]
                             for grp in layer]
                            for layer in layers_tp_groups]
        gpu_pos = {rank_map.get(r, r): pos for r, pos in gpu_pos.items()}
    return layers_tp_groups, gpu_pos


# --------------------------------------------------------------------------
# generation: strategy -> layered JSON
# --------------------------------------------------------------------------
def _entry(split: Dict[int, List[int]], dup: List[int],
           dgu: List[List[int]], type_: str) -> Dict:
    return {"split": {str(d): list(v) for d, v in split.items()},
            "dup": list(dup), "device_group_union": [list(g) for g in dgu],
            "type": type_}


def generate_ds_parallel_config(
        tp_pp_list: Sequence[Tuple[int, int]], num_layers: int,
        ngpus: Optional[int] = None, zero: bool = True,
        model_key: str = "gpt", recompute_layers: Sequence[int] = (),
        gpus_per_node: int = GPUS_PER_NODE,
        hetero_layers: Optional[Sequence[Sequence[int]]] = None,
        rank_map: Optional[Dict[int, int]] = None) -> Dict:
    """Expand a (possibly heterogeneous) tp_pp_list into the layered JSON
    tree (reference generate_ds.py:253 layout: input / <model>{wte, wpe,
    blocks{...}, norm_final} / lm_head / label).  hetero_layers/rank_map:
    non-uniform per-pipeline stage splits and rank relabeling (reference
    parallel_config.py generate_gpt_3d_config)."""
    if ngpus is None:
        ngpus = sum(tp * pp for tp, pp in tp_pp_list)
    ltg, _ = convert_strategy(tp_pp_list, ngpus, num_layers, gpus_per_node,
                              hetero_layers=hetero_layers,
                              rank_map=rank_map)
    dp = len(tp_pp_list)
    if dp == 1:
        zero = False
    dp_union = [dp] * dp
    tp_u = [[len(g) for g in layer] for layer in ltg]    # [layer][pipe]
    used = sorted({r for layer in ltg for grp in layer for r in grp})
    cfg: Dict = {
        "zero": zero,
        "devices": used,
        "input": _entry({0: dp_union}, tp_u[0], ltg[0], "placeholder"),
        model_key: {
            "wte": _entry({0: tp_u[0]}, dp_union, ltg[0], "variable"),
            "wpe": _entry({}, [tp_u[0][i] * dp for i in range(dp)],
                          ltg[0], "variable"),
            "blocks": {},
            "norm_final": _entry({0: tp_u[-1]}, dp_union, ltg[-1],
                                 "variable"),
        },
        "lm_head": _entry({1: tp_u[-1]}, dp_union, ltg[-1], "variable"),
        "label": _entry({0: dp_union}, tp_u[-1], ltg[-1], "placeholder"),
    }
    blocks = cfg[model_key]["blocks"]
    for l in range(num_layers):
        blocks[f"blocks{l}"] = {
            "range": [l],
            "recompute": [l in recompute_layers] * dp,
            "norm1": _entry({0: tp_u[l]}, dp_union, ltg[l], "variable"),
            "attn": {
                "qkv": _entry({1: tp_u[l]}, dp_union, ltg[l], "variable"),
                "dense": _entry({0: tp_u[l]}, dp_union, ltg[l],
                                "variable"),
            },
            "norm2": _entry({0: tp_u[l]}, dp_union, ltg[l], "variable"),
            "mlp": {
                "dense_h_to_4h": _entry({1: tp_u[l]}, dp_union, ltg[l],
                                        "variable"),
                "dense_4h_to_h": _entry({0: tp_u[l]}, dp_union, ltg[l],
                                        "variable"),
            },
        }
    return cfg


def write_ds_parallel_config(cfg: Dict, path: str) -> None:
    with open(path, "w") as f:
        json.dump(cfg, f, indent=2)


# --------------------------------------------------------------------------
# reading: JSON -> DistributedStatesUnion / engine specs
# --------------------------------------------------------------------------
def config2ds(entry: Dict) -> Tuple[DistributedStatesUnion, List[List[int]]]:
    """One leaf entry -> (union, device_group_union).  Placeholders are
    hetero along dim 0 (each pipeline reads its batch slice), variables
    along -1 (each pipeline holds a duplicate copy); a single-pipeline
    entry is homogeneous (reference read_ds.py:26 config2ds)."""
    t = entry["type"]
    if t == "placeholder":
        hetero_dim = 0
    elif t == "variable":
        hetero_dim = -1
    else:
        raise ValueError(f"unsupported entry type {t!r}")
    dgu = entry["device_group_union"]
    K = len(dgu)
    if K == 1:
        hetero_dim = NULL_HETERO_DIM
    ds_list = []
    for k in range(K):
        # each union entry is a full-count ds INCLUDING the union factor
        # on the hetero dim (get_local strips it)
        n = len(dgu[k]) * K
        split = {int(d): v[k] for d, v in entry["split"].items()}
        states = {-1: entry["dup"][k], **split}
        if t == "placeholder":
            order = sorted(split) + [-1]
        else:
            order = [-1] + sorted(split)
        ds_list.append(DistributedStates(
            n, {d: c for d, c in states.items() if c > 1},
            [d for d in order if states.get(d, 1) > 1],
            zero=bool(entry.get("zero", False))))
    return DistributedStatesUnion(ds_list, hetero_dim), \
        [list(g) for g in dgu]


def _is_leaf(node: Dict) -> bool:
    return isinstance(node, dict) and "type" in node \
        and "device_group_union" in node


def read_ds_parallel_config(src: Union[str, Dict]
                            ) -> Dict[str, Tuple[DistributedStatesUnion,
                                                 List[List[int]]]]:
    """Load a config (path or dict) and convert every leaf entry.
    Returns {dotted.module.path: (union, device_group_union)}; the
    top-level `zero` flag is spread into variable entries first
    (reference read_ds.py config_spread_zero)."""
    cfg = src
    if isinstance(src, str):
        with open(src) as f:
            cfg = json.load(f)
    zero = bool(cfg.get("zero", False))
    out: Dict[str, Tuple[DistributedStatesUnion, List[List[int]]]] = {}

    def walk(node, path):
        if _is_leaf(node):
            if node["type"] == "variable" and "zero" not in node:
                node = dict(node, zero=zero)
            out[path] = config2ds(node)
            return
        if isinstance(node, dict):
            for k, v in node.items():
                if isinstance(v, dict):
                    walk(v, f"{path}.{k}" if path else k)

    walk(cfg, "")
    return out


def strategy_to_ds_config(strategy, num_layers: int,
                          model_key: str = "gpt",
                          gpus_per_node: int = GPUS_PER_NODE) -> Dict:
    """Expand a Galvatron search result (galvatron.cost_model.Strategy:
    dp/tp/pp/zero/recompute_layers) into the layered JSON tree — the
    auto-parallel output becomes a runnable, inspectable config file
    (reference flow: search -> generate_ds -> train)."""
    rc = list(range(int(getattr(strategy, "recompute_layers", 0) or 0)))
    return generate_ds_parallel_config(
        [(strategy.tp, strategy.pp)] * strategy.dp, num_layers,
        zero=bool(getattr(strategy, "zero", False)), model_key=model_key,
        recompute_layers=rc, gpus_per_node=gpus_per_node)


def recompute_layers_from_config(cfg: Union[str, Dict],
                                 model_key: Optional[str] = None,
                                 pipeline: int = 0) -> List[int]:
    """Layer indices flagged for activation recompute in `pipeline`'s
    column of the per-block recompute lists (reference generate_ds.py
    per-block `recompute: [bool]*dp`)."""
    if isinstance(cfg, str):
        with open(cfg) as f:
            cfg = json.load(f)
    if model_key is None:
        skip = {"input", "lm_head", "label", "zero", "devices"}
        model_key = next(k for k, v in cfg.items()
                         if k not in skip and isinstance(v, dict))
    out = []
    for blk in cfg[model_key]["blocks"].values():
        flags = blk.get("recompute", [])
        if pipeline < len(flags) and flags[pipeline]:
            out.extend(blk["range"])
    return sorted(out)


def strategy_from_config(cfg: Union[str, Dict],
                         model_key: Optional[str] = None):
    """Recover the engine-level strategy from a config tree.

    Homogeneous single-pipeline configs -> PipelineSpec(pp, dp=1, tp);
    multi-pipeline (hetero dp) -> HeteroSpec of per-pipeline ParallelSpecs
    (pp==1 pipelines only — my HeteroSpec composes with PipelineSpec at
    the engine layer, not here).  Also returns the per-pipeline stage
    rank-groups for pp>1 callers: (spec, stages) where
    stages[k] = ordered distinct tp groups of pipeline k."""
    if isinstance(cfg, str):
        with open(cfg) as f:
            cfg = json.load(f)
    if model_key is None:
        skip = {"input", "lm_head", "label", "zero", "devices"}
        model_key = next(k for k, v in cfg.items()
                         if k not in skip and isinstance(v, dict))
    blocks = cfg[model_key]["blocks"]
    n_layers = sum(len(b["range"]) for b in blocks.values())
    K = len(cfg["input"]["device_group_union"])
    # ordered distinct tp groups per pipeline = that pipeline's stages
    stages: List[List[List[int]]] = [[] for _ in range(K)]
    for l in range(n_layers):
        blk = next(b for b in blocks.values() if l in b["range"])
        dgu = blk["attn"]["qkv"]["device_group_union"]
        for k in range(K):
            g = list(dgu[k])
            if not stages[k] or stages[k][-1] != g:
                stages[k].append(g)
    from ..nn.parallel import ParallelSpec
    from ..parallel.hetero import HeteroSpec
    from ..parallel.pipeline import PipelineSpec
    if K == 1:
        pp = len(stages[0])
        tp = len(stages[0][0])
        return PipelineSpec(pp=pp, dp=1, tp=tp), stages
    shapes = {(len(s), len(s[0])) for s in stages}
    if len(shapes) == 1:
        # identical pipelines: a plain homogeneous pp x dp x tp world
        (pp, tp), = shapes
        return PipelineSpec(pp=pp, dp=K, tp=tp), stages
    pipes = []
    for k in range(K):
        assert len(stages[k]) == 1, \
            "hetero pipelines with pp>1: use the stages return value"
        pipes.append(ParallelSpec(dp=1, tp=len(stages[k][0]),
                                  device_group=stages[k][0]))
    return HeteroSpec(pipelines=pipes), stages
