"""JSON ds_parallel_config generate/read roundtrip (reference
utils/parallel/generate_ds.py + read_ds.py config2ds)."""
import json

from hetu_amd.parallel.dstates import NULL_HETERO_DIM
from hetu_amd.parallel.hetero import HeteroSpec
from hetu_amd.parallel.pipeline import PipelineSpec
from hetu_amd.utils.ds_config import (config2ds, convert_strategy,
                                      generate_ds_parallel_config,
                                      read_ds_parallel_config,
                                      strategy_from_config,
                                      write_ds_parallel_config)


def test_convert_strategy_keeps_tp_in_node():
    ltg, gpu_pos = convert_strategy([(8, 1), (4, 2)], 16, 4,
                                    gpus_per_node=8)
    for layer in ltg:
        assert len(layer) == 2
        for grp in layer:
            nodes = {r // 8 for r in grp}
            assert len(nodes) == 1 or len(grp) % 8 == 0
    assert len(gpu_pos) == 16
    # pipeline 1 (tp4 pp2) has two stages
    stages1 = {gpu_pos[r][1] for r in gpu_pos if gpu_pos[r][0] == 1}
    assert stages1 == {0, 1}


def test_homogeneous_roundtrip(tmp_path):
    cfg = generate_ds_parallel_config([(2, 2)], num_layers=4)
    p = str(tmp_path / "ds.json")
    write_ds_parallel_config(cfg, p)
    entries = read_ds_parallel_config(p)
    assert "input" in entries and "gpt.wte" in entries
    assert "gpt.blocks.blocks0.attn.qkv" in entries
    u, dgu = entries["gpt.blocks.blocks0.attn.qkv"]
    assert u.hetero_dim == NULL_HETERO_DIM and u.size() == 1
    ds = u.get(0)
    assert ds.get_dim(1) == 2 and ds.device_num == 2
    spec, stages = strategy_from_config(p)
    assert isinstance(spec, PipelineSpec)
    assert spec.pp == 2 and spec.tp == 2
    assert stages[0] == [[0, 1], [2, 3]]
    # dp==1 forces zero off
    assert cfg["zero"] is False


def test_hetero_roundtrip():
    cfg = generate_ds_parallel_config([(4, 1), (2, 1), (2, 1)],
                                      num_layers=2)
    entries = read_ds_parallel_config(cfg)
    u, dgu = entries["gpt.wte"]
    assert u.is_hetero() and u.hetero_dim == -1 and u.size() == 3
    # pipeline 0 is tp4: local layout (union stripped) splits dim 0 by 4
    loc = u.get_local(0)
    assert loc.get_dim(0) == 4 and loc.device_num == 4
    loc2 = u.get_local(1)
    assert loc2.get_dim(0) == 2 and loc2.device_num == 2
    # placeholder input is hetero along the batch dim
    ui, _ = entries["input"]
    assert ui.hetero_dim == 0
    spec, stages = strategy_from_config(cfg)
    assert isinstance(spec, HeteroSpec)
    assert sorted(p.tp for p in spec.pipelines) == [2, 2, 4]
    # zero spread into variable entries (dp>1 keeps it on)
    assert cfg["zero"] is True
    assert u.get(0).zero


def test_config2ds_orders():
    e = {"split": {"0": [2]}, "dup": [3], "device_group_union": [[0, 1, 2,
         3, 4, 5]], "type": "placeholder"}
    u, _ = config2ds(e)
    assert u.get(0).order == [0, -1]
    e["type"] = "variable"
    u, _ = config2ds(e)
    assert u.get(0).order == [-1, 0]


def test_interop_with_reference_style_json(tmp_path):
    """A hand-written reference-schema file (llama key, rmsnorm names)
    parses: the reader walks any tree shape."""
    cfg = {
        "zero": True, "devices": [0, 1],
        "input": {"split": {"0": [2, 2]}, "dup": [1, 1],
                  "device_group_union": [[0], [1]], "type": "placeholder"},
        "llama": {"wte": {"split": {"0": [1, 1]}, "dup": [2, 2],
                          "device_group_union": [[0], [1]],
                          "type": "variable"},
                  "blocks": {"blocks0": {
                      "range": [0], "recompute": [False, False],
                      "attn": {"qkv": {"split": {"1": [1, 1]},
                                       "dup": [2, 2],
                                       "device_group_union": [[0], [1]],
                                       "type": "variable"}}}}},
        "label": {"split": {"0": [2, 2]}, "dup": [1, 1],
                  "device_group_union": [[0], [1]], "type": "placeholder"},
    }
    p = str(tmp_path / "ref.json")
    with open(p, "w") as f:
        json.dump(cfg, f)
    entries = read_ds_parallel_config(p)
    u, _ = entries["llama.wte"]
    assert u.is_hetero() and u.size() == 2 and u.get(0).zero
    spec, stages = strategy_from_config(p, model_key="llama")
    # two IDENTICAL tp1 pipelines collapse to homogeneous dp2
    assert isinstance(spec, PipelineSpec)
    assert (spec.pp, spec.dp, spec.tp) == (1, 2, 1)


def test_homogeneous_dp_pipelines_collapse_to_pipeline_spec():
    cfg = generate_ds_parallel_config([(2, 2), (2, 2)], num_layers=4)
    spec, stages = strategy_from_config(cfg)
    assert isinstance(spec, PipelineSpec)
    assert (spec.pp, spec.dp, spec.tp) == (2, 2, 2)


def test_recompute_layers_roundtrip():
    from hetu_amd.utils.ds_config import recompute_layers_from_config
    cfg = generate_ds_parallel_config([(2, 1)], num_layers=4,
                                      recompute_layers=[1, 3])
    assert recompute_layers_from_config(cfg) == [1, 3]


def test_strategy_export_roundtrip():
    """Galvatron Strategy -> ds_parallel_config -> strategy recovery."""
    from hetu_amd.galvatron.cost_model import Strategy
    from hetu_amd.utils.ds_config import strategy_to_ds_config
    s = Strategy(dp=2, tp=2, pp=2, zero=True, recompute_layers=2)
    cfg = strategy_to_ds_config(s, num_layers=8)
    spec, _ = strategy_from_config(cfg)
    assert isinstance(spec, PipelineSpec)
    assert (spec.pp, spec.dp, spec.tp) == (2, 2, 2)
    assert cfg["zero"] is True
    from hetu_amd.utils.ds_config import recompute_layers_from_config
    assert recompute_layers_from_config(cfg) == [0, 1]


def test_convert_strategy_tp_spanning_nodes():
    """tp=16 spans two whole nodes (allowed when tp % gpus_per_node == 0);
    remaining pipelines pack into what's left."""
    ltg, gpu_pos = convert_strategy([(16, 1), (8, 1)], 24, 2,
                                    gpus_per_node=8)
    g16 = [g for g in ltg[0] if len(g) == 16][0]
    assert {r // 8 for r in g16} == {0, 1} or \
        {r // 8 for r in g16} == {1, 2} or \
        {r // 8 for r in g16} == {0, 2}
    g8 = [g for g in ltg[0] if len(g) == 8][0]
    assert len({r // 8 for r in g8}) == 1
    assert len(gpu_pos) == 24


def test_hetero_layers_and_rank_map():
    """Non-uniform per-pipeline stage splits + rank relabeling
    (reference parallel_config.py generate_gpt_3d_config hetero_layers /
    rank_to_device_mapping)."""
    ltg, gpu_pos = convert_strategy(
        [(2, 2), (1, 2)], 6, 4,
        hetero_layers=[[3, 1], [2, 2]],
        rank_map={0: 10, 1: 11})
    # pipeline 0 (tp2): stage 0 owns layers 0-2, stage 1 owns layer 3
    p0_stage_of_layer = [ltg[l][0] for l in range(4)]
    assert p0_stage_of_layer[0] == p0_stage_of_layer[2]
    assert p0_stage_of_layer[3] != p0_stage_of_layer[0]
    # pipeline 1 (tp1): 2+2 split
    p1 = [ltg[l][1] for l in range(4)]
    assert p1[0] == p1[1] and p1[2] == p1[3] and p1[1] != p1[2]
    # ranks 0/1 relabeled to 10/11 everywhere
    all_ranks = {r for layer in ltg for grp in layer for r in grp}
    assert 0 not in all_ranks and 1 not in all_ranks
    assert {10, 11} <= all_ranks
    assert gpu_pos[10][0] in (0, 1)
    # generated config carries only the used (relabeled) devices
    cfg = generate_ds_parallel_config([(2, 2), (1, 2)], num_layers=4,
                                      hetero_layers=[[3, 1], [2, 2]],
                                      rank_map={0: 10, 1: 11})
    assert 0 not in cfg["devices"] and 10 in cfg["devices"]


def test_strategy_roundtrip_property():
    """generate -> strategy_from_config recovers the (tp, pp) multiset
    for a sweep of homogeneous and heterogeneous strategy lists."""
    import itertools
    cases = []
    for tp, pp, dp in itertools.product((1, 2, 4), (1, 2, 4), (1, 2)):
        if tp * pp * dp <= 16:
            cases.append([(tp, pp)] * dp)
    cases += [[(4, 1), (2, 1), (2, 1)], [(2, 1), (1, 1), (1, 1)],
              [(4, 1), (4, 1)]]
    for tp_pp in cases:
        layers = 4
        if any(layers % pp for _, pp in tp_pp):
            continue
        cfg = generate_ds_parallel_config(tp_pp, num_layers=layers)
        spec, stages = strategy_from_config(cfg)
        got = sorted((len(s[0]), len(s)) for s in stages)
        assert got == sorted(tp_pp), (tp_pp, got)
        if isinstance(spec, PipelineSpec):
            assert [(spec.tp, spec.pp)] * spec.dp == list(tp_pp)
