"""Generate a layered ds_parallel_config JSON for a strategy.

Mirrors the reference workflow (models/*/generate_*_4d_config.py +
utils/parallel/generate_ds.py): describe the strategy as one (tp, pp) pair
per data-parallel pipeline — pairs may differ for heterogeneous (Malleus)
worlds — and expand it into the per-module JSON tree that
`hetu_amd.utils.ds_config.read_ds_parallel_config` /
`strategy_from_config` consume.

Examples:
    # homogeneous: 8 GPUs as tp2 x pp2 x dp2
    python generate_ds_config.py --tp-pp 2,2 2,2 --layers 24 -o ds.json
    # heterogeneous: one tp4 pipeline + two tp2 pipelines (Malleus)
    python generate_ds_config.py --tp-pp 4,1 2,1 2,1 --layers 24 -o h.json
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

from hetu_amd.utils.ds_config import (generate_ds_parallel_config,
                                      write_ds_parallel_config)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tp-pp", nargs="+", required=True,
                    metavar="TP,PP", help="one tp,pp pair per pipeline")
    ap.add_argument("--layers", type=int, required=True)
    ap.add_argument("--ngpus", type=int, default=None)
    ap.add_argument("--no-zero", action="store_true")
    ap.add_argument("--model-key", default="gpt")
    ap.add_argument("--recompute-layers", type=int, nargs="*", default=[])
    ap.add_argument("-o", "--out", required=True)
    args = ap.parse_args()
    tp_pp = [tuple(int(x) for x in s.split(",")) for s in args.tp_pp]
    cfg = generate_ds_parallel_config(
        tp_pp, num_layers=args.layers, ngpus=args.ngpus,
        zero=not args.no_zero, model_key=args.model_key,
        recompute_layers=args.recompute_layers)
    write_ds_parallel_config(cfg, args.out)
    print(f"wrote {args.out}: {len(tp_pp)} pipeline(s), "
          f"{args.layers} layers, devices={cfg['devices']}")


if __name__ == "__main__":
    main()
