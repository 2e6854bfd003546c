#!/usr/bin/env python3
"""Driver benchmark contract: flagship training step, tokens/sec whole-job.

Metric (BASELINE.json): tokens/sec (whole node), GPT-3 7B, Galvatron
auto-parallel, at 1/2/4/8 MI355X; synthetic data, random-init weights, bf16
compute.  For N>1 the driver launches this under torch.distributed.run with
one rank per GPU (RCCL); per-GPU work is fixed (weak scaling).

--parallel auto (default) runs the Galvatron-style search
(hetu_amd/galvatron) for this node size and executes the chosen strategy
(dp x tp x cp x pp, zero, micro-batching); explicit strategies like
"dp8", "dp2_tp4", "dp4_pp2_z" are accepted too.
"""
import argparse
import json
import os
import re
import sys
import time

_REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, _REPO)

# hipBLASLt algorithm pinning: if a pre-tuned TunableOp table is committed
# under tunableop/, use it (tuning off -> zero runtime cost, same math)
_TUNED = os.path.join(_REPO, "tunableop", "tunableop_results0.csv")
if os.path.exists(_TUNED) and "PYTORCH_TUNABLEOP_ENABLED" not in os.environ:
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = os.path.join(
        _REPO, "tunableop", "tunableop_results%d.csv")

import torch  # noqa: E402


def parse_strategy(text, world):
    from hetu_amd.galvatron.cost_model import Strategy
    st = Strategy()
    for tok in text.split("_"):
        if tok == "z" or tok == "zero":
            st.zero = True
        elif m := re.fullmatch(r"(dp|tp|pp|cp|mb)(\d+)", tok):
            k, v = m.group(1), int(m.group(2))
            if k == "mb":
                st.micro_batch = v
            else:
                setattr(st, k, v)
        else:
            raise ValueError(f"bad strategy token {tok}")
    assert st.world == world, f"strategy {text} != world {world}"
    return st


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default="gpt3-7b")
    ap.add_argument("--micro-batch", type=int, default=0,
                    help="0 = from strategy search")
    ap.add_argument("--global-batch", type=int, default=0,
                    help="0 = 16 per GPU")
    ap.add_argument("--seq-len", type=int, default=2048)
    ap.add_argument("--parallel", default="auto")
    ap.add_argument("--capture", default="auto")
    args = ap.parse_args()

    if args.capture != "auto":
        os.environ["HETU_AMD_CAPTURE"] = args.capture
    # hipGraph capture happens on the 2nd step (optimizer state must exist
    # before capture); keep it out of the timed region
    args.warmup = max(args.warmup, 2)

    import hetu_amd  # noqa: F401
    from hetu_amd.engine.trainer import Trainer
    from hetu_amd.models.gpt import (GPT_CONFIGS, build_gpt_pipeline_stage,
                                     build_gpt_train_graph)
    from hetu_amd.nn.parallel import ParallelSpec
    from hetu_amd.parallel.comm import comm_backend
    from hetu_amd.galvatron.cost_model import ModelShape
    from hetu_amd.galvatron.search import search

    assert torch.cuda.is_available(), "bench requires a GPU"
    comm = comm_backend()
    rank, ws = comm.rank, comm.world_size
    device = comm.device
    torch.cuda.set_device(device)

    cfg = GPT_CONFIGS[args.model]
    S = args.seq_len
    global_batch = args.global_batch or 16 * ws

    # ---- strategy ---------------------------------------------------------
    if args.parallel == "auto":
        shape = ModelShape(n_layer=cfg.n_layer, hidden=cfg.hidden,
                           ffn_hidden=cfg.ffn_hidden, vocab=cfg.vocab,
                           n_head=cfg.n_head, kind="gpt")
        st, est = search(shape, S, ws, global_batch)
    else:
        st = parse_strategy(args.parallel, ws)
        est = None
    if args.micro_batch:
        st.micro_batch = args.micro_batch
    num_mb = global_batch // (st.dp * st.micro_batch)
    if rank == 0:
        print(f"[bench] strategy={st.name()} micro_batch={st.micro_batch} "
              f"num_micro_batches={num_mb} global_batch={global_batch}",
              file=sys.stderr)

    B = st.micro_batch
    S_loc = S // st.cp
    torch.manual_seed(1234 + rank)

    if st.pp == 1:
        spec = ParallelSpec(dp=st.dp, tp=st.tp, cp=st.cp)
        # dp-only steady state: micro-batches fold into one captured step
        # of batch B*num_mb when no pipeline is involved
        eff_B = B * num_mb
        g, h = build_gpt_train_graph(cfg, micro_batch=eff_B, seq_len=S_loc,
                                     dtype=torch.bfloat16, lr=1e-4,
                                     spec=spec, zero=st.zero)
        trainer = Trainer(g, h, device)
        pool = [torch.randint(0, cfg.vocab, (eff_B, S_loc), device=device)
                for _ in range(4)]
        lpool = [torch.randint(0, cfg.vocab, (eff_B * S_loc,), device=device)
                 for _ in range(4)]

        def step_fn(i):
            return trainer.step({h["input_ids"]: pool[i % 4],
                                 h["labels"]: lpool[i % 4]})
        capture_on = lambda: trainer._cuda_graph is not None  # noqa: E731
    else:
        from hetu_amd.parallel.pipeline import PipelineRunner, PipelineSpec
        pspec = PipelineSpec(pp=st.pp, dp=st.dp, tp=st.tp)
        stage = build_gpt_pipeline_stage(cfg, pspec, micro_batch=B,
                                         seq_len=S_loc,
                                         dtype=torch.bfloat16, lr=1e-4,
                                         zero=st.zero)
        runner = PipelineRunner(pspec, stage, device)
        hs = stage.h
        mbs_pool = []
        for _ in range(2):
            mbs = []
            for _ in range(num_mb):
                feed = {}
                if "input_ids" in hs:
                    feed[hs["input_ids"]] = torch.randint(
                        0, cfg.vocab, (B, S_loc), device=device)
                if "labels" in hs:
                    feed[hs["labels"]] = torch.randint(
                        0, cfg.vocab, (B * S_loc,), device=device)
                mbs.append(feed)
            mbs_pool.append(mbs)

        def step_fn(i):
            return runner.step(mbs_pool[i % 2])
        capture_on = lambda: False  # noqa: E731

    loss = None
    for i in range(args.warmup):
        loss = step_fn(i)

    comm.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        loss = step_fn(i)
    comm.barrier()
    torch.cuda.synchronize()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if ws > 1:
        import torch.distributed as dist
        et = torch.tensor([elapsed], device=device)
        dist.all_reduce(et, op=dist.ReduceOp.MAX)
        elapsed = et.item()

    ms_per_step = elapsed / args.steps * 1000
    tokens_per_step = global_batch * S
    tok_s = tokens_per_step * args.steps / elapsed

    if rank == 0:
        out = {
            "metric": "tokens/sec (whole node) GPT-3 7B Galvatron "
                      "auto-parallel",
            "value": tok_s,
            "unit": "tokens/s",
            "n_gpus": ws,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "loss": float(loss.float().item()) if loss is not None else None,
            "config": {"model": args.model, "global_batch": global_batch,
                       "seq_len": S, "parallelism": st.name(),
                       "micro_batch": st.micro_batch,
                       "capture": capture_on()},
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
