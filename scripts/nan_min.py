#!/usr/bin/env python3
"""Shrink the capture-replay corruption repro: grid over model size."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from hetu_amd.engine.trainer import Trainer  # noqa: E402
from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph  # noqa: E402


def probe(n_layer, hidden, n_head, B, S, ffn_mult=4, vocab=50304,
          steps=6):
    cfg = GPTConfig(n_layer=n_layer, n_head=n_head, n_kv_head=n_head,
                    hidden=hidden, ffn_hidden=ffn_mult * hidden,
                    vocab=vocab, max_seq=S)
    torch.manual_seed(1234)
    dev = torch.device("cuda", 0)
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=torch.bfloat16, lr=1e-4)
    tr = Trainer(g, h, dev)
    pool = [torch.randint(0, vocab, (B, S), device=dev) for _ in range(4)]
    lp = [torch.randint(0, vocab, (B * S,), device=dev) for _ in range(4)]
    adams = [op for op in g.ops if op.type == "AdamStep"]
    verdict = "clean"
    for i in range(steps):
        lv = tr.step({h["input_ids"]: pool[i % 4], h["labels"]: lp[i % 4]})
        torch.cuda.synchronize()
        nbad = sum(1 for op in adams if "m" in op.interface.state and not (
            torch.isfinite(op.interface.state["m"]).all()
            and torch.isfinite(op.interface.state["v"]).all()
            and torch.isfinite(op.interface.state["master"]).all()))
        if nbad or not torch.isfinite(lv.float()):
            verdict = f"BAD@step{i} nbad={nbad} loss={float(lv.float())}"
            break
    print(f"L{n_layer} h{hidden} B{B} S{S} v{vocab}: {verdict}", flush=True)
    # free for next probe
    del tr, g, h
    torch.cuda.empty_cache()
    return verdict


if __name__ == "__main__":
    grids = [
        # many nodes, little memory
        (12, 256, 2, 2, 256),
        (24, 256, 2, 2, 256),
        (24, 128, 1, 2, 128),
        # few nodes, lots of memory
        (4, 2048, 16, 16, 1024),
        (2, 4096, 32, 16, 2048),
        # threshold scan
        (6, 1024, 8, 8, 1024),
        (8, 1024, 8, 8, 1024),
    ]
    for gspec in grids:
        try:
            probe(*gspec)
        except Exception as e:  # noqa: BLE001
            print(f"{gspec}: ERROR {e}", flush=True)
