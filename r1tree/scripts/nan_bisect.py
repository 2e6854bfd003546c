#!/usr/bin/env python3
"""Debug: per-step loss of the 7B bench config under kahn/dfs topo and
capture on/off, to localize the NaN regression."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402


def run(topo, capture, steps=8, model="gpt3-7b", B=16, S=2048):
    os.environ["HETU_AMD_TOPO"] = topo
    os.environ["HETU_AMD_CAPTURE"] = capture
    from hetu_amd.engine.trainer import Trainer
    from hetu_amd.models.gpt import GPT_CONFIGS, build_gpt_train_graph
    cfg = GPT_CONFIGS[model]
    torch.manual_seed(1234)
    dev = torch.device("cuda", 0)
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=torch.bfloat16, lr=1e-4)
    tr = Trainer(g, h, dev)
    pool = [torch.randint(0, cfg.vocab, (B, S), device=dev)
            for _ in range(4)]
    lpool = [torch.randint(0, cfg.vocab, (B * S,), device=dev)
             for _ in range(4)]
    losses = []
    for i in range(steps):
        t0 = time.perf_counter()
        lv = tr.step({h["input_ids"]: pool[i % 4], h["labels"]: lpool[i % 4]})
        torch.cuda.synchronize()
        losses.append(float(lv.float()))
        print(f"[{topo} cap={capture}] step {i}: loss={losses[-1]:.4f} "
              f"({(time.perf_counter()-t0)*1e3:.0f} ms)", flush=True)
    return losses


if __name__ == "__main__":
    topo = sys.argv[1] if len(sys.argv) > 1 else "kahn"
    cap = sys.argv[2] if len(sys.argv) > 2 else "1"
    run(topo, cap)
