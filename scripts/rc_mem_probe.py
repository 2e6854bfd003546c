#!/usr/bin/env python3
import os, sys
os.environ["HETU_AMD_MEM_TRACE"] = "1"
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from hetu_amd.engine.trainer import Trainer
from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph
dev = torch.device("cuda", 0)
cfg = GPTConfig(n_layer=8, n_head=8, n_kv_head=8, hidden=1024,
                ffn_hidden=4096, vocab=50304, max_seq=1024)
B, S = 8, 1024
for rc in (False, True):
    torch.manual_seed(5)
    torch.cuda.empty_cache(); torch.cuda.reset_peak_memory_stats()
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=torch.bfloat16, lr=1e-4,
                                 recompute=rc)
    tr = Trainer(g, h, dev, capture=False)
    for i in range(3):
        lv = tr.step({h["input_ids"]: torch.randint(0, cfg.vocab, (B, S), device=dev),
                      h["labels"]: torch.randint(0, cfg.vocab, (B*S,), device=dev)})
    torch.cuda.synchronize()
    ex = g.executor()
    mp = ex._mem_peak
    print(f"rc={rc}: cuda_peak={torch.cuda.max_memory_allocated()/1e9:.2f}GB "
          f"executor_peak={mp[0]/1e9:.2f}GB at {mp[1]}:{mp[2]} (op {mp[3]}) "
          f"values_total={mp[5]/1e9:.2f}GB", flush=True)
    for nm, nb in mp[4]:
        print(f"    {nm}: {nb/1e6:.0f}MB", flush=True)
    # allocator block histogram right now (post-step, synced)
    import collections
    hist = collections.Counter()
    for seg in torch.cuda.memory_snapshot():
        for blk in seg.get("blocks", []):
            if blk.get("state") == "active_allocated":
                hist[round(blk["size"] / 1e6)] += 1
    big = sorted(((sz, n) for sz, n in hist.items() if sz >= 8),
                 key=lambda x: -x[0] * x[1])[:12]
    tot = sum(sz * n for sz, n in hist.items())
    print(f"    post-step active blocks total {tot/1e3:.2f}GB; "
          f"big: {big}", flush=True)
    del tr, g, h
