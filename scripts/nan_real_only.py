import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from hetu_amd.engine.trainer import Trainer
from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph
dev = torch.device("cuda", 0)
L, H, B, S, V = 12, 256, 2, 256, 50304
torch.manual_seed(1234)
cfg = GPTConfig(n_layer=L, n_head=2, n_kv_head=2, hidden=H,
                ffn_hidden=4*H, vocab=V, max_seq=S)
g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                             dtype=torch.bfloat16, lr=1e-4)
tr = Trainer(g, h, dev)
adams = [op for op in g.ops if op.type == "AdamStep"]
for i in range(6):
    feed = {h["input_ids"]: torch.randint(0, V, (B, S), device=dev),
            h["labels"]: torch.randint(0, V, (B*S,), device=dev)}
    lv = tr.step(feed)
    torch.cuda.synchronize()
    nmv = sum(1 for op in adams if "m" in op.interface.state and not (
        torch.isfinite(op.interface.state["m"]).all()
        and torch.isfinite(op.interface.state["v"]).all()))
    nma = sum(1 for op in adams if "master" in op.interface.state
              and not torch.isfinite(op.interface.state["master"]).all())
    print(f"step {i}: loss={float(lv.float()):.4f} mv={nmv} master={nma}",
          flush=True)
