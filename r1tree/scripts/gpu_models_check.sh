#!/bin/bash
set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0 TMPDIR=/tmp
mkdir -p gpurun_out

# 1. GPT-MoE 8x1.3B: a few real training steps on 1 GPU
timeout 420 python - > gpurun_out/moe_gpu.log 2>&1 <<'PY'
import torch, time
from hetu_amd.models.gpt import GPT_CONFIGS, build_gpt_train_graph
from hetu_amd.engine.trainer import Trainer
cfg = GPT_CONFIGS["gpt-moe-8x1.3b"]
g, h = build_gpt_train_graph(cfg, micro_batch=4, seq_len=2048,
                             dtype=torch.bfloat16, lr=1e-4)
tr = Trainer(g, h, torch.device("cuda", 0))
for i in range(4):
    ids = torch.randint(0, cfg.vocab, (4, 2048), device="cuda")
    lab = torch.randint(0, cfg.vocab, (4 * 2048,), device="cuda")
    t0 = time.time()
    loss = tr.step({h["input_ids"]: ids, h["labels"]: lab})
    torch.cuda.synchronize()
    print(f"moe step {i} loss {float(loss):.4f} {time.time()-t0:.2f}s")
print("MOE OK")
PY

# 2. Llama-13B: two steps
timeout 420 python - > gpurun_out/llama13b.log 2>&1 <<'PY'
import torch, time
from hetu_amd.models.llama import LLAMA_CONFIGS, build_llama_train_graph
from hetu_amd.engine.trainer import Trainer
cfg = LLAMA_CONFIGS["llama-13b"]
g, h = build_llama_train_graph(cfg, 2, 2048, dtype=torch.bfloat16, lr=1e-4)
tr = Trainer(g, h, torch.device("cuda", 0))
for i in range(3):
    ids = torch.randint(0, cfg.vocab, (2, 2048), device="cuda")
    lab = torch.randint(0, cfg.vocab, (2 * 2048,), device="cuda")
    t0 = time.time()
    loss = tr.step({h["input_ids"]: ids, h["labels"]: lab})
    torch.cuda.synchronize()
    print(f"13b step {i} loss {float(loss):.4f} {time.time()-t0:.2f}s")
print("LLAMA13B OK")
PY

# 3. PMC counters on FA fwd
timeout 300 rocprofv3 --kernel-trace --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_INSTS_MFMA SQ_INSTS_VALU SQ_INSTS_LDS -d gpurun_out/fa_pmc -- python scripts/fa_fwd_only.py > gpurun_out/fa_pmc.log 2>&1
tail -2 gpurun_out/moe_gpu.log gpurun_out/llama13b.log gpurun_out/fa_pmc.log
