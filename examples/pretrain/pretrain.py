#!/usr/bin/env python3
"""Pretraining driver from a YAML config (reference examples/pretrain).

Usage (1 GPU):   python examples/pretrain/pretrain.py config/gpt_345m_dp.yaml
Multi-GPU:       python -m torch.distributed.run --nnodes=1 \
                   --nproc-per-node 8 --master-addr 127.0.0.1 \
                   examples/pretrain/pretrain.py config/llama_7b_tp2_dp4.yaml

Synthetic data by default; point `data_path` at a JSON-lines file with
"input_ids" for real tokens.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.engine.trainer import Trainer  # noqa: E402
from hetu_amd.engine.trainer_config import TrainingConfig  # noqa: E402
from hetu_amd.models.gpt import GPT_CONFIGS, build_gpt_train_graph  # noqa
from hetu_amd.models.llama import (LLAMA_CONFIGS,  # noqa: E402
                                   build_llama_train_graph)
from hetu_amd.nn.parallel import ParallelSpec  # noqa: E402
from hetu_amd.parallel.comm import comm_backend  # noqa: E402
from hetu_amd.utils.checkpoint import collect_adam_states, save_model  # noqa


def main():
    cfg_path = sys.argv[1] if len(sys.argv) > 1 else os.path.join(
        os.path.dirname(__file__), "config", "gpt_345m_dp.yaml")
    tc = TrainingConfig.from_yaml(cfg_path)
    comm = comm_backend()
    device = comm.device
    torch.manual_seed(tc.seed + comm.rank)

    spec = ParallelSpec(dp=tc.dp, tp=tc.tp, cp=tc.cp,
                        sequence_parallel=tc.sequence_parallel)
    B = tc.global_batch // tc.dp
    S = tc.seq_len // tc.cp
    if tc.architecture == "LlamaLMHeadModel":
        mcfg = LLAMA_CONFIGS[tc.model]
        g, h = build_llama_train_graph(mcfg, B, S, dtype=tc.dtype(),
                                       lr=tc.lr, spec=spec, zero=tc.zero)
    else:
        mcfg = GPT_CONFIGS[tc.model]
        g, h = build_gpt_train_graph(mcfg, micro_batch=B, seq_len=S,
                                     dtype=tc.dtype(), lr=tc.lr, spec=spec,
                                     zero=tc.zero)
    trainer = Trainer(g, h, device,
                      lr_schedule=tc.lr_schedule())
    t0 = time.time()
    for step in range(tc.steps):
        ids = torch.randint(0, mcfg.vocab, (B, S), device=device)
        labels = torch.randint(0, mcfg.vocab, (B * S,), device=device)
        loss = trainer.step({h["input_ids"]: ids, h["labels"]: labels})
        if comm.rank == 0 and step % 10 == 0:
            print(f"step {step} loss {float(loss):.4f} "
                  f"({(time.time() - t0) / (step + 1):.3f}s/step)")
        if tc.save_every and tc.save_path and (step + 1) % tc.save_every == 0:
            save_model(g.parameters, tc.save_path, comm,
                       optimizer_states=collect_adam_states(g))


if __name__ == "__main__":
    main()
