#!/usr/bin/env python3
"""SFT driver (reference examples/sft): prompt/answer pairs with
prompt-masked loss, byte tokenizer, greedy packing stats, Llama graph.

Run: python examples/sft/sft_train.py          (1 rank, tiny model, CPU ok)
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
import torch  # noqa: E402

from hetu_amd.data.bucket import Bucket  # noqa: E402
from hetu_amd.data.tokenizers import ByteTokenizer  # noqa: E402
from hetu_amd.engine.sft_trainer import SFTTrainer, build_sft_example  # noqa
from hetu_amd.models.llama import LlamaConfig, build_llama_train_graph  # noqa
from hetu_amd.parallel.comm import comm_backend  # noqa: E402

PAIRS = [
    ("What is 2+2? ", "4."),
    ("Capital of France? ", "Paris."),
    ("Color of the sky? ", "Blue."),
    ("Opposite of hot? ", "Cold."),
] * 4


def main():
    comm = comm_backend()
    device = comm.device
    tok = ByteTokenizer()
    S = 64
    cfg = LlamaConfig(n_layer=2, n_head=4, n_kv_head=4, hidden=128,
                      ffn_hidden=256, vocab=tok.vocab_size, max_seq=S)
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    B = 4
    g, h = build_llama_train_graph(cfg, B, S, dtype=dtype, lr=5e-4)
    trainer = SFTTrainer(g, h, device)

    # packing stats for the curious (the padded path below is the simple
    # default; the packed path feeds ht.varlen_attention)
    bucket = Bucket(max_seqlen=S, alignment=16)
    for p, a in PAIRS:
        bucket.add(torch.tensor(tok.encode(p + a, bos=True, eos=True)))
    packed, cus = bucket.pack_data()
    print(f"packing: {len(PAIRS)} seqs -> {packed.shape[0]} bins "
          f"({sum(len(c) - 1 for c in cus)} segments)")

    exs = [build_sft_example(tok.encode(p, bos=True), tok.encode(a, eos=True),
                             S) for p, a in PAIRS]
    for step in range(12):
        batch = [exs[(step * B + i) % len(exs)] for i in range(B)]
        x = torch.stack([b[0] for b in batch]).to(device)
        y = torch.stack([b[1] for b in batch]).to(device)
        loss = trainer.sft_step(x, y)
        if comm.rank == 0 and step % 3 == 0:
            print(f"step {step} loss {float(loss):.4f}")


if __name__ == "__main__":
    main()
