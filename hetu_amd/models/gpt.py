"""GPT model family built on the define-and-run graph API.

Reference parity: python/hetu/models/gpt/gpt_model.py (GPTLMHeadModel) — a
pre-LN transformer with learned position embeddings, fused-HIP LayerNorm,
flash attention, GELU MLP, and sparse softmax CE loss. The 7B config matches
the reference CI "gpt 7b" shape (32 layers x 4096 hidden x 32 heads,
tests/ci_test/scripts/pssh_train_hetu.sh).
"""
from __future__ import annotations

import dataclasses
import math
from typing import Dict, Optional

import torch

from .. import DistributedStates
from ..graph.graph import DefineAndRunGraph, push_graph, pop_graph
from ..graph.ops import api as ht
from ..graph.ops.optim import Adam


@dataclasses.dataclass
class GPTConfig:
    n_layer: int = 12
    n_head: int = 12
    n_kv_head: int = 12
    hidden: int = 768
    ffn_hidden: int = 3072
    vocab: int = 50304
    max_seq: int = 2048
    dropout: float = 0.0
    init_std: float = 0.02
    tie_embeddings: bool = False


GPT_CONFIGS = {
    "gpt2-345m": GPTConfig(n_layer=24, n_head=16, n_kv_head=16, hidden=1024,
                           ffn_hidden=4096, vocab=50304, max_seq=1024),
    "gpt3-7b": GPTConfig(n_layer=32, n_head=32, n_kv_head=32, hidden=4096,
                         ffn_hidden=16384, vocab=50304, max_seq=2048),
    "gpt3-13b": GPTConfig(n_layer=40, n_head=40, n_kv_head=40, hidden=5120,
                          ffn_hidden=20480, vocab=50304, max_seq=2048),
}


def _randn(shape, std, dtype):
    return (torch.randn(*shape) * std).to(dtype)


def build_gpt_forward(cfg: GPTConfig, input_ids, micro_batch: int,
                      seq_len: int, dtype=torch.bfloat16, prefix: str = "gpt",
                      dp: int = 1, device_group=None):
    """Build forward graph ops in the CURRENT graph; returns logits tensor.
    input_ids: graph tensor [B, S] int64 (local shard; ds split(0) for dp>1).
    Data-parallel SPMD: weights duplicated, activations split on dim 0,
    weight grads deduced partial -> allreduce at minimize()."""
    B, S = micro_batch, seq_len
    Hn, Dh = cfg.n_head, cfg.hidden // cfg.n_head
    std = cfg.init_std
    ds_dup = (DistributedStates(dp, {-1: dp}, order=[-1]) if dp > 1
              else None)

    def var(data, name, requires_grad=True):
        return ht.variable(data, name=name, requires_grad=requires_grad,
                           ds=ds_dup, device_group=device_group)

    wte = var(_randn((cfg.vocab, cfg.hidden), std, dtype), f"{prefix}.wte")
    wpe = var(_randn((cfg.max_seq, cfg.hidden), std, dtype), f"{prefix}.wpe")
    pos_ids = var(torch.arange(S, dtype=torch.int64), f"{prefix}.pos",
                  requires_grad=False)

    x = ht.embedding(wte, input_ids)               # [B, S, h]
    pos = ht.embedding(wpe, pos_ids)               # [S, h]
    x = ht.add(x, pos)
    if cfg.dropout > 0:
        x = ht.dropout(x, cfg.dropout)

    proj_std = std / math.sqrt(2 * cfg.n_layer)
    for li in range(cfg.n_layer):
        p = f"{prefix}.h{li}"
        ln1_w = var(torch.ones(cfg.hidden, dtype=dtype), f"{p}.ln1.w")
        ln1_b = var(torch.zeros(cfg.hidden, dtype=dtype), f"{p}.ln1.b")
        wqkv = var(_randn((3 * cfg.hidden, cfg.hidden), std, dtype), f"{p}.wqkv")
        bqkv = var(torch.zeros(3 * cfg.hidden, dtype=dtype), f"{p}.bqkv")
        wproj = var(_randn((cfg.hidden, cfg.hidden), proj_std, dtype), f"{p}.wproj")
        bproj = var(torch.zeros(cfg.hidden, dtype=dtype), f"{p}.bproj")
        ln2_w = var(torch.ones(cfg.hidden, dtype=dtype), f"{p}.ln2.w")
        ln2_b = var(torch.zeros(cfg.hidden, dtype=dtype), f"{p}.ln2.b")
        wfc = var(_randn((cfg.ffn_hidden, cfg.hidden), std, dtype), f"{p}.wfc")
        bfc = var(torch.zeros(cfg.ffn_hidden, dtype=dtype), f"{p}.bfc")
        wfc2 = var(_randn((cfg.hidden, cfg.ffn_hidden), proj_std,
                                  dtype), f"{p}.wfc2")
        bfc2 = var(torch.zeros(cfg.hidden, dtype=dtype), f"{p}.bfc2")

        # ---- attention block ----
        h = ht.layer_norm(x, ln1_w, ln1_b)
        qkv = ht.linear(h, wqkv, bqkv)                     # [B,S,3h]
        qkv = ht.reshape(qkv, (B, S, 3, Hn, Dh))
        q = ht.reshape(ht.slice_(qkv, 2, 0, 1), (B, S, Hn, Dh))
        k = ht.reshape(ht.slice_(qkv, 2, 1, 1), (B, S, Hn, Dh))
        v = ht.reshape(ht.slice_(qkv, 2, 2, 1), (B, S, Hn, Dh))
        q = ht.transpose(q, 1, 2)                          # [B,H,S,D]
        k = ht.transpose(k, 1, 2)
        v = ht.transpose(v, 1, 2)
        attn = ht.attention(q, k, v, causal=True)
        attn = ht.transpose(attn, 1, 2)                    # [B,S,H,D]
        attn = ht.reshape(attn, (B, S, cfg.hidden))
        attn = ht.linear(attn, wproj, bproj)
        if cfg.dropout > 0:
            attn = ht.dropout(attn, cfg.dropout)
        x = ht.add(x, attn)

        # ---- MLP block ----
        h2 = ht.layer_norm(x, ln2_w, ln2_b)
        h2 = ht.gelu(ht.linear(h2, wfc, bfc))
        h2 = ht.linear(h2, wfc2, bfc2)
        if cfg.dropout > 0:
            h2 = ht.dropout(h2, cfg.dropout)
        x = ht.add(x, h2)

    lnf_w = var(torch.ones(cfg.hidden, dtype=dtype), f"{prefix}.lnf.w")
    lnf_b = var(torch.zeros(cfg.hidden, dtype=dtype), f"{prefix}.lnf.b")
    x = ht.layer_norm(x, lnf_w, lnf_b)
    if cfg.tie_embeddings:
        logits = ht.matmul(ht.reshape(x, (B * S, cfg.hidden)), wte,
                           trans_b=True)
    else:
        lm_head = var(_randn((cfg.vocab, cfg.hidden), std, dtype), f"{prefix}.lm_head")
        logits = ht.linear(ht.reshape(x, (B * S, cfg.hidden)), lm_head)
    return logits                                           # [B*S, V]


def build_gpt_train_graph(cfg: GPTConfig, micro_batch: int, seq_len: int,
                          dtype=torch.bfloat16, lr: float = 1e-4,
                          dp: int = 1, device_group=None,
                          graph: Optional[DefineAndRunGraph] = None,
                          zero: bool = False
                          ) -> (DefineAndRunGraph, Dict):
    g = graph or DefineAndRunGraph("gpt_train")
    if dp > 1 and device_group is None:
        device_group = list(range(dp))
    ds_in = (DistributedStates(dp, {0: dp}, order=[0]) if dp > 1 else None)
    push_graph(g)
    try:
        input_ids = ht.placeholder((micro_batch, seq_len),
                                   dtype=torch.int64, name="input_ids",
                                   ds=ds_in, device_group=device_group)
        labels = ht.placeholder((micro_batch * seq_len,),
                                dtype=torch.int64, name="labels",
                                ds=ds_in, device_group=device_group)
        logits = build_gpt_forward(cfg, input_ids, micro_batch, seq_len,
                                   dtype, dp=dp, device_group=device_group)
        per_tok = ht.softmax_cross_entropy_sparse(logits, labels)
        loss = ht.reduce_mean(per_tok)
        loss_report = loss
        if dp > 1:
            loss_report = ht.comm(
                loss, DistributedStates(dp, {-1: dp}, order=[-1]),
                name="loss_allreduce")
        opt = Adam(lr=lr, zero=zero)
        train_op = opt.minimize(loss)
    finally:
        pop_graph()
    return g, {"input_ids": input_ids, "labels": labels,
               "loss": loss_report, "train_op": train_op, "optimizer": opt}
