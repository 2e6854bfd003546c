#!/usr/bin/env python3
"""Second capture bug: GPT repro minus pieces.  Tower is clean now; add
GPT's extras one at a time: embedding front-end, vocab-CE loss, fused-qkv
attention."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
from hetu_amd.graph.graph import (DefineAndRunGraph, pop_graph,  # noqa: E402
                                  push_graph)
from hetu_amd.graph.ops import api as ht  # noqa: E402
from hetu_amd.graph.ops.optim import Adam  # noqa: E402
from hetu_amd.engine.trainer import Trainer  # noqa: E402
from hetu_amd.nn.parallel import vocab_parallel_cross_entropy  # noqa: E402

dev = torch.device("cuda", 0)
L, H, B, S, V = 12, 256, 2, 256, 50304
dtype = torch.bfloat16


def build(embed, vce, attn):
    torch.manual_seed(0)
    g = DefineAndRunGraph("t")
    push_graph(g)
    try:
        if embed:
            ids = ht.placeholder((B, S), dtype=torch.int64, name="ids")
            wte = ht.variable(torch.randn(V, H, dtype=dtype) * 0.02,
                              name="wte.w")
            wpe = ht.variable(torch.randn(S, H, dtype=dtype) * 0.02,
                              name="wpe.w")
            pos = ht.variable(torch.arange(S), name="pos",
                              requires_grad=False)
            x0 = ht.add(ht.vocab_parallel_embedding(wte, ids, V),
                        ht.embedding(wpe, pos))
            cur = ht.reshape(x0, (B, S, H))
            feed_key = ids
        else:
            x = ht.placeholder((B, S, H), dtype=dtype, name="x")
            cur = x
            feed_key = x
        for i in range(L):
            if attn == "gptblock":
                # exact GPTBlock structure: ln1->attn(wqkv,wo)->res,
                # ln2->mlp(4H, biases)->res
                w = ht.variable(torch.ones(H, dtype=dtype), name=f"ln1{i}.w")
                b = ht.variable(torch.zeros(H, dtype=dtype),
                                name=f"ln1{i}.b")
                y = ht.layer_norm(cur, w, b, 1e-5)
                qw = ht.variable(torch.randn(3 * H, H, dtype=dtype) * 0.02,
                                 name=f"a{i}.wqkv")
                qb = ht.variable(torch.zeros(3 * H, dtype=dtype),
                                 name=f"a{i}.bqkv")
                qkv = ht.linear(y, qw, qb)
                nh = H // 128
                o = ht.fused_qkv_attention(qkv, nh, nh, 128, causal=True)
                ow = ht.variable(torch.randn(H, H, dtype=dtype) * 0.005,
                                 name=f"a{i}.wo")
                ob = ht.variable(torch.zeros(H, dtype=dtype),
                                 name=f"a{i}.bo")
                cur = ht.add(cur, ht.linear(o, ow, ob))
                w = ht.variable(torch.ones(H, dtype=dtype), name=f"ln2{i}.w")
                b = ht.variable(torch.zeros(H, dtype=dtype),
                                name=f"ln2{i}.b")
                y = ht.layer_norm(cur, w, b, 1e-5)
                w1 = ht.variable(torch.randn(4 * H, H, dtype=dtype) * 0.02,
                                 name=f"l{i}.w1")
                b1 = ht.variable(torch.zeros(4 * H, dtype=dtype),
                                 name=f"l{i}.b1")
                y = ht.gelu(ht.linear(y, w1, b1))
                w2 = ht.variable(torch.randn(H, 4 * H, dtype=dtype) * 0.005,
                                 name=f"l{i}.w2")
                b2 = ht.variable(torch.zeros(H, dtype=dtype),
                                 name=f"l{i}.b2")
                cur = ht.add(cur, ht.linear(y, w2, b2))
                continue
            w = ht.variable(torch.ones(H, dtype=dtype), name=f"ln{i}.w")
            b = ht.variable(torch.zeros(H, dtype=dtype), name=f"ln{i}.b")
            y = ht.layer_norm(cur, w, b, 1e-5)
            if attn:
                qw = ht.variable(torch.randn(3 * H, H, dtype=dtype) * 0.02,
                                 name=f"a{i}.wqkv")
                qb = ht.variable(torch.zeros(3 * H, dtype=dtype),
                                 name=f"a{i}.bqkv")
                qkv = ht.linear(y, qw, qb)
                nh = H // 128
                y = ht.fused_qkv_attention(qkv, nh, nh, 128, causal=True)
            w1 = ht.variable(torch.randn(2 * H, H, dtype=dtype) * 0.02,
                             name=f"l{i}.w1")
            y = ht.gelu(ht.linear(y, w1))
            w2 = ht.variable(torch.randn(H, 2 * H, dtype=dtype) * 0.02,
                             name=f"l{i}.w2")
            cur = ht.add(cur, ht.linear(y, w2))
        if vce:
            labels = ht.placeholder((B * S,), dtype=torch.int64,
                                    name="labels")
            lmw = ht.variable(torch.randn(V, H, dtype=dtype) * 0.02,
                              name="lm.w")
            logits = ht.linear(ht.reshape(cur, (B * S, H)), lmw)
            per = vocab_parallel_cross_entropy(logits, labels, V)
            loss = ht.reduce_mean(per)
        else:
            labels = None
            loss = ht.reduce_mean(ht.mul(cur, cur))
        train_op = Adam(lr=1e-4).minimize(loss)
    finally:
        pop_graph()
    return g, feed_key, labels, loss, train_op


def probe(tag, embed=False, vce=False, attn=False, steps=6):
    g, fk, lab, loss, train_op = build(embed, vce, attn)
    tr = Trainer(g, {"loss": loss, "train_op": train_op}, dev)
    adams = [op for op in g.ops if op.type == "AdamStep"]
    verdict = "clean"
    for i in range(steps):
        feed = {}
        if fk.meta.dtype == torch.int64:
            feed[fk] = torch.randint(0, V, (B, S), device=dev)
        else:
            feed[fk] = torch.randn(B, S, H, dtype=dtype, device=dev)
        if lab is not None:
            feed[lab] = torch.randint(0, V, (B * S,), device=dev)
        lv = tr.step(feed)
        torch.cuda.synchronize()
        nbad = sum(1 for op in adams if "m" in op.interface.state and not (
            torch.isfinite(op.interface.state["m"]).all()
            and torch.isfinite(op.interface.state["v"]).all()))
        if nbad or not torch.isfinite(lv.float()):
            verdict = f"BAD@step{i} nbad={nbad} loss={float(lv.float()):.3f}"
            break
    print(f"{tag}: {verdict}", flush=True)
    del tr, g
    torch.cuda.empty_cache()


def probe_real(tag, **kw):
    from hetu_amd.models.gpt import GPTConfig, build_gpt_train_graph
    torch.manual_seed(1234)
    cfg = GPTConfig(n_layer=L, n_head=H // 128, n_kv_head=H // 128,
                    hidden=H, ffn_hidden=4 * H, vocab=V, max_seq=S)
    g, h = build_gpt_train_graph(cfg, micro_batch=B, seq_len=S,
                                 dtype=dtype, lr=1e-4, **kw)
    tr = Trainer(g, h, dev)
    adams = [op for op in g.ops if op.type == "AdamStep"]
    verdict = "clean"
    for i in range(6):
        feed = {h["input_ids"]: torch.randint(0, V, (B, S), device=dev),
                h["labels"]: torch.randint(0, V, (B * S,), device=dev)}
        lv = tr.step(feed)
        torch.cuda.synchronize()
        nmv = sum(1 for op in adams if "m" in op.interface.state and not (
            torch.isfinite(op.interface.state["m"]).all()
            and torch.isfinite(op.interface.state["v"]).all()))
        nma = sum(1 for op in adams if "master" in op.interface.state
                  and not torch.isfinite(op.interface.state["master"]).all())
        npb = sum(1 for p in g.parameters
                  if not torch.isfinite(p.get_data().float()).all())
        print(f"  [{tag}] step {i}: loss={float(lv.float()):.3f} "
              f"mv={nmv} master={nma} params={npb}", flush=True)
        if nmv or nma or npb or not torch.isfinite(lv.float()):
            verdict = f"BAD@step{i}"
            break
    print(f"{tag}: {verdict}", flush=True)
    del tr, g
    torch.cuda.empty_cache()


probe("gptblock", embed=True, vce=True, attn="gptblock")
probe_real("real-gpt")
os.environ["HETU_AMD_FUSED_ATTN"] = "0"
probe_real("real-gpt-nofusedattn")
os.environ["HETU_AMD_FUSED_ATTN"] = "1"


# ---- monkeypatch bisect inside the real builder ----
import hetu_amd.models.gpt as GM


def patched_embed_fwd(self, input_ids):
    return self.wte(input_ids)          # drop wpe add


def patched_block_fwd(self, x, B, S):
    return ht.add(x, self.mlp(self.ln2(x)))   # drop attention half


orig_embed = GM.GPTEmbedding.forward
orig_block = GM.GPTBlock.forward

GM.GPTEmbedding.forward = patched_embed_fwd
probe_real("real-nowpe")
GM.GPTEmbedding.forward = orig_embed

GM.GPTBlock.forward = patched_block_fwd
probe_real("real-noattn")
GM.GPTBlock.forward = orig_block

os.environ["HETU_AMD_TORCH_FALLBACK"] = (
    "fa,qkvfa,ln,rms,gelu,silu,swiglu,adam,ce,vce,embed,softmax,rope,"
    "dropout")
import importlib
import hetu_amd.ops.functional as FU
FU._TORCH_FB = frozenset(os.environ["HETU_AMD_TORCH_FALLBACK"].split(","))
probe_real("real-allfallback")


def attn_nocore_fwd(self, x, B, S):
    hl, dh = self.h_local, self.dh
    qkv = self.wqkv(x)
    o = ht.slice_(qkv, 2, 0, hl * dh)     # drop the attention op itself
    return self.wo(o)


orig_attn = GM.GPTAttention.forward
GM.GPTAttention.forward = attn_nocore_fwd
FU._TORCH_FB = frozenset()
probe_real("real-attn-nocore")
GM.GPTAttention.forward = orig_attn

import hetu_amd.nn.parallel as NP
orig_row_fwd = NP.RowParallelLinear.forward


def row_fused_bias_fwd(self, x):
    spec = self.spec
    y = ht.linear(x, self.weight, self.bias)
    if spec.tp > 1:
        dst = (spec.ds_activation_sp(0, 1) if spec.sequence_parallel
               else spec.ds_activation(0))
        y = ht.comm(y, dst, name="row_reduce")
    return y


NP.RowParallelLinear.forward = row_fused_bias_fwd
probe_real("real-row-fusedbias")
NP.RowParallelLinear.forward = orig_row_fwd
