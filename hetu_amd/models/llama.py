"""Llama model family on the define-and-run graph + TP/SP parallel layers.

Reference parity: python/hetu/models/llama/llama_model.py (LlamaLMHeadModel
:446 — RMSNorm, RoPE, SwiGLU MLP, vocab-parallel embedding/lm-head + CE) and
the 4D config generators.  MI355X-native: all hot ops hit the hand-written
HIP kernel set (rmsnorm/rope/swiglu/flash-attn/vocab-CE); TP collectives
ride RCCL over xGMI via CommOp deduction.
"""
from __future__ import annotations

import dataclasses
import os
import math
from typing import Dict, Optional

import torch

from ..graph.graph import DefineAndRunGraph, push_graph, pop_graph
from ..graph.ops import api as ht
from ..graph.ops.optim import Adam
from ..nn import init
from ..nn.module import Module, ModuleList
from ..nn.parallel import (ColumnParallelLinear, ParallelRMSNorm,
                           ParallelSpec, RowParallelLinear,
                           VocabParallelEmbedding,
                           vocab_parallel_cross_entropy)


@dataclasses.dataclass
class LlamaConfig:
    n_layer: int = 12
    n_head: int = 12
    n_kv_head: int = 12
    hidden: int = 768
    ffn_hidden: int = 2048      # per-branch (gate/up) width
    vocab: int = 32000
    max_seq: int = 2048
    rope_theta: float = 10000.0
    rms_eps: float = 1e-6
    init_std: float = 0.02
    dropout: float = 0.0


LLAMA_CONFIGS = {
    "llama-tiny": LlamaConfig(n_layer=2, n_head=4, n_kv_head=4, hidden=128,
                              ffn_hidden=352, vocab=1024, max_seq=256),
    "llama-7b": LlamaConfig(n_layer=32, n_head=32, n_kv_head=32, hidden=4096,
                            ffn_hidden=11008, vocab=32000, max_seq=4096),
    "llama-13b": LlamaConfig(n_layer=40, n_head=40, n_kv_head=40,
                             hidden=5120, ffn_hidden=13824, vocab=32000,
                             max_seq=4096),
}


def rope_tables(cfg: LlamaConfig, seq_len: int, dtype=torch.float32,
                offset: int = 0):
    """offset: global position of the first local token (context
    parallelism gives each cp rank its own chunk of positions)."""
    dh = cfg.hidden // cfg.n_head
    inv = 1.0 / (cfg.rope_theta
                 ** (torch.arange(0, dh, 2, dtype=torch.float32) / dh))
    t = torch.arange(offset, offset + seq_len, dtype=torch.float32)
    freqs = torch.outer(t, inv)                  # [S, dh/2]
    return torch.cos(freqs).to(dtype), torch.sin(freqs).to(dtype)


class LlamaAttention(Module):
    def __init__(self, cfg: LlamaConfig, spec: ParallelSpec, cos, sin,
                 layer_idx: int, dtype):
        super().__init__()
        self.cfg, self.spec = cfg, spec
        tp = spec.tp
        assert cfg.n_head % tp == 0 and cfg.n_kv_head % tp == 0
        self.dh = cfg.hidden // cfg.n_head
        self.h_local = cfg.n_head // tp
        self.kv_local = cfg.n_kv_head // tp
        qkv_out = (cfg.n_head + 2 * cfg.n_kv_head) * self.dh
        p = f"l{layer_idx}.attn"
        proj_std = cfg.init_std / math.sqrt(2 * cfg.n_layer)
        self.wqkv = ColumnParallelLinear(
            cfg.hidden, qkv_out, spec, bias=False, dtype=dtype,
            name=f"{p}.wqkv", init_std=cfg.init_std,
            sections=[cfg.n_head * self.dh, cfg.n_kv_head * self.dh,
                      cfg.n_kv_head * self.dh])
        self.wo = RowParallelLinear(
            cfg.hidden, cfg.hidden, spec, bias=False, dtype=dtype,
            name=f"{p}.wo", init_std=proj_std)
        self.cos, self.sin = cos, sin

    def forward(self, x, B, S):
        spec, cfg = self.spec, self.cfg
        hl, kl, dh = self.h_local, self.kv_local, self.dh
        qkv = self.wqkv(x)                       # [B,S,(h+2kv)*dh / tp]
        if spec.cp == 1 and dh == 128 and os.environ.get(
                "HETU_AMD_FUSED_ATTN", "1") == "1":
            # fused path: in-place RoPE on the q|k sections + strided
            # flash attention, zero layout copies
            o = ht.fused_qkv_attention(qkv, hl, kl, dh, self.cos,
                                       self.sin, causal=True)
            return self.wo(o)
        ds_head = spec._ds({0: spec.dp, 1: spec.cp, 2: spec.tp},
                            [0, 1, 2])
        q = ht.reshape(ht.slice_(qkv, 2, 0, hl * dh), (B, S, hl, dh),
                       ds=ds_head)
        k = ht.reshape(ht.slice_(qkv, 2, hl * dh, kl * dh), (B, S, kl, dh),
                       ds=ds_head)
        v = ht.reshape(ht.slice_(qkv, 2, (hl + kl) * dh, kl * dh),
                       (B, S, kl, dh), ds=ds_head)
        q = ht.rotary(q, self.cos, self.sin)
        k = ht.rotary(k, self.cos, self.sin)
        q = ht.transpose(q, 1, 2)                # [B,hl,S,dh]
        k = ht.transpose(k, 1, 2)
        v = ht.transpose(v, 1, 2)
        if spec.cp > 1:
            o = ht.ring_attention(q, k, v, spec.cp_ranks(), causal=True)
        else:
            o = ht.attention(q, k, v, causal=True)
        o = ht.transpose(o, 1, 2)
        o = ht.reshape(o, (B, S, hl * dh),
                       ds=spec._ds({0: spec.dp, 1: spec.cp, 2: spec.tp},
                                   [0, 1, 2]))
        return self.wo(o)


class LlamaMLP(Module):
    def __init__(self, cfg: LlamaConfig, spec: ParallelSpec, layer_idx: int,
                 dtype):
        super().__init__()
        p = f"l{layer_idx}.mlp"
        proj_std = cfg.init_std / math.sqrt(2 * cfg.n_layer)
        # gate & up fused into one column-parallel GEMM; swiglu kernel
        # consumes the concatenated halves (SwiGLU.cu parity)
        self.w_in = ColumnParallelLinear(
            cfg.hidden, 2 * cfg.ffn_hidden, spec, bias=False, dtype=dtype,
            name=f"{p}.w_in", init_std=cfg.init_std,
            sections=[cfg.ffn_hidden, cfg.ffn_hidden])
        self.w_out = RowParallelLinear(
            cfg.ffn_hidden, cfg.hidden, spec, bias=False, dtype=dtype,
            name=f"{p}.w_out", init_std=proj_std)

    def forward(self, x):
        h = self.w_in(x)            # [B,S,2*ffn/tp] (gate|up local halves)
        h = ht.swiglu(h)
        return self.w_out(h)


class LlamaBlock(Module):
    def __init__(self, cfg, spec, cos, sin, layer_idx, dtype):
        super().__init__()
        self.ln1 = ParallelRMSNorm(cfg.hidden, spec, cfg.rms_eps, dtype,
                                   name=f"l{layer_idx}.ln1")
        self.attn = LlamaAttention(cfg, spec, cos, sin, layer_idx, dtype)
        self.ln2 = ParallelRMSNorm(cfg.hidden, spec, cfg.rms_eps, dtype,
                                   name=f"l{layer_idx}.ln2")
        self.mlp = LlamaMLP(cfg, spec, layer_idx, dtype)

    def forward(self, x, B, S):
        x = ht.add(x, self.attn(self.ln1(x), B, S))
        x = ht.add(x, self.mlp(self.ln2(x)))
        return x

    def forward_chain(self, x, delta, B, S):
        """Pre-norm chain with the residual add fused into each RMSNorm
        (FusedAddRMSOp) — see GPTBlock.forward_chain."""
        if delta is None:
            y1, s1 = self.ln1(x), x
        else:
            y1, s1 = ht.fused_add_rms(x, delta, self.ln1.weight,
                                      self.ln1.eps)
        a = self.attn(y1, B, S)
        y2, s2 = ht.fused_add_rms(s1, a, self.ln2.weight, self.ln2.eps)
        return s2, self.mlp(y2)


class LlamaLMHeadModel(Module):
    """Builds ops in the current graph; forward(input_ids, labels) ->
    (loss, logits)."""

    def __init__(self, cfg: LlamaConfig, spec: Optional[ParallelSpec] = None,
                 micro_batch: int = 1, seq_len: int = 128,
                 dtype=torch.bfloat16, recompute: bool = False):
        super().__init__()
        spec = spec or ParallelSpec()
        self.cfg, self.spec = cfg, spec
        self.recompute = recompute
        self.B, self.S = micro_batch, seq_len
        self.dtype = dtype
        cos_d, sin_d = rope_tables(cfg, seq_len, torch.float32,
                                   offset=spec.my_cp_index() * seq_len)
        self.cos = ht.variable(cos_d, name="rope.cos", requires_grad=False,
                               ds=spec.ds_weight_dup(),
                               device_group=spec.device_group)
        self.sin = ht.variable(sin_d, name="rope.sin", requires_grad=False,
                               ds=spec.ds_weight_dup(),
                               device_group=spec.device_group)
        self.wte = VocabParallelEmbedding(cfg.vocab, cfg.hidden, spec,
                                          dtype=dtype, name="wte",
                                          init_std=cfg.init_std)
        self.layers = ModuleList([
            LlamaBlock(cfg, spec, self.cos, self.sin, i, dtype)
            for i in range(cfg.n_layer)])
        self.lnf = ParallelRMSNorm(cfg.hidden, spec, cfg.rms_eps, dtype,
                                   name="lnf")
        self.lm_head = ColumnParallelLinear(
            cfg.hidden, cfg.vocab, spec, bias=False, dtype=dtype,
            name="lm_head", init_std=cfg.init_std)

    def forward(self, input_ids, labels=None):
        import contextlib
        B, S, cfg, spec = self.B, self.S, self.cfg, self.spec
        x = self.wte(input_ids)
        g = x.graph
        import os as _os
        fused_ln = _os.environ.get("HETU_AMD_FUSED_ADDLN", "1") == "1" \
            and not (spec.sequence_parallel and spec.tp > 1)
        delta = None
        for i, blk in enumerate(self.layers):
            cm = g.recompute_scope(i) if self.recompute \
                else contextlib.nullcontext()
            with cm:
                if fused_ln:
                    x, delta = blk.forward_chain(x, delta, B, S)
                else:
                    x = blk(x, B, S)
        if fused_ln and delta is not None:
            x, _ = ht.fused_add_rms(x, delta, self.lnf.weight,
                                    self.lnf.eps)
        else:
            x = self.lnf(x)
        if spec.sequence_parallel and spec.tp > 1:
            # gather the seq shards back before the LM head (the head's
            # column-parallel GEMM wants the full token set per rank)
            x = ht.comm(x, spec.ds_activation(0), name="sp_final_allgather")
        logits = self.lm_head(
            ht.reshape(x, (B * S, cfg.hidden), ds=spec.ds_tokens(0)))
        if labels is None:
            return None, logits
        per_tok = vocab_parallel_cross_entropy(logits, labels, cfg.vocab)
        loss = ht.reduce_mean(per_tok)
        return loss, logits


def build_llama_pipeline_stage(cfg: LlamaConfig, pspec, micro_batch: int,
                               seq_len: int, dtype=torch.bfloat16,
                               lr: float = 1e-4,
                               stage_layers=None, zero: bool = False):
    """Build THIS rank's pipeline-stage subgraph (see parallel.pipeline).

    Returns a StageModule whose graph exposes fwd (act_out|loss), bwd
    (dx + param grads via grad_in) and update (train_op fed by grad
    placeholders; dp-allreduce of accumulated grads happens here, once per
    step) fetch sets.  zero=True shards the optimizer states over the
    stage's dp group (ZeroAdamStepOp reduce-scatters the still-partial
    accumulated grads itself — no bucket allreduce)."""
    from ..parallel.pipeline import StageModule
    B, S = micro_batch, seq_len
    sid = pspec.my_stage()
    spec = pspec.stage_spec(sid)
    parts = stage_layers or pspec.partition_layers(cfg.n_layer)
    my_layers = parts[sid]
    is_first, is_last = sid == 0, sid == pspec.pp - 1

    g = DefineAndRunGraph(f"llama_stage{sid}")
    push_graph(g)
    try:
        h: Dict = {"act_shape": (B, S, cfg.hidden), "act_dtype": dtype}
        ds_in = spec.ds_activation(0)
        cos_d, sin_d = rope_tables(cfg, seq_len, torch.float32,
                                   offset=spec.my_cp_index() * seq_len)
        cos = ht.variable(cos_d, name="rope.cos", requires_grad=False,
                          ds=spec.ds_weight_dup(),
                          device_group=spec.device_group)
        sin = ht.variable(sin_d, name="rope.sin", requires_grad=False,
                          ds=spec.ds_weight_dup(),
                          device_group=spec.device_group)
        if is_first:
            input_ids = ht.placeholder((B, S), dtype=torch.int64,
                                       name="input_ids", ds=ds_in,
                                       device_group=spec.device_group)
            wte = VocabParallelEmbedding(cfg.vocab, cfg.hidden, spec,
                                         dtype=dtype, name="wte",
                                         init_std=cfg.init_std)
            x = wte(input_ids)
            h["input_ids"] = input_ids
        else:
            act_in = ht.placeholder((B, S, cfg.hidden), dtype=dtype,
                                    name="act_in", ds=ds_in,
                                    device_group=spec.device_group)
            x = act_in
            h["act_in"] = act_in
        blocks = [LlamaBlock(cfg, spec, cos, sin, li, dtype)
                  for li in my_layers]
        for blk in blocks:
            x = blk(x, B, S)
        if is_last:
            labels = ht.placeholder((B * S,), dtype=torch.int64,
                                    name="labels", ds=spec.ds_tokens(0),
                                    device_group=spec.device_group)
            lnf = ParallelRMSNorm(cfg.hidden, spec, cfg.rms_eps, dtype,
                                  name="lnf")
            lm_head = ColumnParallelLinear(
                cfg.hidden, cfg.vocab, spec, bias=False, dtype=dtype,
                name="lm_head", init_std=cfg.init_std)
            xo = lnf(x)
            logits = lm_head(
                ht.reshape(xo, (B * S, cfg.hidden), ds=spec.ds_tokens(0)))
            per_tok = vocab_parallel_cross_entropy(logits, labels, cfg.vocab)
            loss = ht.reduce_mean(per_tok)
            h["labels"] = labels
            h["loss"] = loss
            h["logits"] = logits
        else:
            h["act_out"] = x

        params = list(g.parameters)
        h["params"] = params
        # ---- backward fetch set ----
        xs = params + ([] if is_first else [h["act_in"]])
        if is_last:
            # d-loss seed placeholder so an fp16 GradScaler can scale the
            # backward (PipelineRunner feeds scaler.scale, or 1.0); grads
            # are unscaled after accumulation (engine/amp.py unscale_).
            loss_seed = ht.placeholder(tuple(loss.shape),
                                       dtype=torch.float32,
                                       name="loss_seed", ds=loss.ds,
                                       device_group=spec.device_group)
            h["loss_seed"] = loss_seed
            grads = g.gradients([loss], xs, grad_ys=[loss_seed])
        else:
            grad_in = ht.placeholder((B, S, cfg.hidden), dtype=dtype,
                                     name="grad_in", ds=ds_in,
                                     device_group=spec.device_group)
            h["grad_in"] = grad_in
            grads = g.gradients([x], xs, grad_ys=[grad_in])
        h["param_grads"] = grads[:len(params)]
        if not is_first:
            h["dx"] = grads[len(params)]
        # ---- update graph: grad placeholders -> (dp allreduce) -> Adam ---
        from ..graph.ops.optim import (AdamStepOp, GroupOp, ZeroAdamStepOp,
                                       make_grad_buckets)
        from ..graph.ops.basics import _make
        from ..graph.ops.comm import make_comm
        grad_phs, updates = [], []
        opt_attrs = {"lr": lr, "beta1": 0.9, "beta2": 0.999, "eps": 1e-8,
                     "weight_decay": 0.0}
        use_zero = zero and spec.dp > 1
        phs, pend = [], []
        for p, pg in zip(params, h["param_grads"]):
            gds = pg.ds if pg is not None else None
            ph = ht.placeholder(tuple(p.shape), dtype=torch.float32,
                                name=f"gbuf_{p.name}", ds=gds,
                                device_group=spec.device_group)
            grad_phs.append(ph)
            phs.append((p, ph))
            if not use_zero and gds is not None and p.ds is not None \
                    and not gds.check_equal(p.ds) \
                    and gds.check_allreduce(p.ds):
                pend.append((p, ph))
        reduced = make_grad_buckets(g, pend, name_prefix="gred_bucket") \
            if pend else {}
        for p, ph in phs:
            gt = reduced.get(p.id, ph)
            if use_zero:
                # partial grads flow straight in: ZeroAdamStep does the
                # reduce-scatter + local Adam + allgather itself
                updates.append(_make(g, ZeroAdamStepOp(), [p, gt],
                                     dict(opt_attrs),
                                     name=f"zadam_{p.name}").output())
                continue
            if gt is ph and ph.ds is not None and p.ds is not None \
                    and not ph.ds.check_equal(p.ds):
                gt = make_comm(g, ph, p.ds, name=f"gred_{p.name}")
            updates.append(_make(g, AdamStepOp(), [p, gt], dict(opt_attrs),
                                 name=f"adam_{p.name}").output())
        h["grad_phs"] = grad_phs
        h["train_op"] = _make(g, GroupOp(), updates, name="train_op").output()
    finally:
        pop_graph()
    return StageModule(g, h)


def build_llama_train_graph(cfg: LlamaConfig, micro_batch: int, seq_len: int,
                            dtype=torch.bfloat16, lr: float = 1e-4,
                            spec: Optional[ParallelSpec] = None,
                            graph: Optional[DefineAndRunGraph] = None,
                            zero: bool = False, recompute: bool = False
                            ) -> (DefineAndRunGraph, Dict):
    g = graph or DefineAndRunGraph("llama_train")
    spec = spec or ParallelSpec()
    push_graph(g)
    try:
        ds_in = spec.ds_activation(0)
        input_ids = ht.placeholder((micro_batch, seq_len), dtype=torch.int64,
                                   name="input_ids", ds=ds_in,
                                   device_group=spec.device_group)
        labels = ht.placeholder((micro_batch * seq_len,), dtype=torch.int64,
                                name="labels", ds=spec.ds_tokens(0),
                                device_group=spec.device_group)
        model = LlamaLMHeadModel(cfg, spec, micro_batch, seq_len, dtype,
                                 recompute=recompute)
        loss, logits = model(input_ids, labels)
        loss_report = loss
        if spec.num_devices > 1:
            loss_report = ht.comm(
                loss, spec._ds({-1: spec.num_devices}, [-1]),
                name="loss_allreduce")
        opt = Adam(lr=lr, zero=zero)
        train_op = opt.minimize(loss)
    finally:
        pop_graph()
    return g, {"input_ids": input_ids, "labels": labels,
               "loss": loss_report, "logits": logits, "train_op": train_op,
               "optimizer": opt, "model": model}
