"""GPT model family on the define-and-run graph + TP/SP/CP parallel layers.

Reference parity: python/hetu/models/gpt/gpt_model.py (GPTLMHeadModel) — a
pre-LN transformer with learned position embeddings, fused-HIP LayerNorm,
flash attention, GELU MLP, and vocab-parallel CE loss.  The 7B config
matches the reference CI "gpt 7b" shape (32 layers x 4096 hidden x 32
heads, tests/ci_test/scripts/pssh_train_hetu.sh).  Parallelism (dp x cp x
tp, zero) comes from a ParallelSpec like the Llama family; pipeline stages
via build_gpt_pipeline_stage.
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
from ..nn.parallel import (ColumnParallelLinear, ParallelLayerNorm,
                           ParallelSpec, RowParallelLinear,
                           VocabParallelEmbedding,
                           vocab_parallel_cross_entropy)


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
    # MoE (HetuMoE parity): >0 experts switches the MLP to a gated
    # expert FFN with EP over the device group + (hierarchical) all-to-all
    moe_experts: int = 0
    moe_k: int = 2
    moe_capacity: float = 1.25


GPT_CONFIGS = {
    "gpt2-345m": GPTConfig(n_layer=24, n_head=16, n_kv_head=16, hidden=1024,
                           ffn_hidden=4096, vocab=50304, max_seq=1024),
    "gpt3-7b": GPTConfig(n_layer=32, n_head=32, n_kv_head=32, hidden=4096,
                         ffn_hidden=16384, vocab=50304, max_seq=2048),
    "gpt3-13b": GPTConfig(n_layer=40, n_head=40, n_kv_head=40, hidden=5120,
                          ffn_hidden=20480, vocab=50304, max_seq=2048),
    # BASELINE config 4: GPT-MoE 8 x 1.3B experts (HetuMoE hierarchical a2a)
    "gpt-moe-8x1.3b": GPTConfig(n_layer=24, n_head=16, n_kv_head=16,
                                hidden=2048, ffn_hidden=8192, vocab=50304,
                                max_seq=2048, moe_experts=8, moe_k=2),
    # CI-only config for the multi-process CPU smoke of the bench path
    "gpt-tiny": GPTConfig(n_layer=2, n_head=2, n_kv_head=2, hidden=64,
                          ffn_hidden=128, vocab=128, max_seq=64),
}


class GPTAttention(Module):
    def __init__(self, cfg: GPTConfig, spec: ParallelSpec, layer_idx: int,
                 dtype):
        super().__init__()
        self.cfg, self.spec = cfg, spec
        tp = spec.tp
        assert cfg.n_head % tp == 0
        self.dh = cfg.hidden // cfg.n_head
        self.h_local = cfg.n_head // tp
        p = f"h{layer_idx}.attn"
        proj_std = cfg.init_std / math.sqrt(2 * cfg.n_layer)
        self.wqkv = ColumnParallelLinear(
            cfg.hidden, 3 * cfg.hidden, spec, bias=True, dtype=dtype,
            name=f"{p}.wqkv", init_std=cfg.init_std,
            sections=[cfg.hidden, cfg.hidden, cfg.hidden])
        self.wo = RowParallelLinear(
            cfg.hidden, cfg.hidden, spec, bias=True, dtype=dtype,
            name=f"{p}.wo", init_std=proj_std)

    def forward(self, x, B, S):
        spec = self.spec
        hl, dh = self.h_local, self.dh
        qkv = self.wqkv(x)
        if spec.cp == 1 and dh == 128 and os.environ.get(
                "HETU_AMD_FUSED_ATTN", "1") == "1":
            # fused path: attention reads q/k/v straight out of the qkv
            # GEMM output; no slice/transpose copies (the CPU fallback
            # computes the identical reference math)
            o = ht.fused_qkv_attention(qkv, hl, hl, dh, causal=True)
            return self.wo(o)
        ds_head = spec._ds({0: spec.dp, 1: spec.cp, 2: spec.tp}, [0, 1, 2])
        q = ht.reshape(ht.slice_(qkv, 2, 0, hl * dh), (B, S, hl, dh),
                       ds=ds_head)
        k = ht.reshape(ht.slice_(qkv, 2, hl * dh, hl * dh), (B, S, hl, dh),
                       ds=ds_head)
        v = ht.reshape(ht.slice_(qkv, 2, 2 * hl * dh, hl * dh),
                       (B, S, hl, dh), ds=ds_head)
        q = ht.transpose(q, 1, 2)
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


class GPTMLP(Module):
    def __init__(self, cfg: GPTConfig, spec: ParallelSpec, layer_idx: int,
                 dtype):
        super().__init__()
        p = f"h{layer_idx}.mlp"
        proj_std = cfg.init_std / math.sqrt(2 * cfg.n_layer)
        self.wfc = ColumnParallelLinear(
            cfg.hidden, cfg.ffn_hidden, spec, bias=True, dtype=dtype,
            name=f"{p}.wfc", init_std=cfg.init_std)
        self.wproj = RowParallelLinear(
            cfg.ffn_hidden, cfg.hidden, spec, bias=True, dtype=dtype,
            name=f"{p}.wproj", init_std=proj_std)

    def forward(self, x):
        if self.wfc.spec.tp == 1 and os.environ.get(
                "HETU_AMD_FUSED_MLP", "0") == "1":
            # hipBLASLt epilogue fusion (GELU_AUX_BIAS fwd + DGELU_BGRAD
            # bwd).  Opt-in: the ROCm 7.2 hipBLASLt Tensile catalog has no
            # epilogue solutions at transformer shapes on gfx950 (probed,
            # scripts/lt_probe2.py), and the composed-graph fallback
            # measured ~1.3% faster than FusedMLPOp's fallback composition.
            return ht.fused_mlp(x, self.wfc.weight, self.wfc.bias,
                                self.wproj.weight, self.wproj.bias)
        return self.wproj(ht.gelu(self.wfc(x)))


class GPTMoEMLP(Module):
    """MoE expert FFN in place of the dense MLP (HetuMoE moe_layer.py
    parity): tokens flatten to [B*S, h], dispatch over the EP group."""

    def __init__(self, cfg: GPTConfig, spec: ParallelSpec, layer_idx: int,
                 dtype):
        super().__init__()
        from ..nn.moe import MoEMLP
        self.spec = spec
        self.moe = MoEMLP(cfg.hidden, cfg.ffn_hidden, cfg.moe_experts,
                          spec=spec if spec.num_devices > 1 else None,
                          k=cfg.moe_k, capacity_factor=cfg.moe_capacity,
                          dtype=dtype, name=f"h{layer_idx}.moe")
        self.hidden = cfg.hidden

    def forward(self, x):
        B, S = x.shape[0], x.shape[1]
        flat = ht.reshape(x, (B * S, self.hidden),
                          ds=self.spec.ds_tokens(0))
        y = self.moe(flat)
        return ht.reshape(y, (B, S, self.hidden),
                          ds=self.spec.ds_activation(0))


class GPTBlock(Module):
    def __init__(self, cfg, spec, layer_idx, dtype):
        super().__init__()
        self.ln1 = ParallelLayerNorm(cfg.hidden, spec, 1e-5, dtype,
                                     name=f"h{layer_idx}.ln1")
        self.attn = GPTAttention(cfg, spec, layer_idx, dtype)
        self.ln2 = ParallelLayerNorm(cfg.hidden, spec, 1e-5, dtype,
                                     name=f"h{layer_idx}.ln2")
        if cfg.moe_experts > 0:
            self.mlp = GPTMoEMLP(cfg, spec, layer_idx, dtype)
        else:
            self.mlp = GPTMLP(cfg, spec, layer_idx, dtype)

    def forward(self, x, B, S):
        x = ht.add(x, self.attn(self.ln1(x), B, S))
        x = ht.add(x, self.mlp(self.ln2(x)))
        return x

    def forward_chain(self, x, delta, B, S):
        """Pre-norm chain form: (residual sum, pending delta) in/out, with
        the residual add fused into each LayerNorm (FusedAddLNOp) — no
        standalone elementwise adds in the block."""
        if delta is None:
            y1, s1 = self.ln1(x), x
        else:
            y1, s1 = ht.fused_add_ln(x, delta, self.ln1.weight,
                                     self.ln1.bias, self.ln1.eps)
        a = self.attn(y1, B, S)
        y2, s2 = ht.fused_add_ln(s1, a, self.ln2.weight, self.ln2.bias,
                                 self.ln2.eps)
        return s2, self.mlp(y2)


class GPTEmbedding(Module):
    """Token + learned position embeddings (positions offset per cp rank)."""

    def __init__(self, cfg: GPTConfig, spec: ParallelSpec, seq_len: int,
                 dtype):
        super().__init__()
        self.spec = spec
        self.vocab = cfg.vocab
        self.wte = VocabParallelEmbedding(cfg.vocab, cfg.hidden, spec,
                                          dtype=dtype, name="wte",
                                          init_std=cfg.init_std)
        wpe = init.normal((cfg.max_seq, cfg.hidden), std=cfg.init_std,
                          dtype=dtype, name="wpe.weight")
        off = spec.my_cp_index() * seq_len
        # local slice of the position table: grads stay per-rank and reduce
        # over the dup group like any duplicated parameter
        self.wpe = ht.variable(wpe[off:off + seq_len].contiguous(),
                               name="wpe.weight",
                               ds=spec.ds_weight_dup(),
                               device_group=spec.device_group)
        pos = torch.arange(seq_len, dtype=torch.int64)
        self.pos = ht.variable(pos, name="pos", requires_grad=False,
                               ds=spec.ds_weight_dup(),
                               device_group=spec.device_group)

    def forward(self, input_ids):
        x = self.wte(input_ids)
        p = ht.embedding(self.wpe, self.pos)
        return ht.add(x, p)


class GPTLMHeadModel(Module):
    def __init__(self, cfg: GPTConfig, spec: Optional[ParallelSpec] = None,
                 micro_batch: int = 1, seq_len: int = 128,
                 dtype=torch.bfloat16, recompute: bool = False):
        super().__init__()
        spec = spec or ParallelSpec()
        self.cfg, self.spec = cfg, spec
        self.recompute = recompute
        self.B, self.S = micro_batch, seq_len
        self.embed = GPTEmbedding(cfg, spec, seq_len, dtype)
        self.layers = ModuleList([GPTBlock(cfg, spec, i, dtype)
                                  for i in range(cfg.n_layer)])
        self.lnf = ParallelLayerNorm(cfg.hidden, spec, 1e-5, dtype,
                                     name="lnf")
        if cfg.tie_embeddings:
            # tied wte/lm_head: the SAME graph tensor feeds both the
            # embedding lookup and the output projection; autodiff sums
            # the two gradient paths (reference shared-weight semantics)
            self.lm_head = None
        else:
            self.lm_head = ColumnParallelLinear(
                cfg.hidden, cfg.vocab, spec, bias=False, dtype=dtype,
                name="lm_head", init_std=cfg.init_std)

    def forward(self, input_ids, labels=None):
        import contextlib
        B, S, cfg, spec = self.B, self.S, self.cfg, self.spec
        x = self.embed(input_ids)
        g = x.graph
        fused_ln = os.environ.get("HETU_AMD_FUSED_ADDLN", "1") == "1"
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
            x, _ = ht.fused_add_ln(x, delta, self.lnf.weight,
                                   self.lnf.bias, self.lnf.eps)
        else:
            x = self.lnf(x)
        xr = ht.reshape(x, (B * S, cfg.hidden), ds=spec.ds_tokens(0))
        if self.lm_head is None:
            logits = ht.linear(xr, self.embed.wte.weight)
        else:
            logits = self.lm_head(xr)
        if labels is None:
            return None, logits
        per_tok = vocab_parallel_cross_entropy(logits, labels, cfg.vocab)
        loss = ht.reduce_mean(per_tok)
        return loss, logits


def build_gpt_train_graph(cfg: GPTConfig, micro_batch: int, seq_len: int,
                          dtype=torch.bfloat16, lr: float = 1e-4,
                          dp: int = 1, device_group=None,
                          graph: Optional[DefineAndRunGraph] = None,
                          zero: bool = False,
                          spec: Optional[ParallelSpec] = None,
                          recompute: bool = False,
                          hetero=None
                          ) -> (DefineAndRunGraph, Dict):
    g = graph or DefineAndRunGraph("gpt_train")
    if spec is None:
        spec = ParallelSpec(dp=dp, device_group=device_group)
    push_graph(g)
    try:
        ds_in = spec.ds_activation(0)
        input_ids = ht.placeholder((micro_batch, seq_len),
                                   dtype=torch.int64, name="input_ids",
                                   ds=ds_in, device_group=spec.device_group)
        labels = ht.placeholder((micro_batch * seq_len,), dtype=torch.int64,
                                name="labels", ds=spec.ds_tokens(0),
                                device_group=spec.device_group)
        model = GPTLMHeadModel(cfg, spec, micro_batch, seq_len, dtype,
                               recompute=recompute)
        loss, _ = model(input_ids, labels)
        loss_report = loss
        if spec.num_devices > 1:
            loss_report = ht.comm(
                loss, spec._ds({-1: spec.num_devices}, [-1]),
                name="loss_allreduce")
        opt = Adam(lr=lr, zero=zero, hetero=hetero)
        train_op = opt.minimize(loss)
    finally:
        pop_graph()
    return g, {"input_ids": input_ids, "labels": labels,
               "loss": loss_report, "train_op": train_op,
               "optimizer": opt, "model": model}


def build_gpt_pipeline_stage(cfg: GPTConfig, pspec, micro_batch: int,
                             seq_len: int, dtype=torch.bfloat16,
                             lr: float = 1e-4, stage_layers=None,
                             zero: bool = False):
    """This rank's pipeline-stage subgraph (see parallel.pipeline)."""
    from ..parallel.pipeline import StageModule
    B, S = micro_batch, seq_len
    sid = pspec.my_stage()
    spec = pspec.stage_spec(sid)
    parts = stage_layers or pspec.partition_layers(cfg.n_layer)
    my_layers = parts[sid]
    is_first, is_last = sid == 0, sid == pspec.pp - 1

    g = DefineAndRunGraph(f"gpt_stage{sid}")
    push_graph(g)
    try:
        h: Dict = {"act_shape": (B, S, cfg.hidden), "act_dtype": dtype}
        ds_in = spec.ds_activation(0)
        if is_first:
            input_ids = ht.placeholder((B, S), dtype=torch.int64,
                                       name="input_ids", ds=ds_in,
                                       device_group=spec.device_group)
            embed = GPTEmbedding(cfg, spec, S, dtype)
            x = embed(input_ids)
            h["input_ids"] = input_ids
            if cfg.tie_embeddings:
                h["tied_name"] = embed.wte.weight.name
        else:
            act_in = ht.placeholder((B, S, cfg.hidden), dtype=dtype,
                                    name="act_in", ds=ds_in,
                                    device_group=spec.device_group)
            x = act_in
            h["act_in"] = act_in
        for li in my_layers:
            x = GPTBlock(cfg, spec, li, dtype)(x, B, S)
        if is_last:
            labels = ht.placeholder((B * S,), dtype=torch.int64,
                                    name="labels", ds=spec.ds_tokens(0),
                                    device_group=spec.device_group)
            lnf = ParallelLayerNorm(cfg.hidden, spec, 1e-5, dtype,
                                    name="lnf")
            xo = lnf(x)
            xr = ht.reshape(xo, (B * S, cfg.hidden), ds=spec.ds_tokens(0))
            if cfg.tie_embeddings and is_first:
                # pp == 1: the one stage holds wte — tie directly
                logits = ht.linear(xr, embed.wte.weight)
            elif cfg.tie_embeddings:
                # shared wte/lm_head across first/last stage: the last
                # stage holds its own copy (identical per-name init);
                # PipelineRunner p2p-sums the two stages' grads each step
                # (reference executable_graph.cc:929-933 shared-weight p2p)
                tied_w = VocabParallelEmbedding(
                    cfg.vocab, cfg.hidden, spec, dtype=dtype, name="wte",
                    init_std=cfg.init_std).weight
                h["tied_name"] = tied_w.name
                logits = ht.linear(xr, tied_w)
            else:
                lm_head = ColumnParallelLinear(
                    cfg.hidden, cfg.vocab, spec, bias=False, dtype=dtype,
                    name="lm_head", init_std=cfg.init_std)
                logits = lm_head(xr)
            per_tok = vocab_parallel_cross_entropy(logits, labels, cfg.vocab)
            loss = ht.reduce_mean(per_tok)
            h["labels"] = labels
            h["loss"] = loss
        else:
            h["act_out"] = x

        params = list(g.parameters)
        h["params"] = params
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
        from ..graph.ops.optim import (AdamStepOp, ZeroAdamStepOp, GroupOp,
                                       make_grad_buckets)
        from ..graph.ops.basics import _make
        from ..graph.ops.comm import make_comm
        grad_phs, updates = [], []
        opt_attrs = {"lr": lr, "beta1": 0.9, "beta2": 0.999, "eps": 1e-8,
                     "weight_decay": 0.0}
        cls = ZeroAdamStepOp if zero else AdamStepOp
        phs, pend = [], []
        for p, pg in zip(params, h["param_grads"]):
            gds = pg.ds if pg is not None else None
            ph = ht.placeholder(tuple(p.shape), dtype=torch.float32,
                                name=f"gbuf_{p.name}", ds=gds,
                                device_group=spec.device_group)
            grad_phs.append(ph)
            phs.append((p, ph))
            if not zero and gds is not None and p.ds is not None \
                    and not gds.check_equal(p.ds) \
                    and gds.check_allreduce(p.ds):
                pend.append((p, ph))
        reduced = make_grad_buckets(g, pend, name_prefix="gred_bucket") \
            if pend else {}
        for p, ph in phs:
            gt = reduced.get(p.id, ph)
            if gt is ph and not zero and ph.ds is not None \
                    and p.ds is not None and not ph.ds.check_equal(p.ds):
                gt = make_comm(g, ph, p.ds, name=f"gred_{p.name}")
            updates.append(_make(g, cls(), [p, gt], dict(opt_attrs),
                                 name=f"adam_{p.name}").output())
        h["grad_phs"] = grad_phs
        h["train_op"] = _make(g, GroupOp(), updates, name="train_op").output()
    finally:
        pop_graph()
    return StageModule(g, h)
