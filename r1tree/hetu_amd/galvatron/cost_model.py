"""Galvatron-style cost model, re-derived for MI355X.

Reference parity: tools/Galvatron (VLDB'23) — GalvatronProfiler per-layer
time/memory profiling (core/profiler.py:82-193) and cluster bandwidth
profiling via nccl-tests (:405-534).  Those A100 inputs are replaced by
MI355X-measured constants (this repo's rocprof runs + the microbenchmarks in
hetu_amd/galvatron/profiler.py, which refit them on real hardware):

  * bf16 GEMM (hipBLASLt, transformer shapes):   ~1.4e15 FLOP/s sustained
  * hand FA2 fwd (causal, S=2k-4k):              ~3.3e14 FLOP/s
  * hand FA2 bwd (causal):                       ~3.0e14 FLOP/s
  * HBM3E:                                        8 TB/s peak, ~6.3 measured
  * xGMI: 7 p2p links x ~153 GB/s per GPU; RCCL ring allreduce is per-link
    bound -> effective algorithm bandwidth ~= link_bw * n/(n-1) per GPU
  * HBM capacity: 288 GB/GPU (drives much larger micro-batches / fewer
    recompute layers than A100-class searches)
"""
from __future__ import annotations

import dataclasses
from typing import Dict, List, Optional


@dataclasses.dataclass
class HardwareModel:
    gemm_flops: float = 1.4e15        # sustained bf16 GEMM FLOP/s
    attn_fwd_flops: float = 3.3e14    # fa2 causal fwd
    attn_bwd_flops: float = 3.0e14    # fa2 causal bwd
    hbm_bw: float = 6.3e12            # B/s
    xgmi_link_bw: float = 140e9       # B/s per link, effective
    rccl_allreduce_bw: float = 220e9  # per-GPU bus bandwidth, large buffers
    rccl_p2p_bw: float = 140e9
    hbm_capacity: float = 288e9 * 0.92   # usable bytes
    kernel_overhead: float = 3e-6     # per kernel launch (graph-captured ~0)
    # runtime (not hardware) cost of the uncaptured pipeline path: one
    # python executor pass per micro-batch per stage; the dp/tp path is
    # hipGraph-captured and pays none of this.  Calibrate on hardware.
    pp_mb_overhead: float = 30e-3

    def allreduce_time(self, nbytes: float, n: int) -> float:
        if n <= 1 or nbytes == 0:
            return 0.0
        # ring: 2(n-1)/n x volume over the per-GPU bus bandwidth
        return 2 * (n - 1) / n * nbytes / self.rccl_allreduce_bw + 10e-6

    def reducescatter_time(self, nbytes: float, n: int) -> float:
        if n <= 1:
            return 0.0
        return (n - 1) / n * nbytes / self.rccl_allreduce_bw + 8e-6

    def p2p_time(self, nbytes: float) -> float:
        return nbytes / self.rccl_p2p_bw + 8e-6


@dataclasses.dataclass
class ModelShape:
    n_layer: int
    hidden: int
    ffn_hidden: int       # total ffn width per branch pair as used in GEMMs
    vocab: int
    n_head: int
    kind: str = "gpt"     # gpt: gelu-mlp (2 GEMMs 4h); llama: swiglu (2h ffn)

    @property
    def layer_params(self) -> int:
        if self.kind == "llama":
            mlp = 3 * self.hidden * self.ffn_hidden
        else:
            mlp = 2 * self.hidden * self.ffn_hidden
        return 4 * self.hidden * self.hidden + mlp + 2 * self.hidden

    @property
    def embed_params(self) -> int:
        return 2 * self.vocab * self.hidden   # wte + lm_head


@dataclasses.dataclass
class Strategy:
    dp: int = 1
    tp: int = 1
    pp: int = 1
    cp: int = 1
    micro_batch: int = 1          # per-dp-rank micro batch size
    num_micro_batches: int = 1
    zero: bool = False
    recompute_layers: int = 0     # layers rerun in backward

    @property
    def world(self):
        return self.dp * self.tp * self.pp * self.cp

    def name(self):
        tags = [f"dp{self.dp}"]
        if self.tp > 1:
            tags.append(f"tp{self.tp}")
        if self.pp > 1:
            tags.append(f"pp{self.pp}")
        if self.cp > 1:
            tags.append(f"cp{self.cp}")
        if self.zero:
            tags.append("zero")
        if self.recompute_layers:
            tags.append(f"ckpt{self.recompute_layers}")
        return "_".join(tags)


class CostModel:
    """Per-step time + per-GPU memory estimate for a (model, strategy)."""

    def __init__(self, model: ModelShape, seq_len: int,
                 hw: Optional[HardwareModel] = None):
        self.m = model
        self.s = seq_len
        self.hw = hw or HardwareModel()

    # ---- per-layer compute -----------------------------------------------
    def _layer_flops_fwd(self, tokens: int, tp: int, seq: int) -> float:
        m = self.m
        gemm = 2 * tokens * m.layer_params / tp
        attn = 4 * tokens * seq * m.hidden / tp * 0.5   # causal
        return gemm, attn

    def layer_time(self, tokens: int, tp: int, seq: int,
                   recompute: bool) -> float:
        hw = self.hw
        gemm_f, attn_f = self._layer_flops_fwd(tokens, tp, seq)
        t_fwd = gemm_f / hw.gemm_flops + attn_f / hw.attn_fwd_flops
        t_bwd = 2 * gemm_f / hw.gemm_flops + 2.5 * attn_f / hw.attn_bwd_flops
        t = t_fwd + t_bwd + (t_fwd if recompute else 0.0)
        # TP: 2 allreduces fwd + 2 bwd of [tokens, hidden] bf16
        if tp > 1:
            vol = tokens * self.m.hidden * 2
            t += 4 * hw.allreduce_time(vol, tp)
        return t

    def head_time(self, tokens: int, tp: int) -> float:
        f = 2 * tokens * self.m.vocab * self.m.hidden / tp
        return 3 * f / self.hw.gemm_flops

    # ---- per-layer activation memory (bf16, flash-attention) -------------
    def layer_act_bytes(self, tokens: int, tp: int, recompute: bool) -> float:
        if recompute:
            return 2 * tokens * self.m.hidden   # only the layer input
        m = self.m
        if m.kind == "llama":
            per_tok = m.hidden * (10 + 4) + 2 * m.ffn_hidden * 3
        else:
            per_tok = m.hidden * (10 + 4) + 4 * m.hidden * 2 * 2
        return tokens * (m.hidden * 4 / max(tp, 1) + per_tok * 2 / max(tp, 1))

    # ---- full step -------------------------------------------------------
    def evaluate(self, st: Strategy, global_batch: int) -> Dict:
        m, hw, s = self.m, self.hw, self.s
        assert global_batch % (st.dp * st.micro_batch) == 0, "batch split"
        num_mb = global_batch // (st.dp * st.micro_batch)
        layers_per_stage = (m.n_layer + st.pp - 1) // st.pp
        seq_local = s // st.cp
        tokens_mb = st.micro_batch * seq_local    # per rank per micro-batch

        rec = min(st.recompute_layers, layers_per_stage)
        plain = layers_per_stage - rec

        # per-micro-batch stage time (the pipeline's clock period)
        t_layer = self.layer_time(tokens_mb, st.tp, seq_local, False)
        t_layer_r = self.layer_time(tokens_mb, st.tp, seq_local, True)
        t_stage = plain * t_layer + rec * t_layer_r
        # cp ring: attention exchanged (cp-1) times; KV volume per step
        if st.cp > 1:
            kv_vol = 2 * tokens_mb * m.hidden * 2
            t_stage += 2 * (st.cp - 1) * hw.p2p_time(kv_vol)
        # last stage: head + loss
        t_head = self.head_time(tokens_mb, st.tp)
        # pipeline p2p per micro-batch boundary
        t_p2p = hw.p2p_time(tokens_mb * m.hidden * 2) if st.pp > 1 else 0.0

        # PipeDream-flush: (num_mb + pp - 1) periods of the slowest stage
        t_period = t_stage + t_head / max(st.pp, 1) + 2 * t_p2p
        t_compute = (num_mb + st.pp - 1) * t_period
        if st.pp > 1:
            t_compute += num_mb * hw.pp_mb_overhead

        # dp grad sync (once per step; fp32 buffers in the pp path)
        shard_params = layers_per_stage * m.layer_params / st.tp \
            + m.embed_params / st.tp / max(st.pp, 1)
        if st.dp * st.cp > 1:
            grad_bytes = shard_params * 2
            n = st.dp * st.cp
            if st.zero:
                t_sync = hw.reducescatter_time(grad_bytes, n) \
                    + hw.reducescatter_time(grad_bytes, n)  # + allgather
            else:
                t_sync = hw.allreduce_time(grad_bytes, n)
        else:
            t_sync = 0.0
        t_total = t_compute + t_sync

        # ---- memory ------------------------------------------------------
        p_bytes = shard_params * 2
        opt_bytes = shard_params * 12 / (st.dp * st.cp if st.zero else 1)
        grad_buf = shard_params * 4
        # 1F1B keeps up to pp in-flight micro-batch activations on stage 0
        inflight = min(st.pp, num_mb)
        act = inflight * (plain * self.layer_act_bytes(tokens_mb, st.tp, False)
                          + rec * self.layer_act_bytes(tokens_mb, st.tp, True))
        head_act = 4 * tokens_mb * m.vocab / st.tp  # logits fp32-ish
        mem = p_bytes + opt_bytes + grad_buf + act + head_act

        tokens_per_step = global_batch * s
        return {
            "strategy": st,
            "time": t_total,
            "mem": mem,
            "fits": mem <= hw.hbm_capacity,
            "tokens_per_sec": tokens_per_step / t_total,
            "num_micro_batches": num_mb,
        }
