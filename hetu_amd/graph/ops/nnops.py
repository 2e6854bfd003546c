"""NN ops: activations, norms, dropout, embedding, RoPE, attention, losses.

Reference parity: Relu/Gelu/Silu/Tanh/Sigmoid (hetu/graph/ops/<Name>.cc),
SwiGLU, Softmax.cc, LayerNorm.cc + FusedLayerNorm, RMSNorm.cc, Dropout.cc,
EmbeddingLookup.cc, Rotary.cc, Attention.cc (flash-attn),
SoftmaxCrossEntropySparse.cc, VocabParallelCrossEntropyLoss.cc, MSELoss.cc.
Compute dispatches through hetu_amd.ops.functional (HIP on GPU / torch on
CPU reference).
"""
from __future__ import annotations

from typing import List

import torch

from ...parallel.dstates import DistributedStates
from ..op import OpInterface
from ..tensor import TensorMeta
from .basics import (_g, _make, MulOp, MulScalarOp, _ScalarOp)


# ---------------------------------------------------------------------------
# Simple activations (torch elementwise on both backends: ROCm torch
# elementwise kernels are HBM-bound and already near speed-of-light; fusion
# happens at the module level via swiglu/fused norms)
# ---------------------------------------------------------------------------

class ReluOp(_ScalarOp):
    type = "Relu"

    def compute(self, op, inputs, ctx):
        return [torch.relu(inputs[0])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, ReluGradOp(), [g[0], op.outputs[0]]).output()]


class ReluGradOp(_ScalarOp):
    type = "ReluGrad"

    def compute(self, op, inputs, ctx):
        g, y = inputs
        return [g * (y > 0).to(g.dtype)]


class GeluOp(_ScalarOp):
    type = "Gelu"

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.gelu_fwd(inputs[0])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, GeluGradOp(), [g[0], op.inputs[0]]).output()]


class GeluGradOp(_ScalarOp):
    type = "GeluGrad"

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.gelu_bwd(inputs[0], inputs[1])]


class SiluOp(_ScalarOp):
    type = "Silu"

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.silu_fwd(inputs[0])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, SiluGradOp(), [g[0], op.inputs[0]]).output()]


class SiluGradOp(_ScalarOp):
    type = "SiluGrad"

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.silu_bwd(inputs[0], inputs[1])]


class TanhOp(_ScalarOp):
    type = "Tanh"

    def compute(self, op, inputs, ctx):
        return [torch.tanh(inputs[0])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        y2 = _make(gr, MulOp(), [op.outputs[0], op.outputs[0]]).output()
        one_m = _make(gr, MulScalarOp(), [y2], {"value": -1.0}).output()
        from .basics import AddScalarOp
        one_m = _make(gr, AddScalarOp(), [one_m], {"value": 1.0}).output()
        return [_make(gr, MulOp(), [g[0], one_m]).output()]


class SigmoidOp(_ScalarOp):
    type = "Sigmoid"

    def compute(self, op, inputs, ctx):
        return [torch.sigmoid(inputs[0])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, SigmoidGradOp(), [g[0], op.outputs[0]]).output()]


class SigmoidGradOp(_ScalarOp):
    type = "SigmoidGrad"

    def compute(self, op, inputs, ctx):
        g, y = inputs
        return [(g.float() * y.float() * (1 - y.float())).to(g.dtype)]


# ---------------------------------------------------------------------------
# SwiGLU (fused)
# ---------------------------------------------------------------------------

class SwiGLUOp(OpInterface):
    type = "SwiGLU"

    def infer_meta(self, attrs, inputs):
        shape = list(inputs[0].shape)
        d = shape[-1]
        shape[-1] = d // 2 if not hasattr(d, "value") else d
        return [TensorMeta(shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.swiglu_fwd(inputs[0])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, SwiGLUGradOp(), [g[0], op.inputs[0]]).output()]


class SwiGLUGradOp(OpInterface):
    type = "SwiGLUGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[1].dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.swiglu_bwd(inputs[0], inputs[1])]


# ---------------------------------------------------------------------------
# Softmax
# ---------------------------------------------------------------------------

class SoftmaxOp(_ScalarOp):
    type = "Softmax"

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.softmax_fwd(inputs[0], op.attrs.get("dim", -1))]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, SoftmaxGradOp(), [g[0], op.outputs[0]],
                      dict(op.attrs)).output()]


class SoftmaxGradOp(_ScalarOp):
    type = "SoftmaxGrad"

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.softmax_bwd(inputs[0], inputs[1],
                              op.attrs.get("dim", -1))]


# ---------------------------------------------------------------------------
# Dropout (Philox stateless — seed/offset attrs; per-run offset bump)
# ---------------------------------------------------------------------------

class DropoutOp(OpInterface):
    type = "Dropout"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype),
                TensorMeta(inputs[0].shape, torch.bool)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        if not ctx.training or op.attrs["p"] <= 0.0:
            return [inputs[0], torch.empty(0, dtype=torch.bool,
                                           device=inputs[0].device)]
        y, mask = F.dropout_fwd(inputs[0], op.attrs["p"],
                                op.attrs["seed"], op.attrs.get("offset", 0))
        return [y, mask if mask is not None else
                torch.empty(0, dtype=torch.bool, device=inputs[0].device)]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, DropoutGradOp(), [g[0], op.outputs[1]],
                      dict(op.attrs)).output()]


class DropoutGradOp(OpInterface):
    type = "DropoutGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        if not ctx.training or op.attrs["p"] <= 0.0 or inputs[1].numel() == 0:
            return [inputs[0]]
        return [F.dropout_bwd(inputs[0], inputs[1], op.attrs["p"],
                              op.attrs["seed"], op.attrs.get("offset", 0))]


# ---------------------------------------------------------------------------
# LayerNorm / RMSNorm (fused kernels)
# ---------------------------------------------------------------------------

def _norm_deduce(op):
    """Norm ops: y follows x; row-stat outputs follow x minus last dim.
    Weight/bias inputs are duplicated and do not affect the layout."""
    x = op.inputs[0]
    if x.ds is not None:
        if x.ds.get_dim(x.ndim - 1) > 1:
            raise ValueError(
                f"{op.name}: input split on the normalized dim; use a "
                f"sequence-parallel layer (comm first)")
        if x.ds.partial > 1:
            raise ValueError(f"{op.name}: partial input needs comm first")
        op.outputs[0].ds = x.ds
        for stat in op.outputs[1:]:
            if stat.meta.ndim == x.ndim - 1:
                stat.ds = x.ds
    for out in op.outputs:
        out.device_group = x.device_group


class LayerNormOp(OpInterface):
    type = "LayerNorm"

    def infer_meta(self, attrs, inputs):
        rows = list(inputs[0].shape[:-1])
        return [TensorMeta(inputs[0].shape, inputs[0].dtype),
                TensorMeta(rows, torch.float32),
                TensorMeta(rows, torch.float32)]

    def deduce_states(self, op):
        _norm_deduce(op)

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        y, mean, rstd = F.layernorm_fwd(inputs[0], inputs[1], inputs[2],
                                        op.attrs["eps"])
        return [y, mean, rstd]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        grad_op = _make(gr, LayerNormGradOp(),
                        [g[0], op.inputs[0], op.inputs[1],
                         op.outputs[1], op.outputs[2]])
        return [grad_op.output(0), grad_op.output(1), grad_op.output(2)]


class LayerNormGradOp(OpInterface):
    type = "LayerNormGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[1].dtype),
                TensorMeta(inputs[2].shape, inputs[2].dtype),
                TensorMeta(inputs[2].shape, inputs[2].dtype)]

    def deduce_states(self, op):
        op.outputs[0].ds = op.inputs[1].ds
        # dw/db over token-split input are partial
        x = op.inputs[1]
        w = op.inputs[2]
        if x.ds is not None:
            n = x.ds.device_num
            npart = x.ds.partial
            for d in x.ds.split_dims():
                if d != x.ndim - 1:
                    npart *= x.ds.get_dim(d)
            states = {-2: npart} if npart > 1 else {}
            if n // max(npart, 1) > 1:
                states[-1] = n // max(npart, 1)
            ds = DistributedStates(n, states)
            op.outputs[1].ds = ds
            op.outputs[2].ds = ds
        for o in op.outputs:
            o.device_group = x.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        dy, x, w, mean, rstd = inputs
        dx, dw, db = F.layernorm_bwd(dy, x, w, mean, rstd)
        return [dx, dw, db]


class RMSNormOp(OpInterface):
    type = "RMSNorm"

    def infer_meta(self, attrs, inputs):
        rows = list(inputs[0].shape[:-1])
        return [TensorMeta(inputs[0].shape, inputs[0].dtype),
                TensorMeta(rows, torch.float32)]

    def deduce_states(self, op):
        _norm_deduce(op)

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        y, rstd = F.rmsnorm_fwd(inputs[0], inputs[1], op.attrs["eps"])
        return [y, rstd]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        grad_op = _make(gr, RMSNormGradOp(),
                        [g[0], op.inputs[0], op.inputs[1], op.outputs[1]])
        return [grad_op.output(0), grad_op.output(1)]


class RMSNormGradOp(OpInterface):
    type = "RMSNormGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[1].dtype),
                TensorMeta(inputs[2].shape, inputs[2].dtype)]

    def deduce_states(self, op):
        op.outputs[0].ds = op.inputs[1].ds
        x = op.inputs[1]
        if x.ds is not None:
            n = x.ds.device_num
            npart = x.ds.partial
            for d in x.ds.split_dims():
                if d != x.ndim - 1:
                    npart *= x.ds.get_dim(d)
            states = {-2: npart} if npart > 1 else {}
            if n // max(npart, 1) > 1:
                states[-1] = n // max(npart, 1)
            op.outputs[1].ds = DistributedStates(n, states)
        for o in op.outputs:
            o.device_group = x.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        dy, x, w, rstd = inputs
        dx, dw = F.rmsnorm_bwd(dy, x, w, rstd)
        return [dx, dw]


# ---------------------------------------------------------------------------
# Embedding
# ---------------------------------------------------------------------------

class EmbeddingOp(OpInterface):
    """inputs: table [V, D], ids [...] -> [..., D]."""
    type = "Embedding"

    def infer_meta(self, attrs, inputs):
        table, ids = inputs
        return [TensorMeta(list(ids.shape) + [table.shape[1]], table.dtype)]

    def deduce_states(self, op):
        table, ids = op.inputs
        out = op.outputs[0]
        if table.ds is not None and table.ds.get_dim(0) > 1:
            raise ValueError("use VocabParallelEmbedding for vocab-split "
                             "tables")
        if ids.ds is not None:
            n = ids.ds.device_num
            states = dict(ids.ds.states)
            out.ds = DistributedStates(n, states, list(ids.ds.order))
        out.device_group = ids.device_group or table.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.embedding_fwd(inputs[0], inputs[1])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        dtable = _make(gr, EmbeddingGradOp(), [g[0], op.inputs[1],
                                               op.inputs[0]]).output()
        return [dtable, None]


class EmbeddingGradOp(OpInterface):
    type = "EmbeddingGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[2].shape, inputs[0].dtype)]

    def deduce_states(self, op):
        gy, ids, table = op.inputs
        out = op.outputs[0]
        if gy.ds is not None:
            n = gy.ds.device_num
            npart = gy.ds.partial
            for d in gy.ds.split_dims():
                if d != gy.ndim - 1:
                    npart *= gy.ds.get_dim(d)
            states = {-2: npart} if npart > 1 else {}
            rest = n // max(npart, 1)
            if rest > 1:
                states[-1] = rest
            out.ds = DistributedStates(n, states)
        out.device_group = table.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        gy, ids, table = inputs
        return [F.embedding_bwd(gy, ids, table.shape[0])]


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------

class RotaryOp(OpInterface):
    """inputs: x [B, S, H, D], cos [S, D/2], sin [S, D/2]."""
    type = "Rotary"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.rope_fwd(inputs[0], inputs[1], inputs[2])]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        return [_make(gr, RotaryGradOp(),
                      [g[0], op.inputs[1], op.inputs[2]]).output(),
                None, None]


class RotaryGradOp(OpInterface):
    type = "RotaryGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[0].shape, inputs[0].dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        return [F.rope_bwd(inputs[0], inputs[1], inputs[2])]


# ---------------------------------------------------------------------------
# Flash attention
# ---------------------------------------------------------------------------

class AttentionOp(OpInterface):
    """inputs: q, k, v each [B, H, S, D] (kv heads may differ for GQA)."""
    type = "Attention"

    def infer_meta(self, attrs, inputs):
        q = inputs[0]
        rows = list(q.shape[:-1])
        return [TensorMeta(q.shape, q.dtype),
                TensorMeta(rows, torch.float32)]   # lse

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        out, lse = F.flash_attn_fwd(inputs[0], inputs[1], inputs[2],
                                    op.attrs["causal"],
                                    op.attrs.get("scale"))
        return [out, lse]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        bwd = _make(gr, AttentionGradOp(),
                    [g[0], op.inputs[0], op.inputs[1], op.inputs[2],
                     op.outputs[0], op.outputs[1]], dict(op.attrs))
        return [bwd.output(0), bwd.output(1), bwd.output(2)]


class AttentionGradOp(OpInterface):
    type = "AttentionGrad"

    def infer_meta(self, attrs, inputs):
        _, q, k, v = inputs[:4]
        return [TensorMeta(q.shape, q.dtype),
                TensorMeta(k.shape, k.dtype),
                TensorMeta(v.shape, v.dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        dout, q, k, v, out, lse = inputs
        dq, dk, dv = F.flash_attn_bwd(dout, q, k, v, out, lse,
                                      op.attrs["causal"],
                                      op.attrs.get("scale"))
        return [dq, dk, dv]


# ---------------------------------------------------------------------------
# Losses
# ---------------------------------------------------------------------------

class SoftmaxCrossEntropySparseOp(OpInterface):
    """inputs: logits [N, V], labels [N] int64 -> per-token loss [N] fp32."""
    type = "SoftmaxCrossEntropySparse"

    def infer_meta(self, attrs, inputs):
        rows = list(inputs[0].shape[:-1])
        return [TensorMeta(rows, torch.float32),
                TensorMeta(rows, torch.float32)]   # lse

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        loss, lse = F.softmax_ce_fwd(inputs[0], inputs[1],
                                     op.attrs.get("ignore_index", -100))
        return [loss, lse]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        gin = _make(gr, SoftmaxCrossEntropySparseGradOp(),
                    [g[0], op.inputs[0], op.inputs[1], op.outputs[1]],
                    dict(op.attrs)).output()
        return [gin, None]


class SoftmaxCrossEntropySparseGradOp(OpInterface):
    type = "SoftmaxCrossEntropySparseGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[1].dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        dloss, logits, labels, lse = inputs
        return [F.softmax_ce_bwd(dloss, logits, labels, lse,
                                 op.attrs.get("ignore_index", -100))]


class MSELossOp(OpInterface):
    type = "MSELoss"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta([], torch.float32)]

    def compute(self, op, inputs, ctx):
        return [torch.nn.functional.mse_loss(inputs[0].float(),
                                             inputs[1].float())]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        gin = _make(gr, MSELossGradOp(),
                    [g[0], op.inputs[0], op.inputs[1]]).output()
        return [gin, None]


class MSELossGradOp(OpInterface):
    type = "MSELossGrad"

    def infer_meta(self, attrs, inputs):
        return [TensorMeta(inputs[1].shape, inputs[1].dtype)]

    def compute(self, op, inputs, ctx):
        g, x, y = inputs
        n = x.numel()
        return [(g.float() * 2.0 / n * (x.float() - y.float())).to(x.dtype)]


class FusedQKVAttentionOp(OpInterface):
    """Fused attention straight off the column-parallel qkv GEMM output
    [B, S, (H + 2*Hkv)*D]: optional in-place RoPE on the q|k sections,
    then flash attention with strided q/k/v views — none of the
    slice/reshape/transpose copies of the composed path (reference
    ParallelAttention.cc packs the same way).  NOTE: mutates its qkv
    input in place (rotation is linear; backward never needs the
    pre-rotation values), so qkv must have no other consumer.
    inputs: qkv[, cos, sin]; attrs: n_head, n_kv_head, head_dim, causal.
    outputs: o [B, S, H*D], lse [B, H, S] fp32."""
    type = "FusedQKVAttention"

    def infer_meta(self, attrs, inputs):
        qkv = inputs[0]
        B, S, C = qkv.shape
        H = attrs["n_head"]
        D = attrs["head_dim"]
        return [TensorMeta((B, S, H * D), qkv.dtype),
                TensorMeta((B, H, S), torch.float32)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        a = op.attrs
        cos = inputs[1] if len(inputs) > 1 else None
        sin = inputs[2] if len(inputs) > 2 else None
        o, lse = F.fused_qkv_attention_fwd(
            inputs[0], a["n_head"], a["n_kv_head"], a["head_dim"],
            cos, sin, a.get("causal", True), a.get("scale"))
        return [o, lse]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        ins = [g[0], op.inputs[0], op.outputs[0], op.outputs[1]] \
            + list(op.inputs[1:])
        bwd = _make(gr, FusedQKVAttentionGradOp(), ins, dict(op.attrs),
                    name="fused_qkv_attn_grad")
        return [bwd.output(0)] + [None] * (len(op.inputs) - 1)


class FusedQKVAttentionGradOp(OpInterface):
    type = "FusedQKVAttentionGrad"

    def infer_meta(self, attrs, inputs):
        qkv = inputs[1]
        return [TensorMeta(qkv.shape, qkv.dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        a = op.attrs
        dout, qkv, out, lse = inputs[:4]
        cos = inputs[4] if len(inputs) > 4 else None
        sin = inputs[5] if len(inputs) > 5 else None
        dqkv = F.fused_qkv_attention_bwd(
            dout, qkv, out, lse, a["n_head"], a["n_kv_head"],
            a["head_dim"], cos, sin, a.get("causal", True),
            a.get("scale"))
        return [dqkv]


class VarlenAttentionOp(OpInterface):
    """Packed-varlen flash attention: q/k/v [T, H, D] + cu_seqlens [n+1]
    (reference ParallelAttention.cc packed path)."""
    type = "VarlenAttention"

    def infer_meta(self, attrs, inputs):
        q = inputs[0]
        T, H, D = q.shape
        return [TensorMeta(q.shape, q.dtype),
                TensorMeta((H, T), torch.float32)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        o, lse = F.varlen_attention_fwd(inputs[0], inputs[1], inputs[2],
                                        inputs[3],
                                        op.attrs.get("causal", True),
                                        op.attrs.get("scale"))
        return [o, lse]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        bwd = _make(gr, VarlenAttentionGradOp(),
                    [g[0], op.inputs[0], op.inputs[1], op.inputs[2],
                     op.outputs[0], op.outputs[1], op.inputs[3]],
                    dict(op.attrs))
        return [bwd.output(0), bwd.output(1), bwd.output(2), None]


class VarlenAttentionGradOp(OpInterface):
    type = "VarlenAttentionGrad"

    def infer_meta(self, attrs, inputs):
        _, q, k, v = inputs[:4]
        return [TensorMeta(q.shape, q.dtype), TensorMeta(k.shape, k.dtype),
                TensorMeta(v.shape, v.dtype)]

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        dout, q, k, v, out, lse, cu = inputs
        return list(F.varlen_attention_bwd(dout, q, k, v, out, lse, cu,
                                           op.attrs.get("causal", True),
                                           op.attrs.get("scale")))


class FusedMLPOp(OpInterface):
    """Transformer MLP block with hipBLASLt epilogue fusion:
    y = gelu(x @ wfc^T + b1) @ wproj^T (+ b2 when attrs["with_b2"]).
    gelu rides the fc GEMM (GELU_AUX_BIAS); backward's dgelu and b1-grad
    ride the dgrad GEMM (DGELU_BGRAD) — no standalone gelu/reduce kernels
    (reference keeps Gelu.cu + Reduce.cu around cuBLAS).
    inputs: x [B, S, H], wfc [F, H], b1 [F], wproj [Ho, F][, b2 [Ho]];
    outputs: y [B, S, Ho], a (gelu out, saved), aux (pre-gelu, saved).
    tp>1 layouts should use the composed path (the fused op is the
    single-device / dp hot path)."""
    type = "FusedMLP"

    def infer_meta(self, attrs, inputs):
        x, wfc = inputs[0], inputs[1]
        wproj = inputs[3]
        B, S = x.shape[0], x.shape[1]
        F = wfc.shape[0]
        Ho = wproj.shape[0]
        return [TensorMeta((B, S, Ho), x.dtype),
                TensorMeta((B * S, F), x.dtype),
                TensorMeta((B * S, F), x.dtype)]

    def deduce_states(self, op):
        x = op.inputs[0]
        for out in op.outputs:
            out.ds = x.ds
            out.device_group = x.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        x, wfc, b1, wproj = inputs[:4]
        b2 = inputs[4] if len(inputs) > 4 else None
        B, S, H = x.shape
        x2 = x.reshape(B * S, H)
        a, aux = F.linear_gelu_aux(x2, wfc, b1)
        y = F.linear(a, wproj, b2)
        return [y.reshape(B, S, wproj.shape[0]), a, aux]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        ins = [g[0], op.inputs[0], op.inputs[1], op.inputs[3],
               op.outputs[1], op.outputs[2]]
        bwd = _make(gr, FusedMLPGradOp(),
                    ins, dict(op.attrs), name="fused_mlp_grad")
        grads = [bwd.output(0), bwd.output(1), bwd.output(2),
                 bwd.output(3)]
        if len(op.inputs) > 4:
            grads.append(bwd.output(4))
        return grads


class FusedMLPGradOp(OpInterface):
    """inputs: dy, x, wfc, wproj, a, aux; outputs: dx, dwfc, db1, dwproj
    [, db2]."""
    type = "FusedMLPGrad"

    def infer_meta(self, attrs, inputs):
        dy, x, wfc, wproj = inputs[:4]
        outs = [TensorMeta(x.shape, x.dtype),
                TensorMeta(wfc.shape, wfc.dtype),
                TensorMeta((wfc.shape[0],), wfc.dtype),
                TensorMeta(wproj.shape, wproj.dtype)]
        if attrs.get("with_b2"):
            outs.append(TensorMeta((wproj.shape[0],), wproj.dtype))
        return outs

    def deduce_states(self, op):
        x = op.inputs[1]
        op.outputs[0].ds = x.ds
        if x.ds is not None:
            # weight/bias grads: partial over the token-split (dp) dims,
            # dup over the rest (same rule as LayerNormGradOp)
            n = x.ds.device_num
            npart = x.ds.partial
            for d in x.ds.split_dims():
                npart *= x.ds.get_dim(d)
            states = {}
            if npart > 1:
                states[-2] = npart
            if n // max(npart, 1) > 1:
                states[-1] = n // max(npart, 1)
            wds = DistributedStates(n, states)
            for out in op.outputs[1:]:
                out.ds = wds
        for out in op.outputs:
            out.device_group = x.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        dy, x, wfc, wproj, a, aux = inputs
        B, S, H = x.shape
        Ho = wproj.shape[0]
        dy2 = dy.reshape(B * S, Ho)
        x2 = x.reshape(B * S, H)
        # dwproj = dy^T @ a; da->dgelu(+db1) fused; dwfc = dh^T @ x;
        # dx = dh @ wfc
        dwproj = torch.matmul(dy2.t(), a).to(wproj.dtype)
        dh, db1 = F.dgelu_bgrad(dy2, wproj, aux)
        dwfc = torch.matmul(dh.t(), x2).to(wfc.dtype)
        dx = torch.matmul(dh, wfc).reshape(B, S, H)
        outs = [dx, dwfc, db1, dwproj]
        if op.attrs.get("with_b2"):
            outs.append(F.colsum(dy2).to(dy.dtype))
        return outs


class FusedAddLNOp(OpInterface):
    """Fused residual-add + LayerNorm (transformer pre-norm chain):
    s = x + r; y = LN(s).  One kernel reads x and r once and writes both
    s (the next residual) and y — no standalone elementwise add; the
    backward folds the residual-grad accumulation into the LN-dx kernel
    (reference keeps Add + FusedLayerNorm separate around Arithmetics.cu).
    inputs: x, r, w, b; outputs: y, s, mean, rstd."""
    type = "FusedAddLN"

    def infer_meta(self, attrs, inputs):
        x = inputs[0]
        rows = list(x.shape[:-1])
        return [TensorMeta(x.shape, x.dtype),
                TensorMeta(x.shape, x.dtype),
                TensorMeta(rows, torch.float32),
                TensorMeta(rows, torch.float32)]

    def deduce_states(self, op):
        x = op.inputs[0]
        for out in op.outputs:
            out.ds = x.ds
            out.device_group = x.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        x, r, w, b = inputs
        y, s, mean, rstd = F.layernorm_fwd_res(x, r, w, b,
                                               op.attrs["eps"])
        return [y, s, mean, rstd]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        dy, ds_ext = g[0], g[1]
        ins = [dy, op.outputs[1], op.inputs[2], op.outputs[2],
               op.outputs[3]]
        attrs = {"has_ext": ds_ext is not None}
        if ds_ext is not None:
            ins.append(ds_ext)
        bwd = _make(gr, FusedAddLNGradOp(), ins, attrs,
                    name="fused_addln_grad")
        dsum = bwd.output(0)
        return [dsum, dsum, bwd.output(1), bwd.output(2)]


class FusedAddLNGradOp(OpInterface):
    """inputs: dy, s, w, mean, rstd[, ds_ext];
    outputs: dsum, dw, db."""
    type = "FusedAddLNGrad"

    def infer_meta(self, attrs, inputs):
        s, w = inputs[1], inputs[2]
        return [TensorMeta(s.shape, s.dtype),
                TensorMeta(w.shape, w.dtype),
                TensorMeta(w.shape, w.dtype)]

    def deduce_states(self, op):
        s = op.inputs[1]
        op.outputs[0].ds = s.ds
        if s.ds is not None:
            n = s.ds.device_num
            npart = s.ds.partial
            for d in s.ds.split_dims():
                if d != s.ndim - 1:
                    npart *= s.ds.get_dim(d)
            states = {}
            if npart > 1:
                states[-2] = npart
            if n // max(npart, 1) > 1:
                states[-1] = n // max(npart, 1)
            wds = DistributedStates(n, states)
            op.outputs[1].ds = wds
            op.outputs[2].ds = wds
        for out in op.outputs:
            out.device_group = s.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        dy, s, w, mean, rstd = inputs[:5]
        ds_ext = inputs[5] if op.attrs.get("has_ext") else None
        dsum, dw, db = F.layernorm_bwd_res(dy, s, w, mean, rstd, ds_ext)
        return [dsum, dw.to(w.dtype), db.to(w.dtype)]


class FusedAddRMSOp(OpInterface):
    """Fused residual-add + RMSNorm (Llama pre-norm chain); see
    FusedAddLNOp.  inputs: x, r, w; outputs: y, s, rstd."""
    type = "FusedAddRMS"

    def infer_meta(self, attrs, inputs):
        x = inputs[0]
        rows = list(x.shape[:-1])
        return [TensorMeta(x.shape, x.dtype),
                TensorMeta(x.shape, x.dtype),
                TensorMeta(rows, torch.float32)]

    def deduce_states(self, op):
        x = op.inputs[0]
        for out in op.outputs:
            out.ds = x.ds
            out.device_group = x.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        x, r, w = inputs
        y, s, rstd = F.rmsnorm_fwd_res(x, r, w, op.attrs["eps"])
        return [y, s, rstd]

    def gradient(self, op, g):
        gr = _g(op.outputs[0])
        dy, ds_ext = g[0], g[1]
        ins = [dy, op.outputs[1], op.inputs[2], op.outputs[2]]
        attrs = {"has_ext": ds_ext is not None}
        if ds_ext is not None:
            ins.append(ds_ext)
        bwd = _make(gr, FusedAddRMSGradOp(), ins, attrs,
                    name="fused_addrms_grad")
        dsum = bwd.output(0)
        return [dsum, dsum, bwd.output(1)]


class FusedAddRMSGradOp(OpInterface):
    """inputs: dy, s, w, rstd[, ds_ext]; outputs: dsum, dw."""
    type = "FusedAddRMSGrad"

    def infer_meta(self, attrs, inputs):
        s, w = inputs[1], inputs[2]
        return [TensorMeta(s.shape, s.dtype),
                TensorMeta(w.shape, w.dtype)]

    def deduce_states(self, op):
        s = op.inputs[1]
        op.outputs[0].ds = s.ds
        if s.ds is not None:
            n = s.ds.device_num
            npart = s.ds.partial
            for d in s.ds.split_dims():
                if d != s.ndim - 1:
                    npart *= s.ds.get_dim(d)
            states = {}
            if npart > 1:
                states[-2] = npart
            if n // max(npart, 1) > 1:
                states[-1] = n // max(npart, 1)
            op.outputs[1].ds = DistributedStates(n, states)
        for out in op.outputs:
            out.device_group = s.device_group

    def compute(self, op, inputs, ctx):
        from ...ops import functional as F
        dy, s, w, rstd = inputs[:4]
        ds_ext = inputs[4] if op.attrs.get("has_ext") else None
        dsum, dw = F.rmsnorm_bwd_res(dy, s, w, rstd, ds_ext)
        return [dsum, dw.to(w.dtype)]
