"""Kernel dispatch: hand-written HIP/CDNA4 extension on GPU, torch on CPU.

Every hot op of the reference kernel set
(/root/reference/hetu/impl/kernel/ — RMSNorm.cu, FusedLayerNorm.cu,
SwiGLU.cu, rotary.cu, Softmax.cu, SoftmaxCrossEntropySparse.cu,
VocabParallelCrossEntropyLoss.cu, Dropout.cu, EmbeddingLookup.cu,
Optimizers.cu AdamCuda, FlashAttention.cu, MatMul.cu) has an MI355X-native
equivalent here. On a ROCm GPU the call MUST go through the in-tree
_hetu_hip extension (hand-written gfx950 kernels); if the extension is
missing on a CUDA/HIP device we raise — no silent eager fallback. On CPU the
plain torch implementations below serve as the numerics reference used by
the unit tests.
"""
from __future__ import annotations

import math
import os
from typing import Optional, Tuple

import torch

_EXT = None
_EXT_ERR = None
# norm backward variant: split dx + column-reduce dw/db kernels
_NORM_V2 = os.environ.get("HETU_AMD_NORM_V2", "1") == "1"


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import importlib.util
        here = os.path.dirname(__file__)
        cands = [os.path.join(here, "hip", f) for f in
                 os.listdir(os.path.join(here, "hip"))
                 if f.startswith("_hetu_hip") and f.endswith(".so")] \
            if os.path.isdir(os.path.join(here, "hip")) else []
        if not cands:
            raise ImportError("_hetu_hip extension not built "
                              "(run python setup.py build_ext --inplace)")
        spec = importlib.util.spec_from_file_location("_hetu_hip", cands[0])
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _EXT = mod
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e
        _EXT = None
    return _EXT


def ext():
    e = _load_ext()
    if e is None:
        raise RuntimeError(
            f"hetu_amd HIP extension required on GPU but not available: "
            f"{_EXT_ERR}")
    return e


def has_ext() -> bool:
    return _load_ext() is not None


def _gpu(*ts) -> bool:
    return any(isinstance(t, torch.Tensor) and t.is_cuda for t in ts)


# Debug bisect switch: HETU_AMD_TORCH_FALLBACK="fa,ln,adam,..." routes the
# named kernel families to the plain-torch reference implementations even
# on GPU (they are device-agnostic).  Families: fa, qkvfa, ln, rms, gelu,
# silu, swiglu, adam, ce, vce, embed, softmax, rope, dropout.
_TORCH_FB = frozenset(
    s for s in os.environ.get("HETU_AMD_TORCH_FALLBACK", "").split(",") if s)


def _use_hip(family: str, *ts) -> bool:
    return _gpu(*ts) and family not in _TORCH_FB


# ---------------------------------------------------------------------------
# RMSNorm (fused fwd/bwd; reference RMSNorm.cu — block per row)
# ---------------------------------------------------------------------------

def rmsnorm_fwd(x: torch.Tensor, w: torch.Tensor, eps: float
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    """returns (y, rstd[rows] fp32)"""
    if _use_hip("rms", x):
        return ext().rmsnorm_fwd(x.contiguous(),
                                 w.contiguous().to(x.dtype), eps)
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    y = (xf * rstd) * w.float()
    return y.to(x.dtype), rstd.squeeze(-1)


def rmsnorm_bwd(dy: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                rstd: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    if _use_hip("rms", x):
        if _NORM_V2:
            return ext().rmsnorm_bwd2(dy.contiguous(), x.contiguous(),
                                      w.contiguous().to(x.dtype),
                                      rstd.contiguous())
        return ext().rmsnorm_bwd(dy.contiguous(), x.contiguous(),
                                 w.contiguous().to(x.dtype),
                                 rstd.contiguous())
    xf, dyf, wf = x.float(), dy.float(), w.float()
    r = rstd.unsqueeze(-1)
    xhat = xf * r
    wdy = dyf * wf
    c = (wdy * xhat).mean(-1, keepdim=True)
    dx = (wdy - xhat * c) * r
    dw = (dyf * xhat).reshape(-1, x.shape[-1]).sum(0)
    return dx.to(x.dtype), dw.to(w.dtype)


# ---------------------------------------------------------------------------
# LayerNorm (fused, Welford; reference FusedLayerNorm.cu:455-760)
# ---------------------------------------------------------------------------

def layernorm_fwd(x, w, b, eps):
    if _use_hip("ln", x):
        return ext().layernorm_fwd(x.contiguous(),
                                   w.contiguous().to(x.dtype),
                                   b.contiguous().to(x.dtype), eps)
    xf = x.float()
    mean = xf.mean(-1, keepdim=True)
    var = xf.var(-1, unbiased=False, keepdim=True)
    rstd = torch.rsqrt(var + eps)
    y = (xf - mean) * rstd * w.float() + b.float()
    return y.to(x.dtype), mean.squeeze(-1), rstd.squeeze(-1)


def layernorm_bwd(dy, x, w, mean, rstd):
    if _use_hip("ln", x):
        if _NORM_V2:
            return ext().layernorm_bwd2(dy.contiguous(), x.contiguous(),
                                        w.contiguous().to(x.dtype),
                                        mean.contiguous(),
                                        rstd.contiguous())
        return ext().layernorm_bwd(dy.contiguous(), x.contiguous(),
                                   w.contiguous().to(x.dtype),
                                   mean.contiguous(), rstd.contiguous())
    xf, dyf, wf = x.float(), dy.float(), w.float()
    mu = mean.unsqueeze(-1)
    r = rstd.unsqueeze(-1)
    xhat = (xf - mu) * r
    wdy = dyf * wf
    c1 = wdy.mean(-1, keepdim=True)
    c2 = (wdy * xhat).mean(-1, keepdim=True)
    dx = (wdy - c1 - xhat * c2) * r
    D = x.shape[-1]
    dw = (dyf * xhat).reshape(-1, D).sum(0)
    db = dyf.reshape(-1, D).sum(0)
    return dx.to(x.dtype), dw.to(w.dtype), db.to(w.dtype)


# ---------------------------------------------------------------------------
# SwiGLU: y = silu(x1) * x2 over last-dim halves (reference SwiGLU.cu:14,30)
# ---------------------------------------------------------------------------

def rmsnorm_fwd_res(x, resid, w, eps):
    """Fused residual-add + RMSNorm: s = x + resid; y = RMS(s).
    Returns (y, s, rstd)."""
    if _use_hip("rms", x):
        return tuple(ext().rmsnorm_fwd_res(
            x.contiguous(), resid.contiguous(),
            w.contiguous().to(x.dtype), eps))
    s = x + resid
    y, rstd = rmsnorm_fwd(s, w, eps)
    return y, s, rstd


def rmsnorm_bwd_res(dy, s, w, rstd, ds_ext=None):
    if _use_hip("rms", s):
        empty = torch.empty(0, dtype=s.dtype, device=s.device)
        return tuple(ext().rmsnorm_bwd2_res(
            dy.contiguous(), s.contiguous(), w.contiguous().to(s.dtype),
            rstd.contiguous(),
            ds_ext.contiguous() if ds_ext is not None else empty))
    dx, dw = rmsnorm_bwd(dy, s, w, rstd)
    if ds_ext is not None:
        dx = dx + ds_ext
    return dx, dw


def layernorm_fwd_res(x, resid, w, b, eps):
    """Fused residual-add + LayerNorm: s = x + resid; y = LN(s).
    Returns (y, s, mean, rstd)."""
    if _use_hip("ln", x):
        return tuple(ext().layernorm_fwd_res(
            x.contiguous(), resid.contiguous(),
            w.contiguous().to(x.dtype), b.contiguous().to(x.dtype), eps))
    s = x + resid
    y, mean, rstd = layernorm_fwd(s, w, b, eps)
    return y, s, mean, rstd


def layernorm_bwd_res(dy, s, w, mean, rstd, ds_ext=None):
    """returns (dsum = dLN/ds (+ ds_ext), dw, db); dsum is the grad of
    both add inputs."""
    if _use_hip("ln", s):
        empty = torch.empty(0, dtype=s.dtype, device=s.device)
        return tuple(ext().layernorm_bwd2_res(
            dy.contiguous(), s.contiguous(), w.contiguous().to(s.dtype),
            mean.contiguous(), rstd.contiguous(),
            ds_ext.contiguous() if ds_ext is not None else empty))
    dx, dw, db = layernorm_bwd(dy, s, w, mean, rstd)
    if ds_ext is not None:
        dx = dx + ds_ext
    return dx, dw, db


def swiglu_fwd(x):
    if _use_hip("swiglu", x):
        return ext().swiglu_fwd(x.contiguous())
    x1, x2 = x.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(x1) * x2).to(x.dtype)


def swiglu_bwd(dy, x):
    if _use_hip("swiglu", x):
        return ext().swiglu_bwd(dy.contiguous(), x.contiguous())
    x1, x2 = x.float().chunk(2, dim=-1)
    dyf = dy.float()
    sig = torch.sigmoid(x1)
    silu = x1 * sig
    dsilu = sig * (1 + x1 * (1 - sig))
    dx1 = dyf * x2 * dsilu
    dx2 = dyf * silu
    return torch.cat([dx1, dx2], dim=-1).to(x.dtype)


def colsum(x2d: torch.Tensor) -> torch.Tensor:
    """Column sum of a 2-D tensor -> fp32 [C].  On GPU: hand HIP kernel
    (reduce.hip) — also the hipGraph-replay-safe replacement for the
    at::native column reduce, which returns garbage from the 2nd replay of
    a captured step on some shapes (ROCm 7.2; see
    profiles/r02_capture_replay_bug.md)."""
    if _use_hip("colsum", x2d):
        return ext().colsum(x2d.contiguous())
    return x2d.float().sum(0)


# ---------------------------------------------------------------------------
# Activations (fused fwd/bwd; reference Gelu.cu / Activation.cu)
# ---------------------------------------------------------------------------

def gelu_fwd(x):
    if _use_hip("gelu", x):
        return ext().gelu_fwd(x.contiguous())
    return torch.nn.functional.gelu(x, approximate="tanh")


def gelu_bwd(dy, x):
    if _use_hip("gelu", x):
        return ext().gelu_bwd(dy.contiguous(), x.contiguous())
    xf = x.float()
    c = 0.7978845608028654  # sqrt(2/pi)
    a = 0.044715
    t = torch.tanh(c * (xf + a * xf ** 3))
    dt = (1 - t * t) * c * (1 + 3 * a * xf * xf)
    return (dy.float() * (0.5 * (1 + t) + 0.5 * xf * dt)).to(x.dtype)


def silu_fwd(x):
    if _use_hip("silu", x):
        return ext().silu_fwd(x.contiguous())
    return torch.nn.functional.silu(x)


def silu_bwd(dy, x):
    if _use_hip("silu", x):
        return ext().silu_bwd(dy.contiguous(), x.contiguous())
    s = torch.sigmoid(x.float())
    return (dy.float() * s * (1 + x.float() * (1 - s))).to(x.dtype)


# ---------------------------------------------------------------------------
# RoPE (reference rotary.cu:97-185; NeoX-style half rotation)
# ---------------------------------------------------------------------------

def rope_fwd(x, cos, sin, interleaved: bool = False):
    """x: [B, S, H, D] (or [S, H, D] packed); cos/sin: [S, D/2] fp32."""
    if _use_hip("rope", x):
        return ext().rope_fwd(x.contiguous(), cos.contiguous(),
                              sin.contiguous())
    return _rope_ref(x, cos, sin, False)


def rope_bwd(dy, cos, sin, interleaved: bool = False):
    if _use_hip("rope", dy):
        return ext().rope_bwd(dy.contiguous(), cos.contiguous(),
                              sin.contiguous())
    return _rope_ref(dy, cos, sin, True)


def _rope_ref(x, cos, sin, backward: bool):
    xf = x.float()
    D = x.shape[-1]
    x1, x2 = xf[..., :D // 2], xf[..., D // 2:]
    shape = [1] * x.ndim
    shape[-3] = cos.shape[0]   # seq dim of [B, S, H, D] or packed [S, H, D]
    shape[-1] = D // 2
    c = cos.reshape(shape)
    s = sin.reshape(shape)
    if backward:
        s = -s
    y1 = x1 * c - x2 * s
    y2 = x2 * c + x1 * s
    return torch.cat([y1, y2], dim=-1).to(x.dtype)


# ---------------------------------------------------------------------------
# Softmax (row; reference Softmax.cu:151,202,315)
# ---------------------------------------------------------------------------

def softmax_fwd(x, dim=-1):
    # the HIP kernel vectorizes rows by 8; narrow rows (e.g. MoE gate
    # logits over a handful of experts) take the torch path
    if _use_hip("softmax", x) and dim in (-1, x.ndim - 1) \
            and x.shape[-1] % 8 == 0:
        return ext().softmax_fwd(x.contiguous())
    return torch.softmax(x.float(), dim=dim).to(x.dtype)


def softmax_bwd(dy, y, dim=-1):
    if _use_hip("softmax", dy) and dim in (-1, dy.ndim - 1) \
            and dy.shape[-1] % 8 == 0:
        return ext().softmax_bwd(dy.contiguous(), y.contiguous())
    dyf, yf = dy.float(), y.float()
    dx = (dyf - (dyf * yf).sum(dim, keepdim=True)) * yf
    return dx.to(dy.dtype)


# ---------------------------------------------------------------------------
# Sparse softmax cross-entropy (reference SoftmaxCrossEntropySparse.cu)
# and TP-sharded vocab-parallel CE (VocabParallelCrossEntropyLoss.cu:15,70)
# ---------------------------------------------------------------------------

def softmax_ce_fwd(logits, labels, ignore_index: int = -100):
    """logits [N, V], labels [N] -> (loss[N] fp32, logsumexp[N] fp32)."""
    if _use_hip("ce", logits):
        return ext().softmax_ce_fwd(logits.contiguous(),
                                    labels.contiguous(), ignore_index)
    lf = logits.float()
    lse = torch.logsumexp(lf, dim=-1)
    mask = labels != ignore_index
    safe = labels.clamp(min=0)
    picked = lf.gather(-1, safe.unsqueeze(-1)).squeeze(-1)
    loss = torch.where(mask, lse - picked, torch.zeros_like(lse))
    return loss, lse


def softmax_ce_bwd(dloss, logits, labels, lse, ignore_index: int = -100):
    if _use_hip("ce", logits):
        return ext().softmax_ce_bwd(dloss.contiguous(), logits.contiguous(),
                                    labels.contiguous(), lse.contiguous(),
                                    ignore_index)
    lf = logits.float()
    p = torch.exp(lf - lse.unsqueeze(-1))
    mask = (labels != ignore_index)
    safe = labels.clamp(min=0)
    onehot = torch.zeros_like(lf).scatter_(-1, safe.unsqueeze(-1), 1.0)
    g = (p - onehot) * (dloss * mask.to(dloss.dtype)).unsqueeze(-1)
    return g.to(logits.dtype)


def vocab_parallel_ce_local_stats(logits, labels, vocab_start, vocab_end,
                                  ignore_index: int = -100):
    """Per-rank stage of the vocab-parallel CE: local max, local sum-exp (at
    given max), and predicted-logit for labels owned by this shard.
    Cross-rank max/sum allreduce happens at op level (see graph/ops/loss.py).
    """
    if _use_hip("vce", logits):
        return ext().vp_ce_local(logits.contiguous(), labels.contiguous(),
                                 vocab_start, vocab_end, ignore_index)
    lf = logits.float()
    lmax = lf.max(-1).values
    in_shard = (labels >= vocab_start) & (labels < vocab_end) & \
               (labels != ignore_index)
    local_idx = (labels - vocab_start).clamp(min=0, max=lf.shape[-1] - 1)
    picked = lf.gather(-1, local_idx.unsqueeze(-1)).squeeze(-1)
    picked = torch.where(in_shard, picked, torch.zeros_like(picked))
    return lmax, picked


# ---------------------------------------------------------------------------
# Dropout (Philox-seeded, stateless; reference Dropout.cu)
# ---------------------------------------------------------------------------

def dropout_fwd(x, p: float, seed: int, offset: int):
    if p <= 0.0:
        return x, None
    if _use_hip("dropout", x):
        return ext().dropout_fwd(x.contiguous(), p, seed, offset)
    g = torch.Generator(device="cpu").manual_seed(seed + offset)
    mask = (torch.rand(x.shape, generator=g, device=x.device) >= p)
    y = x * mask.to(x.dtype) / (1.0 - p)
    return y, mask


def dropout_bwd(dy, mask, p: float, seed: int, offset: int):
    if p <= 0.0:
        return dy
    if _use_hip("dropout", dy):
        return ext().dropout_bwd(dy.contiguous(), mask, p, seed, offset)
    return dy * mask.to(dy.dtype) / (1.0 - p)


# ---------------------------------------------------------------------------
# Embedding (reference EmbeddingLookup.cu)
# ---------------------------------------------------------------------------

def embedding_fwd(table, ids):
    if _use_hip("embed", table):
        return ext().embedding_fwd(table.contiguous(), ids.contiguous())
    return table[ids]


def embedding_bwd(dy, ids, num_rows: int):
    if _use_hip("embed", dy):
        return ext().embedding_bwd(dy.contiguous(), ids.contiguous(),
                                   num_rows)
    D = dy.shape[-1]
    g = torch.zeros(num_rows, D, dtype=torch.float32, device=dy.device)
    g.index_add_(0, ids.reshape(-1), dy.reshape(-1, D).float())
    return g.to(dy.dtype)


# ---------------------------------------------------------------------------
# Fused Adam (reference Optimizers.cu:145 AdamCuda — m/v update + bias corr
# + weight decay in one pass; fp32 master weights)
# ---------------------------------------------------------------------------

def adam_step(param32, grad, m, v, lr, beta1, beta2, eps, weight_decay,
              step, param_out16: Optional[torch.Tensor] = None,
              bc_dev: Optional[torch.Tensor] = None):
    """bc_dev: optional fp32 device tensor [2] = (1-b1^t, 1-b2^t); when
    given, the kernel reads bias corrections from it — this keeps a
    hipGraph-captured train step correct across replays (the host updates
    the pinned source of bc_dev between replays)."""
    if _use_hip("adam", param32):
        ext().adam_step(param32, grad, m, v, lr, beta1, beta2, eps,
                        weight_decay, step,
                        param_out16 if param_out16 is not None
                        else grad.new_empty(0),
                        bc_dev if bc_dev is not None
                        else param32.new_empty(0))
        return
    gf = grad.float()
    if weight_decay != 0.0:
        gf = gf + weight_decay * param32
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    update = (m / bc1) / (torch.sqrt(v / bc2) + eps)
    param32.add_(update, alpha=-lr)
    if param_out16 is not None:
        param_out16.copy_(param32.to(param_out16.dtype))


# ---------------------------------------------------------------------------
# Flash attention (reference FlashAttention.cu wraps flash_attn 2;
# here: hand-written CDNA4 MFMA kernel, see ops/hip/attention.hip)
# ---------------------------------------------------------------------------

def flash_attn_fwd(q, k, v, causal: bool, scale: Optional[float] = None):
    """q,k,v: [B, H, S, D] (kv may have fewer heads - GQA).
    Returns (out, lse[B,H,S] fp32)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _use_hip("fa", q):
        return ext().flash_attn_fwd(q.contiguous(), k.contiguous(),
                                    v.contiguous(), causal, scale)
    return _attn_ref_fwd(q, k, v, causal, scale)


def flash_attn_bwd(dout, q, k, v, out, lse, causal: bool,
                   scale: Optional[float] = None):
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if _use_hip("fa", q):
        e = ext()
        if q.shape[1] == k.shape[1] and q.shape[-1] == 128 \
                and hasattr(e, "flash_attn_bwd_v3") \
                and os.environ.get("HETU_AMD_FA_V3", "1") == "1":
            # v3 bwd: 3-deep walking-pointer staging rings + counted
            # vmcnt (bit-exact vs v2, ~6% faster on the causal bwd at
            # S=2048 — gpurun_out/r2_fa_bench.log); non-GQA BHSD only
            return e.flash_attn_bwd_v3(dout.contiguous(), q.contiguous(),
                                       k.contiguous(), v.contiguous(),
                                       out.contiguous(), lse.contiguous(),
                                       causal, scale)
        return e.flash_attn_bwd(dout.contiguous(), q.contiguous(),
                                k.contiguous(), v.contiguous(),
                                out.contiguous(), lse.contiguous(),
                                causal, scale)
    return _attn_ref_bwd(dout, q, k, v, out, lse, causal, scale)


def _repeat_kv(k, n_head):
    if k.shape[1] == n_head:
        return k
    rep = n_head // k.shape[1]
    return k.repeat_interleave(rep, dim=1)


def _attn_ref_fwd(q, k, v, causal, scale):
    B, H, S, D = q.shape
    kf = _repeat_kv(k, H).float()
    vf = _repeat_kv(v, H).float()
    qf = q.float()
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        Skv = kf.shape[2]
        mask = torch.ones(S, Skv, dtype=torch.bool, device=q.device).tril(
            diagonal=Skv - S)
        scores = scores.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)
    p = torch.exp(scores - lse.unsqueeze(-1))
    out = torch.matmul(p, vf)
    return out.to(q.dtype), lse


def _attn_ref_bwd(dout, q, k, v, out, lse, causal, scale):
    # delta MUST come from the global (dout . out) rowsum, not the local
    # (dp*p).sum: under ring attention this reference runs per KV block
    # with the global lse, where the two differ.
    B, H, S, D = q.shape
    Hkv = k.shape[1]
    kf = _repeat_kv(k, H).float()
    vf = _repeat_kv(v, H).float()
    qf, dof = q.float(), dout.float()
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        Skv = kf.shape[2]
        mask = torch.ones(S, Skv, dtype=torch.bool, device=q.device).tril(
            diagonal=Skv - S)
        scores = scores.masked_fill(~mask, float("-inf"))
    p = torch.exp(scores - lse.unsqueeze(-1).float())
    dv = torch.matmul(p.transpose(-1, -2), dof)
    dp = torch.matmul(dof, vf.transpose(-1, -2))
    delta = (dof * out.float()).sum(-1, keepdim=True)
    ds = p * (dp - delta) * scale
    dq = torch.matmul(ds, kf)
    dk = torch.matmul(ds.transpose(-1, -2), qf)
    if Hkv != H:
        rep = H // Hkv
        dk = dk.reshape(B, Hkv, rep, -1, D).sum(2)
        dv = dv.reshape(B, Hkv, rep, -1, D).sum(2)
    return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)


# ---------------------------------------------------------------------------
# GEMM — hand-written MFMA bf16 kernel for the transformer hot path;
# library (hipBLASLt via torch.matmul) for general shapes.
# ---------------------------------------------------------------------------

# GEMM routing: the hand-written MFMA kernel (ops/hip/gemm.hip, measured
# ~910 TF bf16 on MI355X) vs hipBLASLt via torch.matmul (~1380-1550 TF on
# the transformer shapes). Plain projection GEMMs default to the library
# per the "library for plain GEMMs" rule; HETU_AMD_GEMM=hip forces the
# hand kernel (used by tests/microbenchmarks, and the target of the
# 256^2 8-phase upgrade).
_HETU_GEMM = os.environ.get("HETU_AMD_GEMM", "blas")  # hip | blas


def linear(x, w, bias=None, trans_w: bool = True):
    """x [..., K] @ (w [N, K] if trans_w else w [K, N]) + bias."""
    if _gpu(x) and _HETU_GEMM == "hip" and x.dtype == torch.bfloat16 \
            and trans_w:
        xs = x.reshape(-1, x.shape[-1]).contiguous()
        M, K = xs.shape
        N = w.shape[0]
        if M % 128 == 0 and N % 128 == 0 and K % 64 == 0:
            y = ext().gemm_bf16(xs, w.contiguous(), trans_w)
            if bias is not None:
                y = y + bias
            return y.reshape(*x.shape[:-1], N)
    if bias is not None and trans_w:
        # F.linear hits the hipBLASLt bias-epilogue path (one GEMM, no
        # separate broadcast-add kernel over the [M, N] output)
        return torch.nn.functional.linear(x, w.to(x.dtype),
                                          bias.to(x.dtype))
    wm = w.t() if trans_w else w
    if bias is not None:
        xs = x.reshape(-1, x.shape[-1])
        y = torch.addmm(bias.to(x.dtype), xs, wm.to(x.dtype))
        return y.reshape(*x.shape[:-1], wm.shape[-1])
    return torch.matmul(x, wm.to(x.dtype))


# ---------------------------------------------------------------------------
# Blockwise quantization (reference quantization.cu / Quantization.h;
# bitsandbytes-style fp4/nf4/int8 with per-block absmax)
# ---------------------------------------------------------------------------

_FP4_CODE = torch.tensor([
    0.0, 0.0052083333, 0.6666667, 1.0, 0.3333333, 0.5, 0.1666667, 0.25,
    -0.0, -0.0052083333, -0.6666667, -1.0, -0.3333333, -0.5,
    -0.1666667, -0.25])
_NF4_CODE = torch.tensor([
    -1.0, -0.6961928009986877, -0.5250730514526367, -0.39491748809814453,
    -0.28444138169288635, -0.18477343022823334, -0.09105003625154495, 0.0,
    0.07958029955625534, 0.16093020141124725, 0.24611230194568634,
    0.33791524171829224, 0.44070982933044434, 0.5626170039176941,
    0.7229568362236023, 1.0])


def quantize_blockwise(x: torch.Tensor, qtype: str = "nf4",
                       blocksize: int = 64):
    """returns (packed uint8, absmax fp32 [nblocks])."""
    if _use_hip("quant", x):
        return tuple(ext().quantize_blockwise(x.contiguous(), qtype,
                                              blocksize))
    flat = x.float().reshape(-1)
    n = flat.numel()
    nblk = (n + blocksize - 1) // blocksize
    pad = nblk * blocksize - n
    if pad:
        flat = torch.cat([flat, flat.new_zeros(pad)])
    blocks = flat.reshape(nblk, blocksize)
    absmax = blocks.abs().amax(-1)
    if qtype == "int8":
        inv = torch.where(absmax > 0, 127.0 / absmax,
                          torch.zeros_like(absmax))
        q = torch.round(blocks * inv.unsqueeze(-1)) + 128
        return q.reshape(-1)[:n].to(torch.uint8), absmax
    code = _NF4_CODE if qtype == "nf4" else _FP4_CODE
    inv = torch.where(absmax > 0, 1.0 / absmax, torch.zeros_like(absmax))
    norm = blocks * inv.unsqueeze(-1)
    idx = (norm.unsqueeze(-1) - code).abs().argmin(-1).to(torch.uint8)
    idx = idx.reshape(-1)[:n]
    hi, lo = idx[0::2], idx[1::2]
    return (hi << 4) | lo, absmax


def dequantize_blockwise(q: torch.Tensor, absmax: torch.Tensor,
                         qtype: str, blocksize: int, numel: int,
                         dtype=torch.float32) -> torch.Tensor:
    if _use_hip("quant", q):
        return ext().dequantize_blockwise(q, absmax, qtype, blocksize,
                                          numel, dtype)
    if qtype == "int8":
        s = absmax.repeat_interleave(blocksize)[:numel] / 127.0
        return ((q.float() - 128) * s).to(dtype)
    code = _NF4_CODE if qtype == "nf4" else _FP4_CODE
    hi, lo = (q >> 4).long(), (q & 15).long()
    vals = torch.stack([code[hi], code[lo]], -1).reshape(-1)[:numel]
    s = absmax.repeat_interleave(blocksize)[:numel]
    return (vals * s).to(dtype)


def matmul_4bit(x: torch.Tensor, qweight: torch.Tensor,
                absmax: torch.Tensor, qtype: str, blocksize: int,
                shape) -> torch.Tensor:
    """y = x @ dequant(W)^T (reference matmul4bit: dequant then GEMM —
    weight stays 4-bit in HBM, dequant streams through once)."""
    w = dequantize_blockwise(qweight, absmax, qtype, blocksize,
                             shape[0] * shape[1], x.dtype).reshape(shape)
    return x @ w.t()


def fused_qkv_attention_fwd(qkv, n_head: int, n_kv_head: int, head_dim: int,
                            cos=None, sin=None, causal: bool = True,
                            scale: Optional[float] = None):
    """Fused attention over the qkv GEMM output [B, S, (H+2Hkv)*D]:
    in-place RoPE on the q|k sections (optional), then flash attention
    reading q/k/v as strided views — zero slice/transpose copies.
    Mutates qkv (rotation is linear, so backward never needs the
    pre-rotation values).  Returns (o [B,S,H*D], lse [B,H,S])."""
    if scale is None:
        scale = 1.0 / math.sqrt(head_dim)
    if _use_hip("qkvfa", qkv):
        e = ext()
        if cos is not None:
            e.rope_qk_inplace(qkv, cos.contiguous(), sin.contiguous(),
                              n_head + n_kv_head, head_dim, 1)
        o, lse = e.flash_attn_fwd_qkv(qkv, n_head, n_kv_head, head_dim,
                                      causal, scale)
        return o, lse
    # CPU reference: same math with views
    B, S, _ = qkv.shape
    H, Hkv, D = n_head, n_kv_head, head_dim
    qkv4 = qkv.view(B, S, H + 2 * Hkv, D)
    if cos is not None:
        qkv4[:, :, :H + Hkv] = _rope_ref(qkv4[:, :, :H + Hkv], cos, sin,
                                         False).to(qkv.dtype)
    q = qkv4[:, :, :H].permute(0, 2, 1, 3)
    k = qkv4[:, :, H:H + Hkv].permute(0, 2, 1, 3)
    v = qkv4[:, :, H + Hkv:].permute(0, 2, 1, 3)
    o, lse = _attn_ref_fwd(q, k, v, causal, scale)
    return o.permute(0, 2, 1, 3).reshape(B, S, H * D).contiguous(), lse


def fused_qkv_attention_bwd(dout, qkv, out, lse, n_head: int,
                            n_kv_head: int, head_dim: int, cos=None,
                            sin=None, causal: bool = True,
                            scale: Optional[float] = None):
    """Backward of fused_qkv_attention: returns dqkv [B,S,(H+2Hkv)*D]
    (RoPE backward applied in place on the dq|dk sections)."""
    if scale is None:
        scale = 1.0 / math.sqrt(head_dim)
    if _use_hip("qkvfa", qkv):
        e = ext()
        dqkv = e.flash_attn_bwd_qkv(dout.contiguous(), qkv,
                                    out.contiguous(), lse.contiguous(),
                                    n_head, n_kv_head, head_dim, causal,
                                    scale)
        if cos is not None:
            e.rope_qk_inplace(dqkv, cos.contiguous(), sin.contiguous(),
                              n_head + n_kv_head, head_dim, -1)
        return dqkv
    B, S, _ = qkv.shape
    H, Hkv, D = n_head, n_kv_head, head_dim
    qkv4 = qkv.view(B, S, H + 2 * Hkv, D)   # already rotated
    q = qkv4[:, :, :H].permute(0, 2, 1, 3)
    k = qkv4[:, :, H:H + Hkv].permute(0, 2, 1, 3)
    v = qkv4[:, :, H + Hkv:].permute(0, 2, 1, 3)
    do4 = dout.view(B, S, H, D).permute(0, 2, 1, 3)
    o4 = out.view(B, S, H, D).permute(0, 2, 1, 3)
    dq, dk, dv = _attn_ref_bwd(do4, q, k, v, o4, lse, causal, scale)
    dqkv = torch.cat([dq, dk, dv], dim=1).permute(0, 2, 1, 3) \
        .reshape(B, S, (H + 2 * Hkv) * D).contiguous()
    if cos is not None:
        d4 = dqkv.view(B, S, H + 2 * Hkv, D)
        d4[:, :, :H + Hkv] = _rope_ref(d4[:, :, :H + Hkv], cos, sin,
                                       True).to(dqkv.dtype)
    return dqkv


def varlen_attention_fwd(q, k, v, cu_seqlens, causal: bool = True,
                         scale: Optional[float] = None):
    """Packed-varlen attention (reference ParallelAttention.cc packed
    path / FlashAttention.cu mha_varlen_fwd): q/k/v [T, H, D] with
    cu_seqlens [n+1] delimiting the packed sequences; each segment attends
    only within itself.  On GPU (D=128, no GQA mismatch constraint): ONE
    kernel launch with device-side cu offsets — no host loop, no
    `.tolist()` sync; blocks beyond a segment's tiles exit early.
    Returns (out [T, H, D], lse [H, T])."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    T, H, D = q.shape
    if _use_hip("varlen", q) and D == 128:
        cu32 = cu_seqlens.to(device=q.device, dtype=torch.int32)
        o, lse = ext().flash_attn_varlen_fwd(
            q.contiguous(), k.contiguous(), v.contiguous(), cu32, T,
            causal, scale)
        return o, lse
    out = torch.empty_like(q)
    lse = torch.empty(H, T, dtype=torch.float32, device=q.device)
    cu = cu_seqlens.tolist()
    for s0, s1 in zip(cu[:-1], cu[1:]):
        if s1 <= s0:
            continue
        qs = q[s0:s1].permute(1, 0, 2).unsqueeze(0)
        ks = k[s0:s1].permute(1, 0, 2).unsqueeze(0)
        vs = v[s0:s1].permute(1, 0, 2).unsqueeze(0)
        o, l = flash_attn_fwd(qs.contiguous(), ks.contiguous(),
                              vs.contiguous(), causal, scale)
        out[s0:s1] = o[0].permute(1, 0, 2)
        lse[:, s0:s1] = l[0]
    return out, lse


def varlen_attention_bwd(dout, q, k, v, out, lse, cu_seqlens,
                         causal: bool = True,
                         scale: Optional[float] = None):
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    dq = torch.empty_like(q)
    dk = torch.empty_like(k)
    dv = torch.empty_like(v)
    cu = cu_seqlens.tolist()
    for s0, s1 in zip(cu[:-1], cu[1:]):
        if s1 <= s0:
            continue
        args = [t[s0:s1].permute(1, 0, 2).unsqueeze(0).contiguous()
                for t in (dout, q, k, v, out)]
        ls = lse[:, s0:s1].unsqueeze(0).contiguous()
        dqs, dks, dvs = flash_attn_bwd(args[0], args[1], args[2], args[3],
                                       args[4], ls, causal, scale)
        dq[s0:s1] = dqs[0].permute(1, 0, 2)
        dk[s0:s1] = dks[0].permute(1, 0, 2)
        dv[s0:s1] = dvs[0].permute(1, 0, 2)
    return dq, dk, dv


# ---------------------------------------------------------------------------
# hipBLASLt epilogue-fused MLP pieces (ltgemm.cpp): gelu folded into the
# fc GEMM (GELU_AUX_BIAS) and dgelu+bias-grad into the backward GEMM
# (DGELU_BGRAD).  CPU path = the exact reference composition.
# ---------------------------------------------------------------------------

# per-direction capability: this hipBLASLt build (gfx950) has DGELU_BGRAD
# solutions but NO GELU_AUX_BIAS solutions, so the forward composes while
# the backward still fuses dgelu + bias-grad into the dgrad GEMM
_LT_FWD = [None]
_LT_BWD = [None]


def linear_gelu_aux(x2d, w, b):
    """returns (gelu(x@w^T+b), pre-gelu aux)."""
    if _use_hip("ltgemm", x2d) and _LT_FWD[0] is not False:
        try:
            out = tuple(ext().lt_linear_gelu_aux(x2d.contiguous(),
                                                 w.contiguous(),
                                                 b.contiguous()))
            _LT_FWD[0] = True
            return out
        except RuntimeError as e:
            if "no algo" not in str(e):
                raise
            # no GELU_AUX solutions: compose (bias still fused via
            # F.linear; gelu = hand HIP kernel; aux = pre-gelu kept)
            _LT_FWD[0] = False
            print("[hetu_amd] hipBLASLt GELU_AUX epilogue unavailable, "
                  "composing the MLP forward")
    h = torch.nn.functional.linear(x2d, w.to(x2d.dtype), b.to(x2d.dtype))
    return gelu_fwd(h), h


def dgelu_bgrad(dy2d, w, aux):
    """returns (dgelu(dy@w, aux), colsum(dgelu(...)))."""
    if _use_hip("ltgemm", dy2d) and _LT_BWD[0] is not False:
        try:
            out = tuple(ext().lt_dgelu_bgrad(dy2d.contiguous(),
                                             w.contiguous(),
                                             aux.contiguous()))
            _LT_BWD[0] = True
            return out
        except RuntimeError as e:
            if "no algo" not in str(e):
                raise
            _LT_BWD[0] = False
            print("[hetu_amd] hipBLASLt DGELU_BGRAD epilogue unavailable, "
                  "composing the MLP backward")
    da = torch.matmul(dy2d, w.to(dy2d.dtype))
    dh = gelu_bwd(da, aux)
    return dh, colsum(dh).to(dh.dtype)
