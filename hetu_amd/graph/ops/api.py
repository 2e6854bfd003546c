"""User-facing functional API: hetu_amd.<fn>(...) builds graph ops.

Plays the role of the reference's ops.yml-generated `hetu.*` functions
(/root/reference/python/hetu/_binding/codegen/ops.yml): each function makes
an op in the current graph and returns its output Tensor(s).
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import torch

from ...parallel.dstates import DistributedStates
from ..graph import current_graph
from ..tensor import Tensor
from . import basics as B
from . import nnops as N
from . import comm as C


def _cg():
    return current_graph()


def _wrap_const(x, like: Tensor) -> Tensor:
    if isinstance(x, Tensor):
        return x
    raise TypeError("use *_scalar ops or pass a Tensor")


# ---- sources --------------------------------------------------------------

def placeholder(shape, dtype=torch.float32, name="placeholder",
                ds: Optional[DistributedStates] = None, device_group=None
                ) -> Tensor:
    op = _cg().make_op(B.PlaceholderOp(), [], {"shape": tuple(shape),
                                               "dtype": dtype},
                       name=name, ds_list=[ds] if ds else None,
                       device_group=device_group)
    return op.output()


# alias matching the reference's parallel_placeholder
def parallel_placeholder(shape, dtype=torch.float32, ds=None,
                         device_group=None, name="placeholder") -> Tensor:
    return placeholder(shape, dtype, name, ds, device_group)


def variable(data: torch.Tensor, name="variable", requires_grad=True,
             ds: Optional[DistributedStates] = None, device_group=None
             ) -> Tensor:
    g = _cg()
    op = g.make_op(B.VariableOp(), [], {"shape": tuple(data.shape),
                                        "dtype": data.dtype},
                   name=name, ds_list=[ds] if ds else None,
                   device_group=device_group)
    t = op.output()
    t.set_data(data)
    t.is_parameter = requires_grad
    t.requires_grad = requires_grad
    if requires_grad:
        g.parameters.append(t)
    return t


def constant(shape, value, dtype=torch.float32, name="const") -> Tensor:
    return _cg().make_op(B.ConstantOp(), [], {"shape": tuple(shape),
                                              "value": value,
                                              "dtype": dtype},
                         name=name).output()


# ---- arithmetic ------------------------------------------------------------

def add(a: Tensor, b) -> Tensor:
    if not isinstance(b, Tensor):
        return _cg().make_op(B.AddScalarOp(), [a], {"value": float(b)}).output()
    return _cg().make_op(B.AddOp(), [a, b], {}).output()


def sub(a: Tensor, b) -> Tensor:
    if not isinstance(b, Tensor):
        return _cg().make_op(B.AddScalarOp(), [a], {"value": -float(b)}).output()
    return _cg().make_op(B.SubOp(), [a, b], {}).output()


def mul(a: Tensor, b) -> Tensor:
    if not isinstance(b, Tensor):
        return _cg().make_op(B.MulScalarOp(), [a], {"value": float(b)}).output()
    return _cg().make_op(B.MulOp(), [a, b], {}).output()


def div(a: Tensor, b) -> Tensor:
    if not isinstance(b, Tensor):
        return _cg().make_op(B.MulScalarOp(), [a],
                             {"value": 1.0 / float(b)}).output()
    return _cg().make_op(B.DivOp(), [a, b], {}).output()


def neg(a: Tensor) -> Tensor:
    return _cg().make_op(B.NegOp(), [a], {}).output()


def pow(a: Tensor, p: float) -> Tensor:  # noqa: A001
    return _cg().make_op(B.PowScalarOp(), [a], {"value": float(p)}).output()


def exp(a):
    return _cg().make_op(B.ExpOp(), [a], {}).output()


def log(a):
    return _cg().make_op(B.LogOp(), [a], {}).output()


def sqrt(a):
    return _cg().make_op(B.SqrtOp(), [a], {}).output()


def rsqrt(a):
    return _cg().make_op(B.RsqrtOp(), [a], {}).output()


def add_n(ts: Sequence[Tensor]) -> Tensor:
    return B.make_add_n(_cg(), list(ts))


# ---- shape -----------------------------------------------------------------

def reshape(a: Tensor, shape, ds=None) -> Tensor:
    """`ds` overrides the output layout (reshape cannot always map split
    dims mechanically, e.g. [B,S,3h] -> [B,S,3,H,Dh] moves a tp split from
    dim 2 to dim 3 — the caller knows the intent)."""
    return _cg().make_op(B.ReshapeOp(), [a], {"shape": tuple(shape)},
                         ds_list=[ds] if ds is not None else None).output()


def transpose(a: Tensor, dim0: int, dim1: int) -> Tensor:
    return _cg().make_op(B.TransposeOp(), [a], {"dim0": dim0,
                                                "dim1": dim1}).output()


def slice_(a: Tensor, dim: int, start, length) -> Tensor:
    return _cg().make_op(B.SliceOp(), [a], {"dim": dim, "start": start,
                                            "length": length}).output()


def concat(ts: Sequence[Tensor], dim: int = 0) -> Tensor:
    return _cg().make_op(B.ConcatOp(), list(ts), {"dim": dim}).output()


def contiguous(a: Tensor) -> Tensor:
    return _cg().make_op(B.ContiguousOp(), [a], {}).output()


def cast(a: Tensor, dtype) -> Tensor:
    if a.dtype == dtype:
        return a
    return _cg().make_op(B.CastOp(), [a], {"dtype": dtype}).output()


# ---- reductions ------------------------------------------------------------

def reduce_sum(a: Tensor, dim=None, keepdim=False) -> Tensor:
    return _cg().make_op(B.ReduceOp_(), [a], {"mode": "sum", "dim": dim,
                                              "keepdim": keepdim}).output()


def reduce_mean(a: Tensor, dim=None, keepdim=False) -> Tensor:
    return _cg().make_op(B.ReduceOp_(), [a], {"mode": "mean", "dim": dim,
                                              "keepdim": keepdim}).output()


def reduce_max(a: Tensor, dim=None, keepdim=False) -> Tensor:
    return _cg().make_op(B.ReduceOp_(), [a], {"mode": "max", "dim": dim,
                                              "keepdim": keepdim}).output()


def reduce_min(a: Tensor, dim=None, keepdim=False) -> Tensor:
    return _cg().make_op(B.ReduceOp_(), [a], {"mode": "min", "dim": dim,
                                              "keepdim": keepdim}).output()


def reduce_prod(a: Tensor, dim=None, keepdim=False) -> Tensor:
    return _cg().make_op(B.ReduceOp_(), [a], {"mode": "prod", "dim": dim,
                                              "keepdim": keepdim}).output()


def norm(a: Tensor, p=2, dim=None, keepdim=False) -> Tensor:
    from . import extra as E
    return _cg().make_op(E.NormOp(), [a],
                         {"p": p, "dim": dim,
                          "keepdim": keepdim}).output()


def softmax_cross_entropy(logits: Tensor, labels: Tensor,
                          reduction="mean") -> Tensor:
    """Dense soft-label CE; for integer labels use
    softmax_cross_entropy_sparse."""
    from . import extra as E
    return _cg().make_op(E.SoftmaxCrossEntropyOp(), [logits, labels],
                         {"reduction": reduction}).output()


def broadcast_to(a: Tensor, shape) -> Tensor:
    return _cg().make_op(B.BroadcastToOp(), [a],
                         {"shape": tuple(shape)}).output()


def group(*tensors) -> Tensor:
    """Control-dependency join: fetch the returned scalar to force every
    input to execute (reference group.cc)."""
    from .optim import GroupOp
    return _cg().make_op(GroupOp(), list(tensors), {}).output()


def ones_like(a: Tensor) -> Tensor:
    return _cg().make_op(B.OnesLikeOp(), [a], {}).output()


def zeros_like(a: Tensor) -> Tensor:
    return _cg().make_op(B.ZerosLikeOp(), [a], {}).output()


# ---- GEMM ------------------------------------------------------------------

# ---- autocast (reference graph/autocast/autocast.cc:39-92: dtype
# inference inserts DataTransferOp casts around compute ops; here the
# casts are inserted at op-build time while the context is active) --------
_AUTOCAST_STACK: List[torch.dtype] = []


class recompute:
    """with ht.recompute(): ... — ops built inside form one recompute
    scope (reference context.py:223 hetu.recompute): their internal
    activations are dropped after forward and recomputed in backward
    (Graph.apply_recompute).  Scope indices auto-increment per graph."""

    def __enter__(self):
        g = _cg()
        idx = getattr(g, "_rc_auto_idx", 0)
        g._rc_auto_idx = idx + 1
        self._scope = g.recompute_scope(idx)
        self._scope.__enter__()
        return self

    def __exit__(self, *a):
        self._scope.__exit__(*a)


class autocast:
    """with ht.autocast(torch.bfloat16): matmul/linear/bmm/attention
    inputs are cast to the target dtype (fp32 params keep a cast edge, so
    the grads flow back in fp32 — AMP semantics)."""

    def __init__(self, dtype=torch.bfloat16):
        self.dtype = dtype

    def __enter__(self):
        _AUTOCAST_STACK.append(self.dtype)
        return self

    def __exit__(self, *a):
        _AUTOCAST_STACK.pop()


def _ac(t: Tensor) -> Tensor:
    if _AUTOCAST_STACK and t.dtype in (torch.float32, torch.float16,
                                       torch.bfloat16) \
            and t.dtype != _AUTOCAST_STACK[-1]:
        return cast(t, _AUTOCAST_STACK[-1])
    return t


def matmul(a: Tensor, b: Tensor, trans_a=False, trans_b=False) -> Tensor:
    return _cg().make_op(B.MatMul2DOp(), [_ac(a), _ac(b)],
                         {"trans_a": trans_a, "trans_b": trans_b}).output()


def linear(x: Tensor, w: Tensor, bias: Optional[Tensor] = None) -> Tensor:
    ins = [_ac(x), _ac(w)] + ([_ac(bias)] if bias is not None else [])
    return _cg().make_op(B.LinearOp(), ins, {}).output()


def bmm(a: Tensor, b: Tensor) -> Tensor:
    return _cg().make_op(B.BatchMatMulOp(), [_ac(a), _ac(b)], {}).output()


def baddbmm(inp: Tensor, a: Tensor, b: Tensor, beta: float = 1.0,
            alpha: float = 1.0) -> Tensor:
    """beta*inp + alpha*(a @ b), batched (reference Baddbmm.cu) — composed
    over the bmm/add ops so the grads come for free."""
    y = bmm(a, b)
    if alpha != 1.0:
        y = mul(y, alpha)
    return add(mul(inp, beta) if beta != 1.0 else inp, y)


def matvec(a: Tensor, v: Tensor) -> Tensor:
    """[M,K] @ [K] -> [M] (reference MatVecMul.cu): routed through the
    GEMM path on a [K,1] view."""
    y = matmul(a, reshape(v, (int(v.shape[0]), 1)))
    return reshape(y, (int(a.shape[0]),))


# ---- nn --------------------------------------------------------------------

def relu(a):
    return _cg().make_op(N.ReluOp(), [a], {}).output()


def gelu(a):
    return _cg().make_op(N.GeluOp(), [a], {}).output()


def silu(a):
    return _cg().make_op(N.SiluOp(), [a], {}).output()


def tanh(a):
    return _cg().make_op(N.TanhOp(), [a], {}).output()


def sigmoid(a):
    return _cg().make_op(N.SigmoidOp(), [a], {}).output()


def swiglu(a):
    return _cg().make_op(N.SwiGLUOp(), [a], {}).output()


def softmax(a, dim=-1):
    return _cg().make_op(N.SoftmaxOp(), [a], {"dim": dim}).output()


_DROPOUT_SEED = [12345]
_DROPOUT_OFFSET = [0]


def dropout(a, p: float):
    _DROPOUT_OFFSET[0] += 1 << 20
    return _cg().make_op(N.DropoutOp(), [a],
                         {"p": p, "seed": _DROPOUT_SEED[0],
                          "offset": _DROPOUT_OFFSET[0]}).output()


def layer_norm(x, w, b, eps=1e-5):
    return _cg().make_op(N.LayerNormOp(), [x, w, b], {"eps": eps}).output(0)


def rms_norm(x, w, eps=1e-6):
    return _cg().make_op(N.RMSNormOp(), [x, w], {"eps": eps}).output(0)


def embedding(table, ids):
    return _cg().make_op(N.EmbeddingOp(), [table, ids], {}).output()


def rotary(x, cos, sin):
    return _cg().make_op(N.RotaryOp(), [x, cos, sin], {}).output()


def fused_qkv_attention(qkv, n_head, n_kv_head, head_dim, cos=None,
                        sin=None, causal=True, scale=None):
    """Fused attention over the qkv GEMM output (see
    nnops.FusedQKVAttentionOp) -> o [B, S, n_head*head_dim]."""
    ins = [qkv] + ([cos, sin] if cos is not None else [])
    return _cg().make_op(N.FusedQKVAttentionOp(), ins,
                         {"n_head": n_head, "n_kv_head": n_kv_head,
                          "head_dim": head_dim, "causal": causal,
                          "scale": scale}).output()


def fused_add_ln(x, r, w, b, eps=1e-5):
    """Fused residual-add + LayerNorm (see nnops.FusedAddLNOp):
    returns (ln_out, sum)."""
    op = _cg().make_op(N.FusedAddLNOp(), [x, r, w, b], {"eps": eps})
    return op.output(0), op.output(1)


def fused_add_rms(x, r, w, eps=1e-6):
    """Fused residual-add + RMSNorm: returns (rms_out, sum)."""
    op = _cg().make_op(N.FusedAddRMSOp(), [x, r, w], {"eps": eps})
    return op.output(0), op.output(1)


def fused_mlp(x, wfc, b1, wproj, b2=None):
    """Epilogue-fused transformer MLP (see nnops.FusedMLPOp):
    y = gelu(x @ wfc^T + b1) @ wproj^T (+ b2)."""
    ins = [x, wfc, b1, wproj] + ([b2] if b2 is not None else [])
    return _cg().make_op(N.FusedMLPOp(), ins,
                         {"with_b2": b2 is not None}).output()


def varlen_attention(q, k, v, cu_seqlens, causal=True, scale=None):
    """Packed-varlen attention: q/k/v [T, H, D], cu_seqlens [n+1]."""
    return _cg().make_op(N.VarlenAttentionOp(), [q, k, v, cu_seqlens],
                         {"causal": causal, "scale": scale}).output()


def attention(q, k, v, causal=True, scale=None):
    q, k, v = _ac(q), _ac(k), _ac(v)
    return _cg().make_op(N.AttentionOp(), [q, k, v],
                         {"causal": causal, "scale": scale}).output(0)


def softmax_cross_entropy_sparse(logits, labels, ignore_index=-100):
    return _cg().make_op(N.SoftmaxCrossEntropySparseOp(), [logits, labels],
                         {"ignore_index": ignore_index}).output(0)


def mse_loss(x, y):
    return _cg().make_op(N.MSELossOp(), [x, y], {}).output()


def check_finite(tensors) -> Tensor:
    return _cg().make_op(B.CheckFiniteOp(), list(tensors), {}).output()


def ring_attention(q, k, v, cp_ranks, causal=True, scale=None,
                   split=None, seq_lens=None):
    """split: NORMAL (contiguous chunks) or SYM (zigzag [head|tail]
    halves, causal-load-balanced); default from
    HETU_AMD_ATTN_SPLIT (reference HETU_PARALLEL_ATTN_SPLIT_PATTERN)."""
    import os as _os
    from . import parallel_ops as P
    if split is None:
        split = _os.environ.get("HETU_AMD_ATTN_SPLIT", "NORMAL")
    return _cg().make_op(P.RingAttentionOp(), [q, k, v],
                         {"causal": causal, "scale": scale,
                          "split": split,
                          "seq_lens": (list(seq_lens) if seq_lens
                                       else None),
                          "cp_ranks": list(cp_ranks)}).output(0)


def vocab_parallel_embedding(table, ids, vocab: int):
    from . import parallel_ops as P
    return _cg().make_op(P.VocabParallelEmbeddingOp(), [table, ids],
                         {"vocab": vocab}).output()


def vocab_parallel_cross_entropy(logits, labels, vocab: int,
                                 ignore_index: int = -100):
    from . import parallel_ops as P
    return _cg().make_op(P.VocabParallelCrossEntropyOp(), [logits, labels],
                         {"vocab": vocab,
                          "ignore_index": ignore_index}).output(0)


# ---- extended families (einsum / vision / losses / manipulation) -----------

def einsum(equation: str, *ts) -> Tensor:
    from . import extra as E
    return _cg().make_op(E.EinsumOp(), list(ts),
                         {"equation": equation}).output()


def conv2d(x, w, bias=None, stride=1, padding=0, dilation=1, groups=1):
    from . import extra as E
    ins = [x, w] + ([bias] if bias is not None else [])
    return _cg().make_op(E.Conv2dOp(), ins,
                         {"stride": stride, "padding": padding,
                          "dilation": dilation, "groups": groups}).output()


def max_pool2d(x, kernel, stride=None, padding=0):
    from . import extra as E
    return _cg().make_op(E.MaxPool2dOp(), [x],
                         {"kernel": kernel, "stride": stride,
                          "padding": padding}).output()


def avg_pool2d(x, kernel, stride=None, padding=0):
    from . import extra as E
    return _cg().make_op(E.AvgPool2dOp(), [x],
                         {"kernel": kernel, "stride": stride,
                          "padding": padding}).output()


def batch_norm(x, w, b, eps=1e-5):
    from . import extra as E
    return _cg().make_op(E.BatchNormOp(), [x, w, b], {"eps": eps}).output()


def instance_norm(x, eps=1e-5):
    from . import extra as E
    return _cg().make_op(E.InstanceNormOp(), [x], {"eps": eps}).output()


def interpolate(x, scale=None, size=None, mode="nearest"):
    from . import extra as E
    return _cg().make_op(E.InterpolateOp(), [x],
                         {"scale": scale, "size": size,
                          "mode": mode}).output()


def binary_cross_entropy(x, target, reduction="mean"):
    from . import extra as E
    return _cg().make_op(E.BCEOp(), [x, target],
                         {"reduction": reduction}).output()


def kl_div(x, target, reduction="batchmean"):
    from . import extra as E
    return _cg().make_op(E.KLDivOp(), [x, target],
                         {"reduction": reduction}).output()


def nll_loss(x, target, reduction="mean", ignore_index=-100):
    from . import extra as E
    return _cg().make_op(E.NLLOp(), [x, target],
                         {"reduction": reduction,
                          "ignore_index": ignore_index}).output()


def where(cond, a, b):
    from . import extra as E
    return _cg().make_op(E.WhereOp(), [cond, a, b], {}).output()


def triu(a, diagonal=0):
    from . import extra as E
    return _cg().make_op(E.TriuOp(), [a], {"diagonal": diagonal}).output()


def clamp(a, min=None, max=None):  # noqa: A002
    from . import extra as E
    return _cg().make_op(E.ClampOp(), [a], {"min": min, "max": max}).output()


def gather(a, dim, index):
    from . import extra as E
    return _cg().make_op(E.GatherOp(), [a, index], {"dim": dim}).output()


def index_add(a, dim, index, src):
    from . import extra as E
    return _cg().make_op(E.IndexAddOp(), [a, index, src],
                         {"dim": dim}).output()


def masked_fill(a, mask, value):
    from . import extra as E
    return _cg().make_op(E.MaskedFillOp(), [a, mask],
                         {"value": value}).output()


def pad(a, pad_widths, value=0.0):
    from . import extra as E
    return _cg().make_op(E.PadOp(), [a], {"pad": list(pad_widths),
                                          "value": value}).output()


def repeat(a, repeats):
    from . import extra as E
    return _cg().make_op(E.RepeatOp(), [a],
                         {"repeats": list(repeats)}).output()


def roll(a, shifts, dims=None):
    from . import extra as E
    return _cg().make_op(E.RollOp(), [a], {"shifts": shifts,
                                           "dims": dims}).output()


def onehot(ids, num_classes):
    from . import extra as E
    return _cg().make_op(E.OnehotOp(), [ids],
                         {"num_classes": num_classes}).output()


def arange(end, start=0, step=1, dtype=torch.int64):
    from . import extra as E
    return _cg().make_op(E.ArangeOp(), [], {"start": start, "end": end,
                                            "step": step,
                                            "dtype": dtype}).output()


def eye(n, dtype=torch.float32):
    from . import extra as E
    return _cg().make_op(E.EyeOp(), [], {"n": n, "dtype": dtype}).output()


# ---- comm ------------------------------------------------------------------

def comm(x: Tensor, dst_ds: DistributedStates, name="comm") -> Tensor:
    return C.make_comm(_cg(), x, dst_ds, name=name)


# ---- autodiff --------------------------------------------------------------

def gradients(ys, xs, grad_ys=None):
    g = _cg()
    single = isinstance(ys, Tensor)
    ys_l = [ys] if single else list(ys)
    gy_l = None if grad_ys is None else (
        [grad_ys] if isinstance(grad_ys, Tensor) else list(grad_ys))
    return g.gradients(ys_l, list(xs), gy_l)


# ---- bulk unary families ----------------------------------------------------
def _unary_api(op_name):
    from . import extra as _x
    from .basics import _make as _mk
    cls = getattr(_x, f"{op_name}Op")

    def f(a, **attrs):
        return _mk(_cg(), cls(), [a], attrs, name=op_name.lower()).output()
    f.__name__ = op_name.lower()
    return f


abs_ = _unary_api("Abs")
ceil = _unary_api("Ceil")
floor = _unary_api("Floor")
round_ = _unary_api("Round")
sin = _unary_api("Sin")
cos = _unary_api("Cos")
reciprocal = _unary_api("Reciprocal")
leaky_relu = _unary_api("LeakyRelu")
mish = _unary_api("Mish")
elu = _unary_api("Elu")
hardshrink = _unary_api("Hardshrink")
hardsigmoid = _unary_api("Hardsigmoid")
hardswish = _unary_api("Hardswish")
hardtanh = _unary_api("Hardtanh")
logsigmoid = _unary_api("Logsigmoid")
softplus = _unary_api("Softplus")
softshrink = _unary_api("Softshrink")


def outer(a, b):
    from .basics import _make as _mk
    from .extra import OuterOp
    return _mk(_cg(), OuterOp(), [a, b], name="outer").output()


def dot(a, b):
    from .basics import _make as _mk
    from .extra import DotOp
    return _mk(_cg(), DotOp(), [a, b], name="dot").output()


def diagonal(a, offset=0, dim1=0, dim2=1):
    from .basics import _make as _mk
    from .extra import DiagonalOp
    return _mk(_cg(), DiagonalOp(), [a],
                 {"offset": offset, "dim1": dim1, "dim2": dim2},
                 name="diagonal").output()


def split(a, sections: int, dim: int = 0):
    """Even split into `sections` along dim (reference Split.cc) as a list
    of slices."""
    n = a.shape[dim]
    assert n % sections == 0, "uneven split"
    step = n // sections
    return [slice_(a, dim, i * step, step) for i in range(sections)]


def dropout2d(a, p: float, seed: int = 0, offset: int = 0):
    from .basics import _make as _mk
    from .extra import Dropout2dOp
    return _mk(_cg(), Dropout2dOp(), [a],
               {"p": p, "seed": seed, "offset": offset},
               name="dropout2d").output()


def bool_(a):
    from .basics import _make as _mk
    from .extra import BoolOp
    return _mk(_cg(), BoolOp(), [a], name="bool").output()


def range_mask(a, start, end):
    from .basics import _make as _mk
    from .extra import RangeMaskOp
    return _mk(_cg(), RangeMaskOp(), [a], {"start": start, "end": end},
               name="range_mask").output()


def as_strided(a, size, stride, offset=0):
    from .basics import _make as _mk
    from .extra import AsStridedOp
    return _mk(_cg(), AsStridedOp(), [a],
               {"size": list(size), "stride": list(stride),
                "offset": offset}, name="as_strided").output()


def mat_dot(a, b):
    """out[i, j] = a[i, j] * b[i] (reference MatDot.cc)."""
    from .basics import _make as _mk
    from .extra import MatDotOp
    return _mk(_cg(), MatDotOp(), [a, b], name="mat_dot").output()


def dynamic_concat(ts, dim: int = 0):
    from .basics import _make as _mk
    from .extra import DynamicConcatOp
    return _mk(_cg(), DynamicConcatOp(), list(ts), {"dim": dim},
               name="dynamic_concat").output()
