"""GPU numerics tests: every hand-written HIP/CDNA4 kernel vs a plain
PyTorch fp32 reference (same op, CPU/eager path of hetu_amd.ops.functional).

Run on an MI355X: python -m pytest tests -m gpu -x -q
"""
import math

import pytest
import torch

import hetu_amd.ops.functional as F

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda", 0)


def _close(a, b, rtol=2e-2, atol=2e-2, what=""):
    a = a.detach().float().cpu()
    b = b.detach().float().cpu()
    err = (a - b).abs().max().item()
    denom = b.abs().max().item() + 1e-6
    assert torch.allclose(a, b, rtol=rtol, atol=atol), \
        f"{what}: max abs err {err} (ref max {denom})"


@pytest.fixture(autouse=True)
def _require_ext():
    assert F.has_ext(), "HIP extension must be built and loaded on GPU"


class TestNorms:
    @pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
    @pytest.mark.parametrize("shape", [(128, 4096), (3, 17, 1024)])
    def test_rmsnorm(self, dtype, shape):
        x = torch.randn(*shape, dtype=dtype, device=dev())
        w = torch.randn(shape[-1], dtype=dtype, device=dev())
        y, rstd = F.rmsnorm_fwd(x, w, 1e-6)
        yr, rr = F.rmsnorm_fwd(x.cpu().float(), w.cpu().float(), 1e-6)
        _close(y, yr, what="rmsnorm fwd")
        _close(rstd, rr, what="rmsnorm rstd")
        dy = torch.randn_like(x)
        dx, dw = F.rmsnorm_bwd(dy, x, w, rstd)
        dxr, dwr = F.rmsnorm_bwd(dy.cpu().float(), x.cpu().float(),
                                 w.cpu().float(), rr)
        _close(dx, dxr, what="rmsnorm dx")
        _close(dw, dwr, rtol=5e-2, atol=5e-2, what="rmsnorm dw")

    @pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float32])
    def test_layernorm(self, dtype):
        x = torch.randn(256, 2048, dtype=dtype, device=dev())
        w = torch.randn(2048, dtype=dtype, device=dev())
        b = torch.randn(2048, dtype=dtype, device=dev())
        y, mean, rstd = F.layernorm_fwd(x, w, b, 1e-5)
        yr, mr, rr = F.layernorm_fwd(x.cpu().float(), w.cpu().float(),
                                     b.cpu().float(), 1e-5)
        _close(y, yr, what="ln fwd")
        dy = torch.randn_like(x)
        dx, dw, db = F.layernorm_bwd(dy, x, w, mean, rstd)
        dxr, dwr, dbr = F.layernorm_bwd(dy.cpu().float(), x.cpu().float(),
                                        w.cpu().float(), mr, rr)
        _close(dx, dxr, what="ln dx")
        _close(dw, dwr, rtol=5e-2, atol=5e-2, what="ln dw")
        _close(db, dbr, rtol=5e-2, atol=5e-2, what="ln db")


class TestElementwise:
    def test_swiglu(self):
        x = torch.randn(64, 512, dtype=torch.bfloat16, device=dev())
        y = F.swiglu_fwd(x)
        yr = F.swiglu_fwd(x.cpu().float())
        _close(y, yr, what="swiglu fwd")
        dy = torch.randn_like(y)
        dx = F.swiglu_bwd(dy, x)
        dxr = F.swiglu_bwd(dy.cpu().float(), x.cpu().float())
        _close(dx, dxr, what="swiglu bwd")

    def test_rope(self):
        B, S, H, D = 2, 64, 4, 128
        x = torch.randn(B, S, H, D, dtype=torch.bfloat16, device=dev())
        pos = torch.arange(S, device=dev()).float()
        inv = 1.0 / (10000 ** (torch.arange(D // 2, device=dev()).float()
                               / (D // 2)))
        ang = pos[:, None] * inv[None, :]
        cos, sin = ang.cos(), ang.sin()
        y = F.rope_fwd(x, cos, sin)
        yr = F.rope_fwd(x.cpu().float(), cos.cpu(), sin.cpu())
        _close(y, yr, what="rope fwd")
        g = F.rope_bwd(y, cos, sin)
        gr = F.rope_bwd(yr, cos.cpu(), sin.cpu())
        _close(g, gr, what="rope bwd")

    def test_softmax(self):
        x = torch.randn(128, 1024, dtype=torch.bfloat16, device=dev())
        y = F.softmax_fwd(x)
        yr = F.softmax_fwd(x.cpu().float())
        _close(y, yr, rtol=1e-2, atol=1e-3, what="softmax fwd")
        dy = torch.randn_like(x)
        dx = F.softmax_bwd(dy, y)
        dxr = F.softmax_bwd(dy.cpu().float(), yr)
        _close(dx, dxr, rtol=2e-2, atol=1e-2, what="softmax bwd")

    def test_dropout_stats_and_bwd(self):
        x = torch.ones(1 << 20, dtype=torch.bfloat16, device=dev())
        y, mask = F.dropout_fwd(x, 0.3, 1234, 7)
        keep = mask.float().mean().item()
        assert abs(keep - 0.7) < 0.01, f"keep rate {keep}"
        # kept elements scaled by 1/(1-p)
        ys = y[mask.bool()].float()
        assert torch.allclose(ys, torch.full_like(ys, 1 / 0.7), rtol=1e-2)
        dy = torch.randn_like(x)
        dx = F.dropout_bwd(dy, mask, 0.3, 1234, 7)
        ref = dy.float() * mask.float() / 0.7
        _close(dx, ref, what="dropout bwd")


class TestEmbeddingAdam:
    def test_embedding(self):
        V, D = 1000, 512
        tbl = torch.randn(V, D, dtype=torch.bfloat16, device=dev())
        ids = torch.randint(0, V, (32, 128), device=dev())
        y = F.embedding_fwd(tbl, ids)
        _close(y, tbl.cpu()[ids.cpu()], what="embedding fwd")
        dy = torch.randn_like(y)
        g = F.embedding_bwd(dy, ids, V)
        gr = F.embedding_bwd(dy.cpu().float(), ids.cpu(), V)
        _close(g, gr, rtol=5e-2, atol=5e-2, what="embedding bwd")

    def test_adam(self):
        n = 10000
        p = torch.randn(n, device=dev())
        g = torch.randn(n, dtype=torch.bfloat16, device=dev())
        m = torch.zeros(n, device=dev())
        v = torch.zeros(n, device=dev())
        out16 = torch.zeros(n, dtype=torch.bfloat16, device=dev())
        pr, mr, vr = p.cpu().clone(), m.cpu().clone(), v.cpu().clone()
        F.adam_step(p, g, m, v, 1e-3, 0.9, 0.999, 1e-8, 0.01, 1, out16)
        F.adam_step(pr, g.cpu(), mr, vr, 1e-3, 0.9, 0.999, 1e-8, 0.01, 1,
                    None)
        _close(p, pr, rtol=1e-4, atol=1e-5, what="adam p")
        _close(m, mr, rtol=1e-4, atol=1e-5, what="adam m")
        _close(v, vr, rtol=1e-4, atol=1e-5, what="adam v")
        _close(out16, pr.to(torch.bfloat16), rtol=1e-2, atol=1e-2,
               what="adam out16")


class TestCE:
    def test_softmax_ce(self):
        N, V = 512, 32000
        logits = torch.randn(N, V, dtype=torch.bfloat16, device=dev())
        labels = torch.randint(0, V, (N,), device=dev())
        labels[::7] = -100
        loss, lse = F.softmax_ce_fwd(logits, labels, -100)
        lr, lser = F.softmax_ce_fwd(logits.cpu().float(), labels.cpu(), -100)
        _close(loss, lr, what="ce loss")
        _close(lse, lser, what="ce lse")
        dl = torch.randn(N, device=dev())
        g = F.softmax_ce_bwd(dl, logits, labels, lse, -100)
        gr = F.softmax_ce_bwd(dl.cpu(), logits.cpu().float(), labels.cpu(),
                              lser, -100)
        _close(g, gr, rtol=2e-2, atol=1e-3, what="ce grad")

    def test_vp_ce_local(self):
        N, V = 128, 8192
        logits = torch.randn(N, V, dtype=torch.bfloat16, device=dev())
        labels = torch.randint(0, 4 * V, (N,), device=dev())
        lmax, picked = F.vocab_parallel_ce_local_stats(
            logits, labels, V, 2 * V, -100)
        lr, pr = F.vocab_parallel_ce_local_stats(
            logits.cpu().float(), labels.cpu(), V, 2 * V, -100)
        _close(lmax, lr, what="vp lmax")
        _close(picked, pr, what="vp picked")


class TestGemm:
    @pytest.mark.parametrize("mnk", [(256, 256, 256), (512, 1024, 4096),
                                     (2048, 4096, 4096)])
    def test_gemm_tn(self, mnk):
        M, N, K = mnk
        # asymmetric operands (transpose-detecting, guide G9)
        a = torch.randn(M, K, dtype=torch.bfloat16, device=dev())
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev()) \
            + torch.linspace(-1, 1, K, device=dev()).to(torch.bfloat16)
        y = F.ext().gemm_bf16(a, w, True)
        yr = torch.matmul(a.float(), w.float().t())
        _close(y, yr, rtol=3e-2, atol=3e-1, what=f"gemm {mnk}")


class TestAttention:
    @pytest.mark.parametrize("cfg", [
        dict(B=2, H=4, Hkv=4, S=256, D=128, causal=True),
        dict(B=2, H=4, Hkv=4, S=256, D=128, causal=False),
        dict(B=1, H=8, Hkv=2, S=512, D=128, causal=True),   # GQA
        dict(B=2, H=4, Hkv=4, S=192, D=64, causal=True),    # ragged S
        dict(B=1, H=2, Hkv=2, S=300, D=128, causal=True),   # ragged S, v2
        dict(B=1, H=2, Hkv=2, S=1024, D=128, causal=True),  # multi q-block
        dict(B=1, H=2, Hkv=2, S=1024, D=128, causal=False),
    ])
    def test_fwd_bwd(self, cfg):
        B, H, Hkv, S, D = cfg["B"], cfg["H"], cfg["Hkv"], cfg["S"], cfg["D"]
        causal = cfg["causal"]
        torch.manual_seed(0)
        q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev())
        k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev())
        v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device=dev())
        scale = 1.0 / math.sqrt(D)
        o, lse = F.flash_attn_fwd(q, k, v, causal, scale)
        orf, lser = F._attn_ref_fwd(q.cpu().float(), k.cpu().float(),
                                    v.cpu().float(), causal, scale)
        _close(o, orf, rtol=3e-2, atol=3e-2, what="fa fwd")
        _close(lse, lser, rtol=1e-2, atol=1e-2, what="fa lse")
        do = torch.randn_like(o)
        dq, dk, dv = F.flash_attn_bwd(do, q, k, v, o, lse, causal, scale)
        dqr, dkr, dvr = F._attn_ref_bwd(do.cpu().float(), q.cpu().float(),
                                        k.cpu().float(), v.cpu().float(),
                                        orf, lser, causal, scale)
        _close(dq, dqr, rtol=5e-2, atol=5e-2, what="fa dq")
        _close(dk, dkr, rtol=5e-2, atol=5e-2, what="fa dk")
        _close(dv, dvr, rtol=5e-2, atol=5e-2, what="fa dv")

    def test_fwd_spike_forces_rescale(self):
        """Rule 26: force the online-softmax rescale branch with a spiked
        key so a wrong rescale order cannot pass on bounded random data."""
        B, H, S, D = 1, 2, 256, 128
        q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev())
        k = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev())
        v = torch.randn(B, H, S, D, dtype=torch.bfloat16, device=dev())
        k[:, :, 200] = (q[:, :, 255] * 4).to(torch.bfloat16)  # late spike
        scale = 1.0 / math.sqrt(D)
        o, lse = F.flash_attn_fwd(q, k, v, True, scale)
        orf, _ = F._attn_ref_fwd(q.cpu().float(), k.cpu().float(),
                                 v.cpu().float(), True, scale)
        _close(o, orf, rtol=3e-2, atol=3e-2, what="fa spike")


class TestQuant:
    @pytest.mark.parametrize("qtype", ["int8", "nf4", "fp4"])
    def test_quant_roundtrip_matches_cpu(self, qtype):
        torch.manual_seed(0)
        x = torch.randn(8192, dtype=torch.float32, device=dev())
        q, am = F.quantize_blockwise(x, qtype, 64)
        qr, amr = F.quantize_blockwise(x.cpu(), qtype, 64)
        _close(am, amr, rtol=1e-5, atol=1e-6, what="absmax")
        # codes may differ at exact midpoints; compare dequantized values
        y = F.dequantize_blockwise(q, am, qtype, 64, 8192)
        yr = F.dequantize_blockwise(qr, amr, qtype, 64, 8192)
        _close(y, yr, rtol=1e-3, atol=2e-2, what=f"{qtype} dequant")


class TestFusedQKVAttention:
    """GPU fused path (in-place RoPE + strided FA on the qkv buffer) vs
    the CPU fp32 composed reference."""

    @pytest.mark.parametrize("cfg", [
        dict(B=2, H=4, Hkv=4, S=512, rope=True, causal=True),
        dict(B=1, H=8, Hkv=2, S=512, rope=True, causal=True),    # GQA
        dict(B=1, H=4, Hkv=4, S=300, rope=False, causal=True),   # ragged
        dict(B=1, H=2, Hkv=2, S=1024, rope=False, causal=False),
    ])
    def test_fused_vs_cpu_ref(self, cfg):
        B, H, Hkv, S = cfg["B"], cfg["H"], cfg["Hkv"], cfg["S"]
        D = 128
        torch.manual_seed(0)
        C = (H + 2 * Hkv) * D
        qkv = torch.randn(B, S, C, dtype=torch.bfloat16, device=dev())
        cos = sin = None
        if cfg["rope"]:
            t = torch.arange(S, dtype=torch.float32)
            inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2).float() / D))
            fr = torch.outer(t, inv)
            cos, sin = fr.cos().to(dev()), fr.sin().to(dev())
        qkv_cpu = qkv.cpu().float()
        o, lse = F.fused_qkv_attention_fwd(
            qkv, H, Hkv, D, cos, sin, cfg["causal"], None)
        cc = cos.cpu() if cos is not None else None
        ss = sin.cpu() if sin is not None else None
        orf, lser = F.fused_qkv_attention_fwd(
            qkv_cpu, H, Hkv, D, cc, ss, cfg["causal"], None)
        _close(o, orf, rtol=3e-2, atol=3e-2, what="fused fwd")
        _close(lse, lser, rtol=1e-2, atol=1e-2, what="fused lse")
        do = torch.randn_like(o)
        dqkv = F.fused_qkv_attention_bwd(
            do, qkv, o, lse, H, Hkv, D, cos, sin, cfg["causal"], None)
        dref = F.fused_qkv_attention_bwd(
            do.cpu().float(), qkv_cpu, orf, lser, H, Hkv, D, cc, ss,
            cfg["causal"], None)
        _close(dqkv, dref, rtol=6e-2, atol=6e-2, what="fused dqkv")


class TestVocabParallelCEKernels:
    def test_vp_sumexp_and_bwd_match_ref(self):
        torch.manual_seed(0)
        rows, V = 512, 1024
        logits = torch.randn(rows, V, dtype=torch.bfloat16, device=dev())
        labels = torch.randint(0, 4 * V, (rows,), device=dev())
        labels[::17] = -100
        lf = logits.cpu().float()
        gmax = lf.max(-1).values + 0.3   # pretend global max is higher
        gs = F.ext().vp_sumexp(logits, gmax.to(dev()))
        ref = torch.exp(lf - gmax[:, None]).sum(-1)
        _close(gs, ref, rtol=1e-2, atol=1e-3, what="vp_sumexp")
        # bwd: shard covering vocab [V, 2V)
        lse = torch.log(ref) + gmax
        gy = torch.rand(rows)
        dl = F.ext().vp_ce_bwd(gy.to(dev()), logits, labels,
                               lse.to(dev()), V, 2 * V, -100)
        sm = torch.exp(lf - lse[:, None])
        onehot = torch.zeros_like(sm)
        lb = labels.cpu()
        for r in range(rows):
            if V <= lb[r] < 2 * V:
                onehot[r, lb[r] - V] = 1.0
        scale = torch.where(lb == -100, torch.zeros_like(gy), gy)
        refd = (sm - onehot) * scale[:, None]
        _close(dl, refd, rtol=3e-2, atol=3e-3, what="vp_ce_bwd")

    def test_norm_bwd_v2_matches_v1(self):
        torch.manual_seed(1)
        R, D = 2048, 4096
        x = torch.randn(R, D, dtype=torch.bfloat16, device=dev())
        w = torch.randn(D, dtype=torch.bfloat16, device=dev())
        b = torch.randn(D, dtype=torch.bfloat16, device=dev())
        dy = torch.randn(R, D, dtype=torch.bfloat16, device=dev())
        y, mean, rstd = F.layernorm_fwd(x, w, b, 1e-5)
        a = F.ext().layernorm_bwd(dy, x, w, mean, rstd)
        v2 = F.ext().layernorm_bwd2(dy, x, w, mean, rstd)
        for n, (t1, t2) in zip(("dx", "dw", "db"), zip(a, v2)):
            _close(t2, t1.float(), rtol=2e-2, atol=2e-2,
                   what=f"ln bwd2 {n}")
        yr, rs = F.rmsnorm_fwd(x, w, 1e-6)
        a = F.ext().rmsnorm_bwd(dy, x, w, rs)
        v2 = F.ext().rmsnorm_bwd2(dy, x, w, rs)
        for n, (t1, t2) in zip(("dx", "dw"), zip(a, v2)):
            _close(t2, t1.float(), rtol=2e-2, atol=2e-2,
                   what=f"rms bwd2 {n}")


class TestGenerator:
    def test_fa_prefill_matches_fp32_path(self):
        """Serving prefill through the FA kernel (bf16) vs the fp32 torch
        fallback path."""
        from hetu_amd.engine.generator import LlamaGenerator, LlamaKVCache
        from hetu_amd.models.llama import LlamaConfig
        cfg = LlamaConfig(n_layer=2, n_head=4, n_kv_head=2, hidden=512,
                          ffn_hidden=256, vocab=301, max_seq=64)
        torch.manual_seed(0)
        state = {"wte.weight": torch.randn(301, 512) * 0.02,
                 "lnf.weight": torch.ones(512),
                 "lm_head.weight": torch.randn(301, 512) * 0.02}
        for i in range(2):
            state[f"l{i}.ln1.weight"] = torch.ones(512)
            state[f"l{i}.ln2.weight"] = torch.ones(512)
            state[f"l{i}.attn.wqkv.weight"] = torch.randn(1024, 512) * 0.02
            state[f"l{i}.attn.wo.weight"] = torch.randn(512, 512) * 0.02
            state[f"l{i}.mlp.w_in.weight"] = torch.randn(512, 512) * 0.02
            state[f"l{i}.mlp.w_out.weight"] = torch.randn(512, 256) * 0.02
        ids = torch.randint(0, 301, (2, 32), device=dev())
        g16 = LlamaGenerator(cfg, state, device=dev(),
                             dtype=torch.bfloat16)
        g32 = LlamaGenerator(cfg, state, device=dev(),
                             dtype=torch.float32)
        c16 = LlamaKVCache(cfg, 2, 48, dev(), torch.bfloat16)
        c32 = LlamaKVCache(cfg, 2, 48, dev(), torch.float32)
        l16 = g16._forward(ids, c16, 0)
        l32 = g32._forward(ids, c32, 0)
        _close(l16, l32.float(), rtol=6e-2, atol=6e-2, what="prefill")
        # decode a few tokens; caches must stay consistent
        out16 = g16.generate(ids, max_new_tokens=4, temperature=0.0)
        assert out16.shape == (2, 36)


@pytest.mark.gpu
def test_colsum_parity_and_replay():
    """colsum vs fp32 reference, plus bit-stability across hipGraph
    replays (the at::native reduce this replaces corrupts from the 2nd
    replay on some shapes)."""
    import hetu_amd.ops.functional as F
    dev = torch.device("cuda", 0)
    for R, C in [(512, 1024), (512, 768), (7, 8), (4096, 16384),
                 (3, 1000)]:
        x = torch.randn(R, C, dtype=torch.bfloat16, device=dev)
        ref = x.float().sum(0)
        out = F.colsum(x)
        assert out.dtype == torch.float32
        tol = 3e-2 * R ** 0.5 + 1e-3
        assert (out - ref).abs().max().item() < tol, (R, C)
    x = torch.randn(512, 1024, dtype=torch.bfloat16, device=dev)
    first = F.colsum(x).clone()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        out = F.colsum(x)
    for _ in range(3):
        g.replay()
        torch.cuda.synchronize()
        assert torch.equal(out, first)


@pytest.mark.gpu
def test_lt_epilogue_fused_mlp():
    """hipBLASLt GELU_AUX_BIAS / DGELU_BGRAD numerics vs the composed
    reference on GPU."""
    import hetu_amd.ops.functional as F
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    M, H, Ff = 512, 256, 1024
    x = torch.randn(M, H, dtype=torch.bfloat16, device=dev)
    wfc = torch.randn(Ff, H, dtype=torch.bfloat16, device=dev) * 0.05
    b1 = torch.randn(Ff, dtype=torch.bfloat16, device=dev) * 0.1
    wproj = torch.randn(H, Ff, dtype=torch.bfloat16, device=dev) * 0.05
    a, aux = F.linear_gelu_aux(x, wfc, b1)
    href = torch.nn.functional.linear(x.float(), wfc.float(), b1.float())
    aref = torch.nn.functional.gelu(href, approximate="tanh")
    assert (aux.float() - href).abs().max().item() < 0.15
    assert (a.float() - aref).abs().max().item() < 0.15
    dy = torch.randn(M, H, dtype=torch.bfloat16, device=dev)
    dh, db = F.dgelu_bgrad(dy, wproj, aux)
    if F._LT_BWD[0] is not True:
        pytest.skip("hipBLASLt DGELU_BGRAD unavailable on this build")
    da = torch.matmul(dy.float(), wproj.float())
    dh_ref = F.gelu_bwd(da.bfloat16(), aux).float()
    assert (dh.float() - dh_ref).abs().max().item() < 0.2, \
        (dh.float() - dh_ref).abs().max()
    db_ref = dh_ref.sum(0)
    assert (db.float() - db_ref).abs().max().item() < \
        3e-2 * M ** 0.5 + 0.3
    # replay safety: captured calls are bit-stable
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        a2, aux2 = F.linear_gelu_aux(x, wfc, b1)
        dh2, db2 = F.dgelu_bgrad(dy, wproj, aux2)
    g.replay()
    torch.cuda.synchronize()
    first = (a2.clone(), dh2.clone(), db2.clone())
    for _ in range(3):
        g.replay()
    torch.cuda.synchronize()
    assert torch.equal(a2, first[0])
    assert torch.equal(dh2, first[1])
    assert torch.equal(db2, first[2])


@pytest.mark.gpu
def test_varlen_single_kernel():
    """Single-launch packed-varlen fwd vs the per-segment reference, with
    many ragged segments (VERDICT: 64 segments, one launch)."""
    import hetu_amd.ops.functional as F
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    H, D = 4, 128
    lens = torch.randint(16, 257, (64,))
    cu = torch.zeros(65, dtype=torch.int64)
    cu[1:] = lens.cumsum(0)
    T = int(cu[-1])
    q = torch.randn(T, H, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(T, H, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(T, H, D, dtype=torch.bfloat16, device=dev)
    for causal in (True, False):
        o, lse = F.varlen_attention_fwd(q, k, v, cu.to(dev), causal)
        # reference: per-segment dense flash
        oref = torch.empty_like(o)
        lref = torch.empty(H, T, dtype=torch.float32, device=dev)
        for s0, s1 in zip(cu[:-1].tolist(), cu[1:].tolist()):
            qs = q[s0:s1].permute(1, 0, 2).unsqueeze(0).contiguous()
            ks = k[s0:s1].permute(1, 0, 2).unsqueeze(0).contiguous()
            vs = v[s0:s1].permute(1, 0, 2).unsqueeze(0).contiguous()
            ob, lb = F.flash_attn_fwd(qs, ks, vs, causal, None)
            oref[s0:s1] = ob[0].permute(1, 0, 2)
            lref[:, s0:s1] = lb[0]
        assert (o.float() - oref.float()).abs().max().item() < 2e-2, causal
        assert (lse - lref).abs().max().item() < 1e-3, causal


@pytest.mark.gpu
def test_fused_add_ln_kernels():
    """layernorm_fwd_res / layernorm_bwd2_res numerics vs composed."""
    import hetu_amd.ops.functional as F
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    R, D = 512, 1024
    x = torch.randn(R, D, dtype=torch.bfloat16, device=dev)
    r = torch.randn(R, D, dtype=torch.bfloat16, device=dev)
    w = (torch.rand(D, device=dev) + 0.5).bfloat16()
    b = (torch.randn(D, device=dev) * 0.1).bfloat16()
    y, s, mean, rstd = F.layernorm_fwd_res(x, r, w, b, 1e-5)
    s_ref = (x.float() + r.float()).bfloat16()
    y_ref, m_ref, r_ref = F.layernorm_fwd(s_ref, w, b, 1e-5)
    assert torch.equal(s, s_ref)
    # the fused kernel takes stats over the fp32 sum; the composed ref
    # rounds s to bf16 first — up to ~2 bf16 ulps apart (fused is the
    # more accurate one)
    assert (y.float() - y_ref.float()).abs().max().item() < 6e-2
    assert (mean - m_ref).abs().max().item() < 1e-3
    dy = torch.randn_like(y)
    ds_ext = torch.randn_like(y)
    dsum, dw, db = F.layernorm_bwd_res(dy, s, w, mean, rstd, ds_ext)
    dx_ref, dw_ref, db_ref = F.layernorm_bwd(dy, s, w, mean, rstd)
    assert (dsum.float() - (dx_ref.float() + ds_ext.float())
            ).abs().max().item() < 3e-2
    assert (dw.float() - dw_ref.float()).abs().max().item() < 1.0
    assert (db.float() - db_ref.float()).abs().max().item() < 1.0
    # no-ext variant
    dsum2, _, _ = F.layernorm_bwd_res(dy, s, w, mean, rstd, None)
    assert (dsum2.float() - dx_ref.float()).abs().max().item() < 2e-2


@pytest.mark.gpu
def test_fused_add_rms_kernels():
    import hetu_amd.ops.functional as F
    dev = torch.device("cuda", 0)
    torch.manual_seed(1)
    R, D = 256, 512
    x = torch.randn(R, D, dtype=torch.bfloat16, device=dev)
    r = torch.randn(R, D, dtype=torch.bfloat16, device=dev)
    w = (torch.rand(D, device=dev) + 0.5).bfloat16()
    y, s, rstd = F.rmsnorm_fwd_res(x, r, w, 1e-6)
    s_ref = (x.float() + r.float()).bfloat16()
    y_ref, r_ref = F.rmsnorm_fwd(s_ref, w, 1e-6)
    assert torch.equal(s, s_ref)
    assert (y.float() - y_ref.float()).abs().max().item() < 6e-2
    dy = torch.randn_like(y)
    ds_ext = torch.randn_like(y)
    dsum, dw = F.rmsnorm_bwd_res(dy, s, w, rstd, ds_ext)
    dx_ref, dw_ref = F.rmsnorm_bwd(dy, s, w, rstd)
    assert (dsum.float() - (dx_ref.float() + ds_ext.float())
            ).abs().max().item() < 3e-2
    assert (dw.float() - dw_ref.float()).abs().max().item() < 1.0
