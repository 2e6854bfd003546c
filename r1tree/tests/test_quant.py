"""Blockwise quantization: roundtrip error bounds (CPU reference path;
the GPU kernel parity test lives in test_gpu_kernels)."""
import torch

import hetu_amd.ops.functional as F


def test_int8_roundtrip():
    x = torch.randn(1024)
    q, am = F.quantize_blockwise(x, "int8", 64)
    y = F.dequantize_blockwise(q, am, "int8", 64, 1024)
    assert (y - x).abs().max() < x.abs().max() / 100


def test_nf4_roundtrip():
    x = torch.randn(4096)
    q, am = F.quantize_blockwise(x, "nf4", 64)
    assert q.numel() == 2048
    y = F.dequantize_blockwise(q, am, "nf4", 64, 4096)
    # nf4: coarse but bounded relative error per block
    assert (y - x).abs().max() < 0.2 * x.abs().max()
    assert torch.corrcoef(torch.stack([x, y]))[0, 1] > 0.98


def test_fp4_roundtrip():
    x = torch.randn(512)
    q, am = F.quantize_blockwise(x, "fp4", 64)
    y = F.dequantize_blockwise(q, am, "fp4", 64, 512)
    assert torch.corrcoef(torch.stack([x, y]))[0, 1] > 0.9


def test_matmul_4bit():
    x = torch.randn(8, 64)
    w = torch.randn(32, 64)
    q, am = F.quantize_blockwise(w, "nf4", 64)
    y = F.matmul_4bit(x, q, am, "nf4", 64, (32, 64))
    ref = x @ w.t()
    rel = (y - ref).norm() / ref.norm()
    assert rel < 0.1, rel


def test_matmul_4bit_and_qlinear():
    """4-bit matmul path (reference Quantization.h matmul4bit): y from the
    packed weight must match y from the dequantized weight exactly, and
    stay close to the fp32 original."""
    import torch

    from hetu_amd.ops import functional as F
    from hetu_amd.peft.lora import QLinear
    torch.manual_seed(0)
    w = torch.randn(32, 64)
    x = torch.randn(4, 64)
    q, amax = F.quantize_blockwise(w.reshape(-1), "nf4", 64)
    y = F.matmul_4bit(x, q, amax, "nf4", 64, (32, 64))
    wd = F.dequantize_blockwise(q, amax, "nf4", 64, 32 * 64,
                                torch.float32).reshape(32, 64)
    assert torch.allclose(y, x @ wd.t(), atol=1e-5)
    rel = (y - x @ w.t()).abs().max() / w.abs().max()
    assert rel < 0.5           # nf4 quantization noise, not garbage
    ql = QLinear(w, bias=torch.randn(32), qtype="nf4")
    y2 = ql(x)
    assert y2.shape == (4, 32)
    assert torch.allclose(y2, x @ wd.t() + ql.bias, atol=1e-5)
