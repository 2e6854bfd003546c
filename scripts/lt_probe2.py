#!/usr/bin/env python3
"""DGELU_BGRAD algo availability across shapes + bias dtypes."""
import ctypes, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hetu_amd.ops.functional as F
dev = torch.device("cuda", 0)
e = F.ext()
for (M, Ff, H) in [(512, 1024, 256), (8192, 4096, 1024),
                   (32768, 16384, 4096)]:
    dy = torch.randn(M, H, dtype=torch.bfloat16, device=dev)
    wp = torch.randn(H, Ff, dtype=torch.bfloat16, device=dev)
    aux = torch.randn(M, Ff, dtype=torch.bfloat16, device=dev)
    try:
        dh, db = tuple(e.lt_dgelu_bgrad(dy, wp, aux))
        torch.cuda.synchronize()
        print(f"dg M{M} F{Ff} H{H}: OK")
    except RuntimeError as ex:
        print(f"dg M{M} F{Ff} H{H}: {str(ex)[:60]}")
    try:
        a, aux2 = tuple(e.lt_linear_gelu_aux(
            torch.randn(M, H, dtype=torch.bfloat16, device=dev),
            torch.randn(Ff, H, dtype=torch.bfloat16, device=dev),
            torch.randn(Ff, dtype=torch.bfloat16, device=dev)))
        print(f"fg M{M} N{Ff} K{H}: OK")
    except RuntimeError as ex:
        print(f"fg M{M} N{Ff} K{H}: {str(ex)[:60]}")
