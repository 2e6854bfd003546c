#!/usr/bin/env python3
"""Which hipBLASLt epilogues have solutions on this build/arch?"""
import ctypes, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import hetu_amd.ops.functional as F
dev = torch.device("cuda", 0)
x = torch.randn(512, 256, dtype=torch.bfloat16, device=dev)
w = torch.randn(1024, 256, dtype=torch.bfloat16, device=dev)
b = torch.randn(1024, dtype=torch.bfloat16, device=dev)
try:
    a, aux = tuple(F.ext().lt_linear_gelu_aux(x, w, b))
    print("GELU_AUX_BIAS: OK")
except RuntimeError as e:
    print(f"GELU_AUX_BIAS: {e}")
wp = torch.randn(256, 1024, dtype=torch.bfloat16, device=dev)
dy = torch.randn(512, 256, dtype=torch.bfloat16, device=dev)
aux = torch.randn(512, 1024, dtype=torch.bfloat16, device=dev)
try:
    dh, db = tuple(F.ext().lt_dgelu_bgrad(dy, wp, aux))
    print("DGELU_BGRAD: OK")
except RuntimeError as e:
    print(f"DGELU_BGRAD: {e}")
