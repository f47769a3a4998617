#!/usr/bin/env python3
"""Profiling target: fp8 gate-up only, serving shape, N iterations."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from beta9_amd.ops import gemm as g
ext = g._load()
M, hidden, inter = 256, 4096, 14336
a = (torch.randn(M, hidden, device="cuda") * 0.3).to(torch.bfloat16)
wg = (torch.randn(inter, hidden, device="cuda") * 0.3).to(torch.bfloat16)
aq, a_s = g.quantize_fp8_tensor(a)
gq, g_s = g.quantize_fp8_rowwise(wg)
uq, u_s = g.quantize_fp8_rowwise(wg)
for _ in range(int(sys.argv[1]) if len(sys.argv) > 1 else 50):
    ext.gemm_gate_up_silu_fp8(aq, gq, uq, g_s, u_s, a_s, 0, False)
torch.cuda.synchronize()
print("done")
