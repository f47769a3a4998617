#!/usr/bin/env python3
"""fp8 gate-up BNJ tile A/B at the serving shape (M=256, N=14336, K=4096):
BNJ=2 gives 448 WGs (full chip, smaller tiles), BNJ=4 gives 224 WGs
(underfilled, more reuse). Numerics-checked both ways."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from beta9_amd.ops import gemm as g

ext = g._load()


def t_us(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    M, K, N = 256, 4096, 14336
    a = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
    wg = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
    wu = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
    aq, a_s = g.quantize_fp8_tensor(a)
    gq, g_s = g.quantize_fp8_rowwise(wg)
    uq, u_s = g.quantize_fp8_rowwise(wu)
    flops = 2 * 2 * M * K * N
    for bnj in (1, 2, 4):
        us = t_us(lambda: ext.gemm_gate_up_silu_fp8(aq, gq, uq, g_s, u_s, a_s, bnj))
        out = ext.gemm_gate_up_silu_fp8(aq, gq, uq, g_s, u_s, a_s, bnj).float()
        adq = aq.view(torch.float8_e4m3fn).float() * a_s
        gdq = gq.view(torch.float8_e4m3fn).float() * g_s[:, None]
        udq = uq.view(torch.float8_e4m3fn).float() * u_s[:, None]
        ref = torch.nn.functional.silu(adq @ gdq.t()) * (adq @ udq.t())
        err = ((out - ref).abs().max() / ref.abs().max()).item()
        print(f"BNJ={bnj}: {us:7.1f} us  {flops/us/1e6:6.0f} TF-equiv  maxrel={err:.4f}")


if __name__ == "__main__":
    main()
