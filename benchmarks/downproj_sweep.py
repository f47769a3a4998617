#!/usr/bin/env python3
"""Split-K sweep for the serving down-projection (M=256,N=4096,K=14336) —
hipBLASLt holds 39.9% of the serving step at ~477 TF effective."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from beta9_amd.ops import _gemm_ext as ext

def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

for (M, N, K) in [(256, 4096, 14336), (512, 4096, 14336)]:
    a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
    fl = 2 * M * N * K
    ref = a.float() @ w.float().t()
    dt = bench(lambda: a @ w.t())
    print(f"{M}x{N}x{K} hipBLASLt : {dt*1e6:7.1f} us  {fl/dt/1e12:6.1f} TF")
    for sk in (1, 2, 4, 8, 16):
        if (K // 64) % sk:
            continue
        out = ext.gemm_bf16_nt(a, w, False, sk)
        err = ((out.float() - ref).abs().max() / ref.abs().max()).item()
        dt = bench(lambda: ext.gemm_bf16_nt(a, w, False, sk))
        print(f"{M}x{N}x{K} native sk{sk:2d}: {dt*1e6:7.1f} us  {fl/dt/1e12:6.1f} TF  relerr {err:.4f}")
    for sk in (1, 2, 4, 8, 16):
        if (K // 64) % sk:
            continue
        dt = bench(lambda: ext.gemm_bf16_nt(a, w, True, sk))
        print(f"{M}x{N}x{K} nat+xcd sk{sk:2d}: {dt*1e6:7.1f} us  {fl/dt/1e12:6.1f} TF")
