#!/usr/bin/env python3
"""FP8 down-projection A/B at the serving shape: bf16 split-K (+fused
residual) vs fp8 split-K (+fused residual/dequant) vs hipBLASLt+add."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from beta9_amd.ops import gemm as g


def t_ms(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    M, K, N = 256, 14336, 4096  # serving down-proj shape
    a = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
    r = torch.randn(M, N, device="cuda").to(torch.bfloat16)
    aq, a_s = g.quantize_fp8_tensor(a)
    wq, w_s = g.quantize_fp8_rowwise(w)
    flops = 2 * M * K * N

    bf16 = t_ms(lambda: g.gemm_nt_residual(a, w, r))
    fp8 = t_ms(lambda: g.mlp_down_fp8(aq, wq, w_s, a_s, r))
    lib = t_ms(lambda: r + a @ w.t())
    print(f"shape {M}x{K}x{N} (+residual)")
    print(f"bf16 split-K fused : {bf16*1e3:8.1f} us  {flops/bf16/1e9:7.0f} TF")
    print(f"fp8  split-K fused : {fp8*1e3:8.1f} us  {flops/fp8/1e9:7.0f} TF-equiv  ({bf16/fp8:.2f}x vs bf16)")
    print(f"hipBLASLt + add    : {lib*1e3:8.1f} us  {flops/lib/1e9:7.0f} TF")
    # numerics sanity
    out = g.mlp_down_fp8(aq, wq, w_s, a_s, r).float()
    adq = aq.view(torch.float8_e4m3fn).float() * a_s
    wdq = wq.view(torch.float8_e4m3fn).float() * w_s[:, None]
    ref = r.float() + adq @ wdq.t()
    err = ((out - ref).abs().max() / ref.abs().max()).item()
    print(f"max-rel vs dequant ref: {err:.4f}")


if __name__ == "__main__":
    main()
