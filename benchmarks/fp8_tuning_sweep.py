#!/usr/bin/env python3
"""fp8 serving kernel tuning sweep: XCD swizzle A/B for gate-up + down,
split-K sweep for the fp8 down projection (partials traffic vs fill)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from beta9_amd.ops import gemm as g

ext = g._load()


def t_us(fn, iters=60):
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    M = 256
    hidden, inter = 4096, 14336
    a = (torch.randn(M, hidden, device="cuda") * 0.3).to(torch.bfloat16)
    wg = (torch.randn(inter, hidden, device="cuda") * 0.3).to(torch.bfloat16)
    act = (torch.randn(M, inter, device="cuda") * 0.3).to(torch.bfloat16)
    wd = (torch.randn(hidden, inter, device="cuda") * 0.3).to(torch.bfloat16)
    r = torch.randn(M, hidden, device="cuda").to(torch.bfloat16)
    aq, a_s = g.quantize_fp8_tensor(a)
    gq, g_s = g.quantize_fp8_rowwise(wg)
    uq, u_s = g.quantize_fp8_rowwise(wg)
    actq, act_s = g.quantize_fp8_tensor(act)
    dq, d_s = g.quantize_fp8_rowwise(wd)

    gu_flops = 2 * 2 * M * hidden * inter
    for sw in (False, True):
        us = t_us(lambda: ext.gemm_gate_up_silu_fp8(aq, gq, uq, g_s, u_s, a_s, 0, sw))
        print(f"gate-up fp8 xcd={int(sw)}: {us:7.1f} us  {gu_flops/us/1e6:6.0f} TF-eq")

    d_flops = 2 * M * inter * hidden
    for sk in (2, 4, 8, 16):
        for sw in (False, True):
            us = t_us(lambda: ext.gemm_down_fp8(actq, dq, d_s, act_s, r, sk, sw))
            print(f"down fp8 splitk={sk:2d} xcd={int(sw)}: {us:7.1f} us  {d_flops/us/1e6:6.0f} TF-eq")
    # numerics spot check on the best-config candidates
    out = ext.gemm_down_fp8(actq, dq, d_s, act_s, r, 4, True).float()
    adq = actq.view(torch.float8_e4m3fn).float() * act_s
    wdq = dq.view(torch.float8_e4m3fn).float() * d_s[:, None]
    ref = r.float() + adq @ wdq.t()
    print("down maxrel:", ((out - ref).abs().max() / ref.abs().max()).item())


if __name__ == "__main__":
    main()
