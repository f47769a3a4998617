#!/usr/bin/env python3
"""p8 GEMM prefetch-lead A/B with race screen (guide §5.4 two-lane
discipline: LEAD>5 stages into regions whose last reader is 1-2 phases
away, covered only by HBM landing latency — so every LEAD value must be
multi-seed race-screened before a perf claim)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from beta9_amd.ops.gemm import _load

ext = _load()


def t_ms(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def race_screen(lead: int) -> bool:
    for n in (256, 512, 4096):
        for seed in (0, 3, 7, 11, 13):
            torch.manual_seed(seed)
            a = (torch.randn(n, n, device="cuda") * 0.5).to(torch.bfloat16)
            b = (torch.randn(n, n, device="cuda") * 0.5).to(torch.bfloat16)
            ref = (a.float() @ b.float().t()).to(torch.bfloat16).float()
            for _rep in range(3):
                out = ext.gemm_bf16_nt_p8(a, b, True, lead).float()
                err = (out - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
                if err > 0.02:
                    print(f"RACE/NUMERICS FAIL lead={lead} n={n} seed={seed} err={err}")
                    return False
    return True


def main():
    for n in (4096, 8192):
        a = (torch.randn(n, n, device="cuda") * 0.5).to(torch.bfloat16)
        b = (torch.randn(n, n, device="cuda") * 0.5).to(torch.bfloat16)
        flops = 2 * n * n * n
        line = [f"N={n}:"]
        for lead in (5, 6, 7):
            ms = t_ms(lambda: ext.gemm_bf16_nt_p8(a, b, True, lead))
            line.append(f"lead{lead}={flops/ms/1e9:.0f}TF")
        lib = t_ms(lambda: a @ b.t())
        line.append(f"hipBLASLt={flops/lib/1e9:.0f}TF")
        print(" ".join(line))
    for lead in (6, 7):
        ok = race_screen(lead)
        print(f"race screen lead={lead}: {'PASS' if ok else 'FAIL'}")


if __name__ == "__main__":
    main()
