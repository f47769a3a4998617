#!/usr/bin/env python3
"""Race screen + A/B for the phase-split GEMM variants (p8, p2) (guide §5.4 two-lane
discipline: sync-structure edits get multi-run race screens at
256/512/4096 plus within-probe A/B vs the unmodified kernel)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from beta9_amd.ops import _gemm_ext as ext


def relerr(out, ref):
    return ((out.float() - ref).abs().max() / (ref.abs().max() + 1e-6)).item()


def main():
    torch.manual_seed(int(sys.argv[1]) if len(sys.argv) > 1 else 0)
    ok = True
    for run in range(5):
        for n in (256, 512, 1024, 4096):
            a = (torch.randn(n, n, device="cuda") * 0.5).to(torch.bfloat16)
            b = (torch.randn(n, n, device="cuda") * 0.5).to(torch.bfloat16)
            ref = (a.float() @ b.float().t())
            for vname, vfn in (("p8", ext.gemm_bf16_nt_p8), ("p2", ext.gemm_bf16_nt_p2)):
                e = relerr(vfn(a, b, False), ref)
                if e > 0.02:
                    print(f"RACE/NUMERICS FAIL {vname} run={run} n={n} relerr={e}")
                    ok = False
    print("race screen:", "PASS" if ok else "FAIL")
    if not ok:
        return 1

    # within-probe A/B vs the 2-barrier 256^2 kernel
    for (M, N, K) in [(4096, 4096, 4096), (8192, 8192, 8192), (2048, 14336, 4096)]:
        a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
        b = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
        for name, fn in [("base", lambda: ext.gemm_bf16_nt(a, b, False, 1)),
                         ("p2  ", lambda: ext.gemm_bf16_nt_p2(a, b, False)),
                         ("p8  ", lambda: ext.gemm_bf16_nt_p8(a, b, False)),
                         ("base", lambda: ext.gemm_bf16_nt(a, b, False, 1)),
                         ("p2  ", lambda: ext.gemm_bf16_nt_p2(a, b, False)),
                         ("p8  ", lambda: ext.gemm_bf16_nt_p8(a, b, False))]:
            for _ in range(3):
                fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(10):
                fn()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 10
            print(f"{M}x{N}x{K} {name}: {2*M*N*K/dt/1e12:7.1f} TF")
    return 0


if __name__ == "__main__":
    sys.exit(main())
