#!/usr/bin/env python3
"""FP8 (e4m3, w8a8) fused gate+up+SiLU: numerics vs the dequantized fp32
reference + timing vs the bf16 fused kernel. Multi-seed (race screen)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from beta9_amd.ops import _gemm_ext as ext

E4M3_MAX = 448.0


def quant_rowwise(w):  # [N,K] bf16 -> u8 e4m3 + scale[N]
    amax = w.float().abs().amax(dim=1).clamp(min=1e-8)
    scale = amax / E4M3_MAX
    q = (w.float() / scale[:, None]).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale.float()


def quant_tensor(a):  # per-tensor
    amax = a.float().abs().amax().clamp(min=1e-8)
    scale = (amax / E4M3_MAX).item()
    q = (a.float() / scale).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale


def main():
    ok = True
    for seed in (0, 7, 13):
        torch.manual_seed(seed)
        for (M, N, K) in [(128, 64, 128), (256, 512, 256), (256, 14336, 4096)]:
            a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
            wg = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
            wu = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
            aq, a_s = quant_tensor(a)
            gq, g_s = quant_rowwise(wg)
            uq, u_s = quant_rowwise(wu)
            out = ext.gemm_gate_up_silu_fp8(aq, gq, uq, g_s, u_s, a_s)
            # reference on the SAME quantized values (isolates kernel bugs
            # from quantization error)
            adq = aq.view(torch.float8_e4m3fn).float() * a_s
            gdq = gq.view(torch.float8_e4m3fn).float() * g_s[:, None]
            udq = uq.view(torch.float8_e4m3fn).float() * u_s[:, None]
            ref = torch.nn.functional.silu(adq @ gdq.t()) * (adq @ udq.t())
            e = ((out.float() - ref).abs().max() / (ref.abs().max() + 1e-6)).item()
            # end-to-end vs unquantized bf16 math (quantization error incl.)
            full = torch.nn.functional.silu(a.float() @ wg.float().t()) * (a.float() @ wu.float().t())
            e2e = ((out.float() - full).abs().max() / (full.abs().max() + 1e-6)).item()
            if e > 0.02:
                print(f"KERNEL FAIL seed={seed} {M}x{N}x{K} relerr={e:.4f}")
                ok = False
            if seed == 0:
                print(f"{M}x{N}x{K}: kernel-vs-dequant relerr {e:.4f}, e2e-vs-bf16 {e2e:.3f}")
    print("fp8 numerics:", "PASS" if ok else "FAIL")
    if not ok:
        return 1

    def bench(fn, iters=50):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    for (M, N, K) in [(256, 14336, 4096), (512, 14336, 4096), (1024, 14336, 4096)]:
        a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
        wg = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
        wu = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
        aq, a_s = quant_tensor(a)
        gq, g_s = quant_rowwise(wg)
        uq, u_s = quant_rowwise(wu)
        fl = 2 * 2 * M * N * K
        dt8 = bench(lambda: ext.gemm_gate_up_silu_fp8(aq, gq, uq, g_s, u_s, a_s))
        dt16 = bench(lambda: ext.gemm_gate_up_silu(a, wg, wu, False))
        print(f"{M}x{N}x{K}: fp8 {dt8*1e6:6.1f} us ({fl/dt8/1e12:6.1f} TF-equiv)  "
              f"bf16 {dt16*1e6:6.1f} us ({fl/dt16/1e12:6.1f} TF)  speedup {dt16/dt8:.2f}x")
    return 0


if __name__ == "__main__":
    sys.exit(main())
