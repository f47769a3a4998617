import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from beta9_amd.ops import _gemm_ext as ext

def relerr(out, ref):
    return ((out.float() - ref).abs().max() / (ref.abs().max() + 1e-6)).item()

# numerics: multi-seed on shapes hitting the 256-row kernel
ok = True
for seed in (0, 7, 13):
    torch.manual_seed(seed)
    for (M, N, K) in [(2048, 1024, 256), (4096, 512, 128), (2048, 14336, 512)]:
        a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
        g = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
        u = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
        out = ext.gemm_gate_up_silu(a, g, u, False)
        ref = torch.nn.functional.silu(a.float() @ g.float().t()) * (a.float() @ u.float().t())
        e = relerr(out, ref)
        if e > 0.02:
            print(f"FAIL seed={seed} {M}x{N}x{K} relerr={e}")
            ok = False
print("numerics:", "PASS" if ok else "FAIL")

def bench(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

for (M, N, K) in [(2048, 14336, 4096), (4096, 14336, 4096), (256, 14336, 4096)]:
    a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    g = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
    u = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
    fl = 2 * 2 * M * N * K
    dt = bench(lambda: ext.gemm_gate_up_silu(a, g, u, False))
    print(f"{M}x{N}x{K} fused    : {dt*1e6:7.1f} us  {fl/dt/1e12:6.1f} TF")
    dt = bench(lambda: torch.nn.functional.silu(a @ g.t()) * (a @ u.t()))
    print(f"{M}x{N}x{K} torch    : {dt*1e6:7.1f} us  {fl/dt/1e12:6.1f} TF")
