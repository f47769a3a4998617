#!/usr/bin/env python3
"""Kernel microbenchmarks: MFMA GEMM TF/s, elementwise bandwidth, arena
streaming bandwidth. Run on an MI355X box; summaries land in stdout (commit
under profiles/ alongside rocprofv3 output)."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench_gpu(fn, warmup=5, iters=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    from beta9_amd.ops import gemm as ops
    from beta9_amd.ops import hipsnap

    assert ops.has_native()
    out = {}

    # -- GEMM: random data (guide §5.4 rule 25: quote the random number) ----
    for (M, N, K) in [(1024, 1024, 1024), (4096, 4096, 4096), (8192, 8192, 8192),
                      (256, 14336, 4096), (2048, 14336, 4096)]:
        a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
        w = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
        for swz in (False, True):
            dt = bench_gpu(lambda: ops.gemm_nt(a, w, xcd_swizzle=swz))
            tf = 2 * M * N * K / dt / 1e12
            out[f"gemm_{M}x{N}x{K}_swz{int(swz)}"] = round(tf, 1)
        # torch (hipBLASLt) comparison
        dt = bench_gpu(lambda: a @ w.t())
        out[f"gemm_{M}x{N}x{K}_torch"] = round(2 * M * N * K / dt / 1e12, 1)

    # -- fused gate+up+silu vs unfused (2 gemm + silu_mul) -----------------
    for (M, N, K) in [(256, 14336, 4096), (2048, 14336, 4096)]:
        a = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
        wg = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
        wu = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
        dt = bench_gpu(lambda: ops.mlp_gate_up(a, wg, wu))
        out[f"fused_gateup_{M}x{N}x{K}_TF"] = round(4 * M * N * K / dt / 1e12, 1)
        dt = bench_gpu(lambda: ops.silu_mul(ops.gemm_nt(a, wg), ops.gemm_nt(a, wu)))
        out[f"unfused_gateup_{M}x{N}x{K}_TF"] = round(4 * M * N * K / dt / 1e12, 1)
        dt = bench_gpu(
            lambda: torch.nn.functional.silu(a @ wg.t()) * (a @ wu.t())
        )
        out[f"torch_gateup_{M}x{N}x{K}_TF"] = round(4 * M * N * K / dt / 1e12, 1)

    # -- elementwise bandwidth ---------------------------------------------
    n = 64 << 20
    g = torch.randn(n, device="cuda").to(torch.bfloat16)
    u = torch.randn(n, device="cuda").to(torch.bfloat16)
    dt = bench_gpu(lambda: ops.silu_mul(g, u))
    out["silu_mul_TBps"] = round(3 * n * 2 / dt / 1e12, 2)

    x = torch.randn(4096, 8192, device="cuda").to(torch.bfloat16)
    wn = torch.randn(8192, device="cuda").to(torch.bfloat16)
    dt = bench_gpu(lambda: ops.rmsnorm(x, wn))
    out["rmsnorm_TBps"] = round(2 * x.numel() * 2 / dt / 1e12, 2)

    # -- arena H2D streaming (pinned double-buffered) ----------------------
    arena = hipsnap.Arena(device=0, size_bytes=8 << 30, staging_bytes=512 << 20)
    data = os.urandom(2 << 30)
    off = arena.alloc("blob", len(data))
    t0 = time.perf_counter()
    arena.upload(off, data)
    dt = time.perf_counter() - t0
    out["arena_h2d_GBps"] = round(len(data) / dt / 1e9, 1)
    t0 = time.perf_counter()
    _ = arena.download(off, len(data))
    dt = time.perf_counter() - t0
    out["arena_d2h_GBps"] = round(len(data) / dt / 1e9, 1)

    # file -> HBM (the cold-start weight path) incl. page-cache-warm read
    path = "/tmp/b9_weights.bin"
    with open(path, "wb") as f:
        f.write(data)
    t0 = time.perf_counter()
    arena._impl.upload_file(off, path, 0, len(data))
    dt = time.perf_counter() - t0
    out["arena_file2hbm_GBps"] = round(len(data) / dt / 1e9, 1)
    os.unlink(path)

    # snapshot/restore bandwidth
    t0 = time.perf_counter()
    info = hipsnap.snapshot(arena, "/tmp/b9snap", "bench")
    out["snapshot_GBps"] = round(info["bytes"] / info["seconds"] / 1e9, 1)
    arena2 = hipsnap.Arena(device=0, size_bytes=8 << 30, staging_bytes=512 << 20)
    t0 = time.perf_counter()
    hipsnap.restore_snapshot(arena2, "bench", store_dir="/tmp/b9snap")
    dt = time.perf_counter() - t0
    out["restore_GBps"] = round(len(data) / dt / 1e9, 1)

    print(json.dumps(out, indent=2))


if __name__ == "__main__":
    main()
