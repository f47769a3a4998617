#!/usr/bin/env python3
"""Minimal profiling targets for rocprofv3 runs (one workload per
invocation so kernel stats stay attributable)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    what = sys.argv[1] if len(sys.argv) > 1 else "gemm"
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    from beta9_amd.ops import gemm as ops

    if what == "gemm":
        a = (torch.randn(4096, 4096, device="cuda") * 0.5).to(torch.bfloat16)
        w = (torch.randn(4096, 4096, device="cuda") * 0.5).to(torch.bfloat16)
        for _ in range(iters):
            ops.gemm_nt(a, w)
    elif what == "mlp":
        from beta9_amd.models.mlp import SyntheticServingModel

        m = SyntheticServingModel(hidden=4096, inter=14336, layers=8)
        for _ in range(iters):
            m.infer(batch_tokens=256)
    elif what == "stream":
        from beta9_amd.ops import hipsnap

        arena = hipsnap.Arena(device=0, size_bytes=8 << 30, staging_bytes=512 << 20)
        data = os.urandom(1 << 30)
        off = arena.alloc("b", len(data))
        for _ in range(iters):
            arena.upload(off, data)
    torch.cuda.synchronize()
    print("done", what)


if __name__ == "__main__":
    main()
