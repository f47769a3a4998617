#!/usr/bin/env python3
"""Serving-step soak for rocprofv3: N forward passes of the synthetic
serving model at the bench shape. --backend fp8 runs the full w8a8 chain
(rmsnorm_fp8 -> gate-up Q8 -> fp8 down + fused residual)."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--tokens", type=int, default=256)
    p.add_argument("--hidden", type=int, default=4096)
    p.add_argument("--inter", type=int, default=14336)
    p.add_argument("--layers", type=int, default=8)
    p.add_argument("--backend", default="")
    args = p.parse_args()
    if args.backend:
        os.environ["B9_GEMM_BACKEND"] = args.backend
    import torch

    from beta9_amd.models.mlp import SyntheticServingModel

    m = SyntheticServingModel(args.hidden, args.inter, args.layers, "cuda:0")
    for _ in range(3):  # warmup + fp8 calibration calls
        m.infer(batch_tokens=args.tokens)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        m.infer(batch_tokens=args.tokens)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{args.steps} steps in {dt*1000:.1f} ms = {dt/args.steps*1000:.3f} ms/step "
          f"backend={args.backend or 'native'}")


if __name__ == "__main__":
    main()
