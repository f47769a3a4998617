#!/usr/bin/env python3
"""Concurrent serving soak for rocprofv3: N threads each run the serving
model on their own HIP stream (ops/streams.py) — the shape the runner's
thread pool produces under concurrent requests. Use with
`rocprofv3 --stats` to show overlapped first-party kernels."""

import argparse
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--threads", type=int, default=8)
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
    from beta9_amd.ops import streams

    streams.prewarm(args.threads)
    m = SyntheticServingModel(args.hidden, args.inter, args.layers, "cuda:0")
    for _ in range(3):  # warmup + fp8 calibration
        m.infer(batch_tokens=args.tokens)
    torch.cuda.synchronize()

    def worker():
        for _ in range(args.steps):
            m.infer(batch_tokens=args.tokens)

    t0 = time.perf_counter()
    ts = [threading.Thread(target=worker) for _ in range(args.threads)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    total = args.threads * args.steps
    print(f"{total} infers across {args.threads} threads in {dt*1000:.1f} ms "
          f"= {total/dt:.0f} infers/s ({dt/total*1000:.3f} ms each amortized) "
          f"backend={args.backend or 'native'}")


if __name__ == "__main__":
    main()
