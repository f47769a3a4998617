#!/usr/bin/env python3
"""Parallel sandbox creation benchmark — the reference's headline suite
(reference: benchmarks/sandbox_parallel.py + b9bench suite_defs/
sandbox-stage-cold.yaml: N sandboxes at parallelism N, ready-timeout,
exec-output verification; validators.py anti-fooling checks).

Usage: python benchmarks/sandbox_parallel.py [--n 20] [--gpu] [--json]
Runs against an in-process cluster (gateway + embedded worker) and reports
p50/p95 create latency with exec-evidence validation (each sandbox must
execute a unique command and echo it back — a scheduler that fakes creates
fails validation).
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import statistics
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


async def main(n: int, use_gpu: bool, parallelism: int) -> dict:
    from concurrent.futures import ThreadPoolExecutor

    from tests.helpers import Cluster

    # to_thread's default pool is 32 threads; each blocking create() holds
    # one, so parallelism>32 would starve the exec/terminate calls
    asyncio.get_event_loop().set_default_executor(
        ThreadPoolExecutor(max_workers=parallelism * 3 + 8)
    )
    with tempfile.TemporaryDirectory() as tmp:
        c = Cluster(tmp)
        await c.start(n_workers=1, cpu_per_worker=max(64000, n * 200))
        try:
            import beta9_amd

            sem = asyncio.Semaphore(parallelism)
            latencies = []
            evidence = []

            async def one(i: int) -> None:
                async with sem:
                    sb_def = beta9_amd.Sandbox(cpu=100, memory=192)
                    t0 = time.perf_counter()
                    sb = await asyncio.to_thread(sb_def.create, 240.0)
                    dt = (time.perf_counter() - t0) * 1000
                    # evidence: unique exec output + distinct container ids
                    marker = f"sb-evidence-{i}-{os.getpid()}"
                    r = await asyncio.to_thread(sb.exec, f"echo {marker}")
                    assert marker in r["stdout"], f"evidence failed for {i}"
                    latencies.append(dt)
                    evidence.append(sb.container_id)
                    await asyncio.to_thread(sb.terminate)

            t_all = time.perf_counter()
            await asyncio.gather(*(one(i) for i in range(n)))
            wall = time.perf_counter() - t_all

            assert len(set(evidence)) == n, "container ids not distinct"
            latencies.sort()
            return {
                "suite": "sandbox-parallel-cold",
                "n": n,
                "parallelism": parallelism,
                "p50_ms": round(statistics.median(latencies), 1),
                "p95_ms": round(latencies[int(0.95 * (len(latencies) - 1))], 1),
                "max_ms": round(latencies[-1], 1),
                "wall_s": round(wall, 2),
                "creates_per_sec": round(n / wall, 2),
                "validated": True,
            }
        finally:
            await c.stop()


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=20)
    p.add_argument("--parallelism", type=int, default=0, help="0 = n")
    p.add_argument("--gpu", action="store_true")
    args = p.parse_args()
    result = asyncio.run(main(args.n, args.gpu, args.parallelism or args.n))
    print(json.dumps(result))
