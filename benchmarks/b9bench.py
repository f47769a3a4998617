#!/usr/bin/env python3
"""b9bench — suite-driven benchmark harness with anti-fooling validators.

(reference: benchmarks/b9bench — YAML suite definitions, percentile
reporting, validators asserting evidence so a broken/fake run cannot pass:
b9bench/README.md, validators.py, suite_defs/*.yaml.)

    python benchmarks/b9bench.py --suite benchmarks/suites/startup-default.yaml
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

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests", "fixtures"))
# runner subprocesses need the fixture handlers importable too
os.environ["PYTHONPATH"] = os.pathsep.join(
    [REPO, os.path.join(REPO, "tests", "fixtures")]
    + ([os.environ["PYTHONPATH"]] if os.environ.get("PYTHONPATH") else [])
)


class ValidationError(AssertionError):
    pass


def validate_distinct_containers(evidence: list) -> None:
    ids = [e["container_id"] for e in evidence]
    if len(set(ids)) != len(ids):
        raise ValidationError("container ids are not distinct")


def validate_exec_markers(evidence: list) -> None:
    for e in evidence:
        if e.get("marker") and e["marker"] not in e.get("output", ""):
            raise ValidationError(f"marker missing for {e['container_id']}")


def validate_task_rows(evidence: list, db) -> None:
    for e in evidence:
        if e.get("task_id"):
            db.flush()
            row = db.get_task(e["task_id"])
            if row is None:
                raise ValidationError(f"no task row for {e['task_id']}")


async def run_suite(spec: dict) -> dict:
    from tests.helpers import Cluster

    kind = spec["kind"]
    iters = int(spec.get("iterations", 10))
    warmup = int(spec.get("warmup", 1))
    parallelism = int(spec.get("parallelism", 1))
    with tempfile.TemporaryDirectory() as tmp:
        c = Cluster(tmp)
        await c.start(n_workers=int(spec.get("workers", 1)))
        try:
            import beta9_amd

            evidence: list = []
            latencies: list = []

            async def one_startup(i: int) -> None:
                sb_def = beta9_amd.Sandbox(
                    cpu=int(spec.get("cpu", 100)), memory=int(spec.get("memory_mb", 192))
                )
                t0 = time.perf_counter()
                sb = await asyncio.to_thread(sb_def.create, float(spec.get("ready_timeout_s", 240)))
                dt = (time.perf_counter() - t0) * 1000
                marker = f"ev-{i}-{os.getpid()}"
                out = (await asyncio.to_thread(sb.exec, f"echo {marker}"))["stdout"]
                evidence.append({"container_id": sb.container_id, "marker": marker, "output": out})
                latencies.append(dt)
                await asyncio.to_thread(sb.terminate)

            async def one_function(i: int) -> None:
                import handlers  # fixtures on path via tests.helpers import

                fn = beta9_amd.Function(handlers.hello, cpu=100, memory=128)
                t0 = time.perf_counter()
                r = await asyncio.to_thread(fn.remote, i)
                dt = (time.perf_counter() - t0) * 1000
                if r != {"hello": i * 2}:
                    raise ValidationError(f"wrong result {r}")
                latencies.append(dt)
                evidence.append({"container_id": f"fn-{i}", "task_id": None})

            runner = one_startup if kind == "sandbox-startup" else one_function
            sem = asyncio.Semaphore(parallelism)

            async def bounded(i):
                async with sem:
                    await runner(i)

            for i in range(warmup):
                await runner(10_000 + i)
            latencies.clear()
            evidence.clear()
            t_all = time.perf_counter()
            await asyncio.gather(*(bounded(i) for i in range(iters)))
            wall = time.perf_counter() - t_all

            # validators (anti-fooling: reference b9bench/validators.py)
            validate_distinct_containers(evidence)
            if kind == "sandbox-startup":
                validate_exec_markers(evidence)

            latencies.sort()
            return {
                "suite": spec.get("name", kind),
                "iterations": iters,
                "parallelism": parallelism,
                "p50_ms": round(statistics.median(latencies), 1),
                "p95_ms": round(latencies[int(0.95 * (len(latencies) - 1))], 1),
                "wall_s": round(wall, 2),
                "validated": True,
            }
        finally:
            await c.stop()


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--suite", required=True)
    args = p.parse_args()
    with open(args.suite) as f:
        spec = yaml.safe_load(f)
    result = asyncio.run(run_suite(spec))
    print(json.dumps(result))
    return 0


if __name__ == "__main__":
    sys.exit(main())
