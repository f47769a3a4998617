#!/usr/bin/env python3
"""@task_queue sustained throughput (BASELINE config 4: queue-depth
autoscaler across MI355X GPUs; on one box: 1 GPU, N queue workers).

    python benchmarks/taskqueue_bench.py --tasks 100 [--gpu]
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
os.environ["PYTHONPATH"] = os.pathsep.join(
    [REPO] + ([os.environ["PYTHONPATH"]] if os.environ.get("PYTHONPATH") else [])
)


async def main(n_tasks: int, use_gpu: bool, tokens: int) -> dict:
    from tests.helpers import Cluster

    with tempfile.TemporaryDirectory() as tmp:
        c = Cluster(tmp)
        await c.start(n_workers=1)
        if use_gpu:
            from beta9_amd.worker.gpu import GPUManager
            from beta9_amd.worker.repo_client import DirectRepoClient
            from beta9_amd.worker.worker import Worker

            gm = GPUManager()
            use_gpu = gm.total > 0
            if use_gpu:
                repo = DirectRepoClient(c.gateway.worker_repo, c.gateway.container_repo, events=c.gateway.events)
                w = Worker(
                    worker_id="tq-gpu", repo=repo, cpu_limit=64000, memory_limit_mb=262144,
                    gpu_type=gm.gpu_type, gpu_manager=gm,
                    gateway_host="127.0.0.1", gateway_grpc_port=c.gateway.grpc_port,
                )
                await w.start()
                await w.wait_prewarmed(180.0)
                c.workers.append(w)
        try:
            from beta9_amd.types import AutoscalerConfig, RuntimeConfig, StubConfig

            cfg = StubConfig(
                runtime=RuntimeConfig(
                    cpu=4000, memory_mb=32768,
                    gpu="MI355X" if use_gpu else "", gpu_count=1 if use_gpu else 0,
                    gpus=["MI355X"] if use_gpu else [],
                ),
                handler="beta9_amd.models.bench_handler:infer",
                on_start="beta9_amd.models.bench_handler:load",
                keep_warm_seconds=120,
                workers=4,
                env={"B9_BENCH_HIDDEN": "4096", "B9_BENCH_INTER": "14336", "B9_BENCH_LAYERS": "8"},
                autoscaler=AutoscalerConfig(max_containers=1 if use_gpu else 4, tasks_per_container=8),
                timeout=600,
            )
            ws = c.creds["workspace_id"]
            stub = c.gateway.db.get_or_create_stub(ws, "tq-bench", "taskqueue/deployment", cfg)

            # enqueue everything, then measure drain
            t_enq = time.perf_counter()
            task_ids = []
            for i in range(n_tasks):
                t = await c.gateway.dispatcher.send_and_execute(
                    "taskqueue", ws, stub["id"], {"args": [], "kwargs": {"tokens": tokens}}
                )
                task_ids.append(t.task_id)
            enq_s = time.perf_counter() - t_enq

            t0 = time.perf_counter()
            done = 0
            failed = 0
            deadline = t0 + 600
            pending = set(task_ids)
            while pending and time.perf_counter() < deadline:
                for tid in list(pending):
                    t = await c.gateway.dispatcher.get_task(tid)
                    if t and t.status.is_terminal:
                        pending.discard(tid)
                        if t.status.value == "COMPLETE":
                            done += 1
                        else:
                            failed += 1
                await asyncio.sleep(0.05)
            wall = time.perf_counter() - t0
            return {
                "suite": "taskqueue-sustained",
                "gpu": use_gpu,
                "tasks": n_tasks,
                "completed": done,
                "failed": failed,
                "enqueue_s": round(enq_s, 2),
                "drain_s": round(wall, 2),
                "tasks_per_sec": round(done / wall, 2) if wall else 0,
            }
        finally:
            await c.stop()


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--tasks", type=int, default=50)
    p.add_argument("--tokens", type=int, default=256)
    p.add_argument("--gpu", action="store_true")
    args = p.parse_args()
    print(json.dumps(asyncio.run(main(args.tasks, args.gpu, args.tokens))))
