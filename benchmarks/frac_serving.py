#!/usr/bin/env python3
"""Fractional GPU sharing demo (Thunder analog — pkg/worker/thunder.go):
two gpu="MI355X:0.5" containers run serving bursts CONCURRENTLY on ONE
device through the full platform (SDK -> gateway -> scheduler -> worker
-> runner, HBM budget fenced per process). Reports solo vs shared
throughput so the time-sharing cost is measured, not guessed.

Run on a 1-GPU box: python benchmarks/frac_serving.py
"""

import asyncio
import json
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
FIXTURES = os.path.join(REPO, "tests", "fixtures")
sys.path.insert(0, REPO)
sys.path.insert(0, FIXTURES)
sys.path.insert(0, os.path.join(REPO, "tests"))
# container runners import the handler by module name — they need the
# fixtures dir on PYTHONPATH too (same as tests/conftest.py does)
_pp = os.environ.get("PYTHONPATH", "")
os.environ["PYTHONPATH"] = os.pathsep.join(
    [FIXTURES, REPO] + ([_pp] if _pp else [])
)


async def main() -> None:
    import beta9_amd
    import handlers
    from tests.helpers import cluster

    from beta9_amd.worker.gpu import GPUManager
    from beta9_amd.worker.repo_client import DirectRepoClient
    from beta9_amd.worker.worker import Worker

    steps = int(os.environ.get("FRAC_STEPS", "150"))
    async with cluster(n_workers=1) as c:
        gm = GPUManager()
        assert gm.total >= 1, "no GPU visible"
        repo = DirectRepoClient(c.gateway.worker_repo, c.gateway.container_repo)
        w = Worker(
            worker_id="gpuw", repo=repo, cpu_limit=32000, memory_limit_mb=131072,
            gpu_type=gm.gpu_type, gpu_manager=gm,
            gateway_host="127.0.0.1", gateway_grpc_port=c.gateway.grpc_port,
        )
        await w.start()
        half = beta9_amd.Function(
            handlers.frac_serving_steps, cpu=4000, memory=16384, gpu="MI355X:0.5"
        )
        # solo baseline: one half-GPU container alone on the device
        solo = await asyncio.to_thread(half.remote, steps)
        # shared: two half-GPU containers concurrently on the SAME device
        a, b = await asyncio.gather(
            asyncio.to_thread(half.remote, steps),
            asyncio.to_thread(half.remote, steps),
        )
        assert a["visible"] == b["visible"], (a, b)
        assert a["fraction"] == "0.5" and b["fraction"] == "0.5"
        agg = a["steps_per_s"] + b["steps_per_s"]
        print(json.dumps({
            "solo_steps_per_s": round(solo["steps_per_s"], 1),
            "shared_each_steps_per_s": [round(a["steps_per_s"], 1), round(b["steps_per_s"], 1)],
            "shared_aggregate_steps_per_s": round(agg, 1),
            "aggregate_vs_solo": round(agg / solo["steps_per_s"], 3),
            "ms_per_step_solo": round(solo["ms_per_step"], 3),
            "ms_per_step_shared": [round(a["ms_per_step"], 3), round(b["ms_per_step"], 3)],
            "device": a["visible"],
            "steps": steps,
        }))
        await w.shutdown()


if __name__ == "__main__":
    asyncio.run(main())
