"""Task-queue runner: worker loop popping tasks from the queue.

TaskQueuePop -> claim -> handler -> TaskQueueComplete, with heartbeats and
cancellation monitoring (reference: sdk/src/beta9/runner/taskqueue.py:166-405).
Runs until the container is stopped by the autoscaler (keep-warm governs).
"""

from __future__ import annotations

import asyncio
import logging
import sys

from .common import (
    GPUBoot,
    RunnerClient,
    RunnerEnv,
    call_handler,
    load_callable,
    run_on_start,
    setup_environment,
)

log = logging.getLogger("b9.runner.taskqueue")


async def worker_loop(env: RunnerEnv, client: RunnerClient) -> None:
    fn = load_callable(env.handler)
    while True:
        try:
            r = await client.rpc.call(
                "TaskQueueService",
                "TaskQueuePop",
                {"stub_id": env.stub_id, "container_id": env.container_id, "timeout": 1.0},
                timeout=15.0,
            )
        except Exception:
            log.exception("pop failed; retrying")
            await asyncio.sleep(0.5)
            continue
        task_id = r.get("task_id")
        if not task_id:
            continue
        client.start_heartbeat(task_id)
        status, result = "COMPLETE", None
        try:
            result = await asyncio.wait_for(call_handler(fn, r.get("args")), timeout=env.timeout)
        except asyncio.TimeoutError:
            status = "TIMEOUT"
        except Exception as e:
            log.exception("task %s failed", task_id)
            status, result = "ERROR", {"error": str(e)}
        finally:
            client.stop_heartbeat()
        try:
            await client.rpc.call(
                "TaskQueueService",
                "TaskQueueComplete",
                {"task_id": task_id, "result": result, "status": status},
            )
            await client.set_keep_warm()
        except Exception:
            log.exception("complete failed for %s", task_id)


async def main() -> int:
    env = RunnerEnv.from_env()
    setup_environment(env)
    gpu = GPUBoot(env)
    gpu.boot()
    client = RunnerClient(env)
    try:
        await run_on_start(env)
        if env.checkpoint_enabled and not gpu.restored:
            try:
                await client.checkpoint_ready()
            except Exception:
                pass
        # `workers` concurrent pop loops within one process (reference runs
        # N worker processes; event-loop concurrency covers the same).
        loops = [asyncio.create_task(worker_loop(env, client)) for _ in range(max(1, env.workers))]
        await asyncio.gather(*loops)
        return 0
    finally:
        await client.close()


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    sys.exit(asyncio.run(main()))
