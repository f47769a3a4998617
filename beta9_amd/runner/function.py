"""Function runner: one task per container.

StartTask -> GetTaskArgs -> handler() -> EndTask (reference:
sdk/src/beta9/runner/function.py; result flow pkg/abstractions/function/
function.go FunctionGetArgs/SetResult).
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

log = logging.getLogger("b9.runner.function")


async def main() -> int:
    env = RunnerEnv.from_env()
    setup_environment(env)
    gpu = GPUBoot(env)
    gpu.boot()
    client = RunnerClient(env)
    task_id = env.task_id
    if not task_id:
        log.error("no TASK_ID set for function runner")
        return 2
    try:
        ok = await client.start_task(task_id)
        if not ok:
            log.error("task %s already claimed", task_id)
            return 3
        client.start_heartbeat(task_id)
        try:
            await run_on_start(env)
            args = await client.get_args(task_id)
            fn = load_callable(env.handler)
            result = await asyncio.wait_for(call_handler(fn, args), timeout=env.timeout)
            await client.end_task(task_id, result, status="COMPLETE")
            return 0
        except asyncio.TimeoutError:
            await client.end_task(task_id, None, status="TIMEOUT")
            return 4
        except Exception as e:
            log.exception("handler failed")
            await client.end_task(task_id, {"error": str(e)}, status="ERROR")
            return 1
        finally:
            client.stop_heartbeat()
    finally:
        await client.close()


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    sys.exit(asyncio.run(main()))
