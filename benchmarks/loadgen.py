#!/usr/bin/env python3
"""Load generator subprocess for bench.py.

Serving clients are not part of the gateway process in production; running
the request loop inside rank 0 (which hosts the gateway + worker) makes
the shared event loop the bottleneck and under-reports platform
throughput. Each bench rank spawns one of these per stub instead.

Protocol (stdin/stdout lines):
  argv: addr token stub_id concurrency steps warmup tokens timeout
  1. warms up (`warmup` steps of `concurrency` concurrent invokes)
  2. prints READY, waits for GO on stdin (parent holds the dist barrier)
  3. runs `steps` timed steps, prints one JSON line {"elapsed", "tasks"}
"""

import asyncio
import json
import sys
import time


async def invoke(client, stub_id: str, tokens: int, affinity: str, timeout: float) -> None:
    r = await client.call(
        "EndpointService",
        "EndpointInvoke",
        {"stub_id": stub_id, "args": {"tokens": tokens}, "affinity": affinity,
         "timeout": timeout},
        timeout=timeout + 20.0,
    )
    ok = (
        r.get("status") == "COMPLETE"
        and isinstance(r.get("result"), dict)
        and r["result"].get("status_code") == 200
    )
    if not ok:
        raise RuntimeError(f"invoke failed: {str(r)[:300]}")


async def main() -> int:
    addr, token, stub_id = sys.argv[1:4]
    concurrency, steps, warmup, tokens = (int(x) for x in sys.argv[4:8])
    timeout = float(sys.argv[8])

    from beta9_amd.rpc.channel import RpcClient

    client = RpcClient(addr, token=token)

    async def step() -> int:
        await asyncio.gather(
            *(invoke(client, stub_id, tokens, "lg", timeout) for _ in range(concurrency))
        )
        return concurrency

    for _ in range(warmup):
        await step()
    print("READY", flush=True)
    line = await asyncio.to_thread(sys.stdin.readline)
    if line.strip() != "GO":
        raise RuntimeError(f"bad GO line: {line!r}")
    done = 0
    t0 = time.perf_counter()
    for _ in range(steps):
        done += await step()
    elapsed = time.perf_counter() - t0
    print(json.dumps({"elapsed": elapsed, "tasks": done}), flush=True)
    await client.close()
    return 0


if __name__ == "__main__":
    sys.exit(asyncio.run(main()))
