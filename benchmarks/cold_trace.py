#!/usr/bin/env python3
"""Cold-start hop timeline (diagnostic): stamps gateway-side hops and
aligns runner-side wall-clock stamps from container logs."""
import argparse
import asyncio
import os
import sys
import time
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
print("importing bench...", flush=True)
import bench  # noqa: E402
print("bench imported", flush=True)

T0 = time.perf_counter()
W0 = time.time()


def stamp(name):
    print(f"  [{(time.perf_counter()-T0)*1000:8.2f}ms] {name}", flush=True)


async def main():
    global T0, W0
    small = bool(os.environ.get("B9_TRACE_SMALL"))
    args = argparse.Namespace(gpus=1, steps=3, warmup=1, concurrency=4, tokens=64,
                              hidden=256 if small else 4096,
                              inter=512 if small else 14336,
                              layers=2 if small else 8, model="mlp")
    print("building node...", flush=True)
    with tempfile.TemporaryDirectory() as tmpdir:
        gw, worker, creds, gpu_manager = await bench.build_node(tmpdir, 0, args)
        has_gpu = gpu_manager is not None and gpu_manager.total > 0
        stub_id, ws = await bench.deploy_endpoint(gw, creds, args, has_gpu)
        print("deployed; waiting prewarm...", flush=True)
        await worker.wait_prewarmed(timeout=180.0)
        print("prewarmed", flush=True)

        sched_run = gw.scheduler.run
        async def run2(req):
            stamp(f"scheduler.run({req.container_id})")
            return await sched_run(req)
        gw.scheduler.run = run2

        orig_rc = worker.lifecycle.run_container
        async def rc2(req):
            stamp("worker.run_container enter")
            r = await orig_rc(req)
            stamp("worker.run_container returned")
            return r
        worker.lifecycle.run_container = rc2

        from beta9_amd.abstractions.common.buffer import RequestBuffer

        async def probe2(self, slot):
            try:
                if slot.rpc is None:
                    from beta9_amd.rpc.channel import RpcClient
                    slot.rpc = RpcClient(slot.addr)
                r = await slot.rpc.call("Runner", "Health", {}, timeout=2.0)
                stamp(f"probe {slot.addr} -> ok={r.get('ok')}")
                return bool(r.get("ok"))
            except Exception as e:
                stamp(f"probe {slot.addr} EXC {type(e).__name__}: {str(e)[:90]}")
                if slot.rpc is not None:
                    rpc, slot.rpc = slot.rpc, None
                    asyncio.ensure_future(rpc.close())
                return False
        RequestBuffer._probe = probe2

        T0 = time.perf_counter(); W0 = time.time()
        stamp("first request start")
        r = await bench.one_request(gw, ws, stub_id, args.tokens)
        stamp(f"first request done ok={r['ok']}")
        for key in list(gw.store._lists.keys()):
            if "log" in str(key):
                for it in gw.store._lists.get(key, []):
                    line = it.get("line", "") if isinstance(it, dict) else str(it)
                    if "boottrace" in line:
                        try:
                            ts = float(line.strip().rsplit(" ", 1)[-1])
                            print(f"  [{(ts-W0)*1000:8.2f}ms] {line.strip().rsplit(' ', 1)[0]}", flush=True)
                        except ValueError:
                            print("  " + line.strip(), flush=True)
        await worker.shutdown(); await gw.shutdown()


if __name__ == "__main__":
    asyncio.run(main())
