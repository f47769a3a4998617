#!/usr/bin/env python3
"""Flagship serving bench: GPU-container cold start + sustained tasks/sec
through the full platform path.

BASELINE.json metric: "p50 GPU-container cold-start (ms) + tasks/sec at
1/2/4/8 concurrent MI355X". Each rank (one per GPU, launched by
torch.distributed.run for N>1) runs a complete in-process node — gateway +
scheduler + dispatcher + embedded worker pinned to its GPU — deploys the
bf16 synthetic MLP endpoint (random-init weights; no network for datasets),
and drives requests through the REAL dispatch path: gateway task dispatch ->
request buffer -> container subprocess (HTTP) -> hand-written gfx950 MFMA
kernels. A step = `--concurrency` requests per rank completed end-to-end.

Cold start (scale 0 -> first healthy response, containing HIP init + HBM
arena prealloc + weight init) is measured per rank during setup and reported
as cold_start_p50_ms; the timed region measures serving tasks/sec.
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

REPO = os.path.dirname(os.path.abspath(__file__))
if REPO not in sys.path:
    sys.path.insert(0, REPO)


def get_dist():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    return rank, world, local_rank


async def build_node(tmpdir: str, local_rank: int, args):
    """One in-process node: gateway + embedded worker pinned to one GPU."""
    from beta9_amd.common.config import AppConfig
    from beta9_amd.gateway.gateway import Gateway
    from beta9_amd.worker.gpu import GPUManager, discover_devices, _torch_fallback_devices
    from beta9_amd.worker.repo_client import DirectRepoClient
    from beta9_amd.worker.worker import Worker

    cfg = AppConfig().with_root(tmpdir)
    cfg.gateway.grpc_port = 0
    cfg.gateway.http_port = 0
    cfg.database.path = ":memory:"
    gw = Gateway(cfg)
    await gw.start()
    creds = gw.bootstrap_workspace("bench")

    devices = discover_devices(cfg.gpu.kfd_topology_path) or _torch_fallback_devices()
    mine = [d for d in devices if d.ordinal == local_rank]
    gpu_manager = GPUManager(devices=mine) if mine else None
    repo = DirectRepoClient(gw.worker_repo, gw.container_repo, events=gw.events)
    worker = Worker(
        worker_id=f"bench-w{local_rank}",
        repo=repo,
        cpu_limit=64000,
        memory_limit_mb=262144,
        gpu_type=gpu_manager.gpu_type if gpu_manager else "",
        gpu_manager=gpu_manager,
        gateway_host=cfg.gateway.host,
        gateway_grpc_port=gw.grpc_port,
        # GPU bench: shells for the GPU only (no CPU shells; at N=8 ranks
        # that is 16 fewer concurrent torch imports), CPU dev box: one
        prewarm_cpu_shells=0 if (gpu_manager and gpu_manager.total) else 1,
    )
    await worker.start()
    return gw, worker, creds, gpu_manager


async def deploy_endpoint(gw, creds, args, has_gpu: bool):
    from beta9_amd.types import RuntimeConfig, StubConfig, TaskPolicy

    cfg = StubConfig(
        runtime=RuntimeConfig(
            cpu=8000,
            memory_mb=65536,
            gpu="MI355X" if has_gpu else "",
            gpus=["MI355X"] if has_gpu else [],
            gpu_count=1 if has_gpu else 0,
        ),
        handler=("beta9_amd.models.bench_handler:infer_batched"
                 if getattr(args, "dyn_batch", False)
                 else "beta9_amd.models.bench_handler:infer"),
        on_start="beta9_amd.models.bench_handler:load",
        keep_warm_seconds=600.0,
        workers=1,
        concurrent_requests=max(32, args.concurrency * 2),
        env={
            "B9_BENCH_HIDDEN": str(args.hidden),
            "B9_BENCH_INTER": str(args.inter),
            "B9_BENCH_LAYERS": str(args.layers),
            "B9_BENCH_MODEL": args.model,
        },
        autoscaler=__import__("beta9_amd.types", fromlist=["AutoscalerConfig"]).AutoscalerConfig(
            max_containers=1, tasks_per_container=max(32, args.concurrency * 2)
        ),
        timeout=600.0,
        task_policy=TaskPolicy(max_retries=0, timeout=600.0),
    )
    ws = creds["workspace_id"]
    app = gw.db.get_or_create_app(ws, "bench")
    stub = gw.db.get_or_create_stub(ws, "bench-endpoint", "endpoint/deployment", cfg, app_id=app["id"])
    gw.db.create_deployment(ws, "bench-endpoint", stub["id"], "endpoint/deployment", app_id=app["id"])
    return stub["id"], ws


async def one_request(gw, ws: str, stub_id: str, tokens: int) -> dict:
    task = await gw.dispatcher.send_and_execute(
        "endpoint", ws, stub_id, {"method": "POST", "path": "/", "json": {"tokens": tokens}}
    )
    done = await gw.dispatcher.wait(task.task_id, timeout=600.0)
    result = await gw.dispatcher.get_result(task.task_id)
    ok = done.status.value == "COMPLETE" and isinstance(result, dict) and result.get("status_code") == 200
    return {"ok": ok, "result": result}


async def run_rank(args, rank: int, world: int, local_rank: int) -> dict:
    import torch

    with tempfile.TemporaryDirectory() as tmpdir:
        gw, worker, creds, gpu_manager = await build_node(tmpdir, local_rank, args)
        has_gpu = gpu_manager is not None and gpu_manager.total > 0
        stub_id, ws = await deploy_endpoint(gw, creds, args, has_gpu)
        # prewarm shells (torch+HIP context) park before the measured cold
        # start — they are part of worker boot, not of a container start
        await worker.wait_prewarmed(timeout=180.0)

        # ---- cold start: scale 0 -> first healthy response -----------------
        t0 = time.perf_counter()
        first = await one_request(gw, ws, stub_id, args.tokens)
        cold_ms = (time.perf_counter() - t0) * 1000.0
        if not first["ok"]:
            raise RuntimeError(f"cold-start request failed: {str(first)[:400]}")
        try:
            first_load = first["result"]["body"]["result"]["load"]
        except (KeyError, TypeError):
            first_load = {}
        boot_phases = {}
        try:
            addrs = await gw.container_repo.addresses_for_stub(stub_id)
            if addrs:
                from beta9_amd.rpc.channel import RpcClient

                rc = RpcClient(list(addrs.values())[0])
                h = await rc.call("Runner", "Health", {}, timeout=5.0)
                boot_phases = h.get("gpu_phases", {})
                await rc.close()
        except Exception:
            pass

        async def step() -> int:
            results = await asyncio.gather(
                *(one_request(gw, ws, stub_id, args.tokens) for _ in range(args.concurrency))
            )
            failed = sum(1 for r in results if not r["ok"])
            if failed:
                raise RuntimeError(f"{failed}/{len(results)} requests failed in step")
            return len(results)

        # ---- warmup --------------------------------------------------------
        for _ in range(args.warmup):
            await step()

        # ---- timed region --------------------------------------------------
        use_dist = world > 1
        if use_dist:
            import torch.distributed as dist

            if torch.cuda.is_available() and local_rank < torch.cuda.device_count():
                torch.cuda.set_device(local_rank)
            dist.barrier()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t_start = time.perf_counter()
        done_tasks = 0
        for _ in range(args.steps):
            done_tasks += await step()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        elapsed = time.perf_counter() - t_start
        if use_dist:
            import torch.distributed as dist

            dist.barrier()
            t = torch.tensor([elapsed], dtype=torch.float64)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elapsed_max = float(t.item())
            c = torch.tensor([cold_ms], dtype=torch.float64)
            colds = [torch.zeros(1, dtype=torch.float64) for _ in range(world)]
            dist.all_gather(colds, c)
            cold_list = [float(x.item()) for x in colds]
        else:
            elapsed_max = elapsed
            cold_list = [cold_ms]

        spans = {}
        try:
            cids = list((await gw.container_repo.containers_for_stub(stub_id)).keys())
            if cids:
                spans = {
                    s["name"]: round(s["duration_ms"], 2)
                    for s in await gw.container_repo.get_spans(cids[0])
                }
        except Exception:
            pass

        for w in [worker]:
            await w.shutdown()
        await gw.shutdown()

        return {
            "elapsed_max": elapsed_max,
            "tasks": done_tasks,
            "cold_list": cold_list,
            "native_gpu": has_gpu,
            "spans": spans,
            "first_load": first_load,
            "boot_phases": boot_phases,
        }


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--concurrency", type=int, default=32, help="requests per rank per step")
    p.add_argument("--tokens", type=int, default=256)
    p.add_argument("--hidden", type=int, default=4096)
    p.add_argument("--inter", type=int, default=14336)
    p.add_argument("--layers", type=int, default=8)
    p.add_argument("--model", choices=["mlp", "transformer"], default="mlp")
    p.add_argument("--dyn-batch", action="store_true",
                   help="serve through the dynamic-batching handler "
                        "(one coalesced forward per group of concurrent "
                        "requests; same total tokens)")
    args = p.parse_args()

    rank, world, local_rank = get_dist()
    if world > 1:
        import torch.distributed as dist

        dist.init_process_group(backend="gloo", init_method="env://")

    r = asyncio.run(run_rank(args, rank, world, local_rank))

    if rank == 0:
        n = world if world > 1 else args.gpus
        total_tasks = r["tasks"] * (world if world > 1 else 1)
        value = total_tasks / r["elapsed_max"]
        cold_sorted = sorted(r["cold_list"])
        p50 = statistics.median(cold_sorted)
        out = {
            "metric": "tasks_per_sec",
            "value": round(value, 3),
            "unit": "tasks/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(r["elapsed_max"] / args.steps * 1000.0, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "cold_start_p50_ms": round(p50, 1),
            "cold_start_all_ms": [round(x, 1) for x in cold_sorted],
            "cold_start_phases_ms": r["spans"],
            "first_container_load": r["first_load"],
            "container_boot_phases_ms": r["boot_phases"],
            "native_gpu_path": r["native_gpu"],
            "config": {
                "model": f"synthetic-{args.model}-{args.layers}L-{args.hidden}h-{args.inter}i",
                "global_batch": args.concurrency * (world if world > 1 else 1),
                "seq_len": args.tokens,
                "dyn_batch": bool(getattr(args, "dyn_batch", False)), "parallelism": f"serving dp{world if world > 1 else 1} (1 GPU-container per rank)",
                "requests_per_step_per_gpu": args.concurrency,
                "baseline_metric": "p50 GPU-container cold-start (ms) + tasks/sec at 1/2/4/8 concurrent MI355X",
            },
        }
        print(json.dumps(out))
    if world > 1:
        import torch.distributed as dist

        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
