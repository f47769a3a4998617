"""Multi-GPU endpoint runner: in-container RCCL-over-xGMI rendezvous.

For gpu_count>1 containers the worker provisions the rendezvous (loopback
MASTER_ADDR/PORT, WORLD_SIZE, xGMI-connected ordinal map in
B9_RDZV_ORDINALS — beta9_amd/worker/gpu.py provision_rendezvous); this
runner turns that into a running process group: one process per GPU (HIP
visibility pinned per rank BEFORE context creation), torch.distributed over
RCCL ("nccl" backend on ROCm; "gloo" on CPU test hosts), rank 0 serving
HTTP and broadcasting each request's args to the workers so handlers can run
tensor-parallel forward passes with RCCL collectives.

The reference has no equivalent — its multi-GPU containers just see N
devices and user code runs its own NCCL; here the
platform owns the bootstrap, which is what makes gpu_count=8 endpoints and
their checkpoints (one hipsnap arena per rank) first-class.
"""

from __future__ import annotations

import asyncio
import logging
import os
import subprocess
import sys
import time
from typing import Any, Dict, List, Optional

log = logging.getLogger("b9.runner.mgpu")

_STOP = {"__b9_stop__": True}


class Rendezvous:
    def __init__(self, rank: int, world: int, backend: str = ""):
        import torch
        import torch.distributed as dist

        self.torch = torch
        self.dist = dist
        if not backend:
            backend = os.environ.get("B9_RDZV_BACKEND", "")
        if not backend:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        self.backend = backend
        if backend == "nccl" and torch.cuda.is_available():
            torch.cuda.set_device(0)  # each rank sees exactly its GPU
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
        self.rank = rank
        self.world = world

    def broadcast(self, obj: Optional[Any] = None) -> Any:
        lst = [obj]
        self.dist.broadcast_object_list(lst, src=0)
        return lst[0]

    def shutdown(self) -> None:
        try:
            self.dist.destroy_process_group()
        except Exception:
            pass


_rendezvous: Optional[Rendezvous] = None


def get_rendezvous() -> Optional[Rendezvous]:
    """Handlers use this for their collectives (rank/world/process group)."""
    return _rendezvous


def _ordinals(world: int) -> List[str]:
    raw = os.environ.get("B9_RDZV_ORDINALS", "")
    if raw:
        return raw.split(",")
    return [str(i) for i in range(world)]


def _pin_env(env: Dict[str, str], ordinal: str, rank: int) -> None:
    env["RANK"] = str(rank)
    env["LOCAL_RANK"] = str(rank)
    env["HIP_VISIBLE_DEVICES"] = ordinal
    env["ROCR_VISIBLE_DEVICES"] = ordinal
    env["WORKER_GPU_DEVICES"] = ordinal


def run_multi_gpu(world: int) -> int:
    """Entry from the endpoint runner when WORLD_SIZE > 1."""
    rank = int(os.environ.get("RANK", "-1") or -1)
    ordinals = _ordinals(world)
    if rank < 0:
        # launcher: spawn ranks 1..world-1, then become rank 0
        procs = []
        for r in range(1, world):
            child = dict(os.environ)
            _pin_env(child, ordinals[r], r)
            procs.append(
                subprocess.Popen(
                    [sys.executable, "-u", "-m", "beta9_amd.runner.endpoint"],
                    env=child,
                    start_new_session=False,
                )
            )
        _pin_env(os.environ, ordinals[0], 0)
        try:
            return _serve_rank0(world, procs)
        finally:
            for p in procs:
                if p.poll() is None:
                    p.terminate()
            for p in procs:
                try:
                    p.wait(timeout=10)
                except subprocess.TimeoutExpired:
                    p.kill()
    elif rank == 0:
        return _serve_rank0(world, [])
    else:
        return _rank_worker(rank, world)


def _serve_rank0(world: int, procs: List[subprocess.Popen]) -> int:
    global _rendezvous
    from .common import GPUBoot, RunnerEnv, setup_environment

    env = RunnerEnv.from_env()
    env.gpu_devices = [0]
    setup_environment(env)
    rdzv = Rendezvous(0, world)
    _rendezvous = rdzv
    gpu = GPUBoot(env)
    gpu.boot()
    from .endpoint import run_rpc_server

    try:
        return asyncio.run(run_rpc_server(env, gpu, rdzv=rdzv))
    except KeyboardInterrupt:
        return 0
    finally:
        try:
            rdzv.broadcast(_STOP)
        except Exception:
            pass
        rdzv.shutdown()


def _rank_worker(rank: int, world: int) -> int:
    global _rendezvous
    from .common import (
        GPUBoot,
        RunnerEnv,
        call_handler,
        load_callable,
        run_on_start,
        setup_environment,
    )

    env = RunnerEnv.from_env()
    setup_environment(env)
    rdzv = Rendezvous(rank, world)
    _rendezvous = rdzv
    gpu = GPUBoot(env)
    gpu.boot()

    async def amain() -> int:
        await run_on_start(env)
        handler = load_callable(env.handler) if env.handler else None
        log.info("rank %d/%d ready (backend=%s)", rank, world, rdzv.backend)
        while True:
            args = rdzv.broadcast(None)
            if isinstance(args, dict) and args.get("__b9_stop__"):
                return 0
            try:
                if handler is not None:
                    await call_handler(handler, args)
            except Exception:
                log.exception("rank %d handler failed", rank)

    return asyncio.run(amain())


class Rank0Dispatcher:
    """Serializes distributed request dispatch on rank 0: broadcast the args
    to the workers, then run the handler locally (all ranks now execute the
    same handler call and can meet in RCCL collectives)."""

    def __init__(self, rdzv: Rendezvous):
        self.rdzv = rdzv
        self._lock = asyncio.Lock()

    async def dispatch(self, handler, args: Any):
        from .common import call_handler

        async with self._lock:
            loop = asyncio.get_event_loop()
            await loop.run_in_executor(None, self.rdzv.broadcast, args)
            return await call_handler(handler, args)

    def stop(self) -> None:
        try:
            self.rdzv.broadcast(_STOP)
        except Exception:
            pass
