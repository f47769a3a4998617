"""Generic autoscaler + autoscaled instance.

Reference machinery (reference: pkg/abstractions/common/autoscaler.go:13-74 —
1 Hz sampling with windowed samples + manual trigger; instance.go:57-326 —
AutoscaledInstance diffing desired vs current and calling start/stop
functions; instance.go:444-640 InstanceController rehydration).

Scale functions reproduced:
- endpoint deployment: containers = ceil(inflight / concurrent_requests),
  capped by max_containers and gateway stub limits
  (reference: endpoint/autoscaler.go:39-76)
- endpoint/taskqueue serve: 1 while the serve lock exists, else 0
  (reference: endpoint/autoscaler.go:77-97)
- taskqueue deployment: containers = ceil(queue_len / tasks_per_container)
  (reference: taskqueue/autoscaler.go:53-80)
"""

from __future__ import annotations

import asyncio
import logging
import math
import time
from typing import Awaitable, Callable, Dict, List, Optional

from ...common import keys
from ...common.state import StateStore
from ...types import ContainerRequest, StubConfig

log = logging.getLogger(__name__)

SampleFn = Callable[[], Awaitable[float]]  # returns the load sample
# returns desired container count given the sample
ScaleFn = Callable[[float], int]


class Autoscaler:
    def __init__(
        self,
        sample_fn: SampleFn,
        scale_fn: ScaleFn,
        apply_fn: Callable[[int], Awaitable[None]],
        interval_s: float = 1.0,
        window: int = 60,
    ):
        self.sample_fn = sample_fn
        self.scale_fn = scale_fn
        self.apply_fn = apply_fn
        self.interval_s = interval_s
        self.window = window
        self.samples: List[float] = []
        self._task: Optional[asyncio.Task] = None
        self._trigger = asyncio.Event()

    def start(self) -> None:
        if self._task is None:
            self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None

    def trigger(self) -> None:
        """Manual trigger: scale immediately (e.g. on request arrival)."""
        self._trigger.set()

    async def step(self) -> int:
        sample = await self.sample_fn()
        self.samples.append(sample)
        if len(self.samples) > self.window:
            self.samples.pop(0)
        desired = self.scale_fn(sample)
        await self.apply_fn(desired)
        return desired

    async def _loop(self) -> None:
        while True:
            try:
                self._trigger.clear()
                await self.step()
                try:
                    await asyncio.wait_for(self._trigger.wait(), timeout=self.interval_s)
                except asyncio.TimeoutError:
                    pass
            except asyncio.CancelledError:
                raise
            except Exception:
                log.exception("autoscaler step failed")
                await asyncio.sleep(self.interval_s)


def deployment_scale_fn(cfg: StubConfig, stub_limit_max_replicas: int) -> ScaleFn:
    per = max(1, cfg.autoscaler.tasks_per_container)
    max_c = min(max(1, cfg.autoscaler.max_containers), stub_limit_max_replicas)
    min_c = max(0, cfg.autoscaler.min_containers)

    def fn(sample: float) -> int:
        desired = math.ceil(sample / per)
        return max(min_c, min(desired, max_c))

    return fn


class AutoscaledInstance:
    """Holds a stub's container set; reconciles desired vs running."""

    def __init__(
        self,
        gateway,  # Gateway
        stub_id: str,
        cfg: StubConfig,
        stub_row: Dict,
        make_request_fn: Callable[[], ContainerRequest],
        sample_fn: SampleFn,
        scale_fn: Optional[ScaleFn] = None,
        serve_mode: bool = False,
    ):
        self.gw = gateway
        self.store: StateStore = gateway.store
        self.stub_id = stub_id
        self.cfg = cfg
        self.stub_row = stub_row
        self.make_request_fn = make_request_fn
        self.serve_mode = serve_mode
        if scale_fn is None:
            scale_fn = deployment_scale_fn(cfg, gateway.cfg.gateway.stub_limits_max_replicas)
        if serve_mode:
            inner = scale_fn

            def serve_scale(sample: float) -> int:
                return 1  # while the serve lock exists (checked in apply)

            scale_fn = serve_scale
        self.autoscaler = Autoscaler(sample_fn, scale_fn, self._apply)

    def start(self) -> None:
        self.autoscaler.start()

    async def shutdown(self) -> None:
        await self.autoscaler.stop()
        for cid in list((await self.gw.container_repo.containers_for_stub(self.stub_id)).keys()):
            await self.gw.scheduler.stop(cid)

    async def current_containers(self) -> List[str]:
        return list((await self.gw.container_repo.containers_for_stub(self.stub_id)).keys())

    async def _apply(self, desired: int) -> None:
        if self.serve_mode:
            lock_key = keys.scheduler_serve_lock(self.stub_row["workspace_id"], self.stub_id)
            if await self.store.get(lock_key) is None:
                desired = 0
        current = await self.current_containers()
        n = len(current)
        if desired > n:
            for _ in range(desired - n):
                request = self.make_request_fn()
                await self.gw.scheduler.run(request)
        elif desired < n:
            await self._stop_excess(current, n - desired)

    async def _stop_excess(self, current: List[str], count: int) -> None:
        """Stop containers lacking a live keep-warm key (reference:
        endpoint/instance.go:152-228 — keep-warm key per container with TTL
        refreshed after each request; containers lacking it are stoppable)."""
        stopped = 0
        for cid in current:
            if stopped >= count:
                return
            kw = await self.store.get(keys.container_keep_warm(self.stub_id, cid))
            if kw is None:
                await self.gw.scheduler.stop(cid)
                stopped += 1
