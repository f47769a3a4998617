"""Dynamic request batching for endpoint runners.

Opt-in via the @beta9_amd.batchable decorator: concurrent Invoke calls are
coalesced (up to max_size, or wait_ms after the first arrival) into ONE
handler call `handler(batch=[args, ...]) -> [result, ...]`, so one GPU
forward serves every in-flight request — the dynamic-batching idea every
serving engine uses (the reference gets it implicitly from vLLM inside its
vllm integration; here it is a first-class platform feature usable by any
handler). No request's work is skipped: the handler processes each batch
element; batching only shares the kernel-launch/weight-read cost.
"""

from __future__ import annotations

import asyncio
import inspect
import logging
from typing import Any, Callable, List, Optional

log = logging.getLogger(__name__)


class DynamicBatcher:
    def __init__(self, handler: Callable, max_size: int = 32, wait_ms: float = 2.0):
        from .common import unwrap_handler

        self.fn = unwrap_handler(handler)
        self.max_size = max(1, max_size)
        self.wait_s = max(0.0, wait_ms) / 1000.0
        self._pending: List[tuple] = []  # (args, future)
        self._flush_handle: Optional[asyncio.TimerHandle] = None
        self._lock = asyncio.Lock()

    async def submit(self, args: Any) -> Any:
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        async with self._lock:
            self._pending.append((args, fut))
            if len(self._pending) >= self.max_size:
                self._schedule_flush(0.0)
            elif self._flush_handle is None:
                self._schedule_flush(self.wait_s)
        return await fut

    def _schedule_flush(self, delay: float) -> None:
        if self._flush_handle is not None:
            self._flush_handle.cancel()
        loop = asyncio.get_event_loop()
        self._flush_handle = loop.call_later(
            delay, lambda: asyncio.ensure_future(self._flush())
        )

    async def _flush(self) -> None:
        async with self._lock:
            self._flush_handle = None
            batch, self._pending = self._pending, []
        if not batch:
            return
        args_list = [a for a, _ in batch]
        try:
            if inspect.iscoroutinefunction(self.fn):
                results = await self.fn(batch=args_list)
            else:
                results = await asyncio.get_event_loop().run_in_executor(
                    None, lambda: self.fn(batch=args_list)
                )
            if not isinstance(results, list) or len(results) != len(batch):
                raise ValueError(
                    f"batchable handler must return a list of {len(batch)} results"
                )
            for (_, fut), r in zip(batch, results):
                if not fut.done():
                    fut.set_result(r)
        except Exception as e:
            log.exception("batched handler failed")
            for _, fut in batch:
                if not fut.done():
                    fut.set_exception(e)


def batchable(max_size: int = 32, wait_ms: float = 2.0):
    """Mark a handler for dynamic batching: it will be called as
    fn(batch=[args, ...]) and must return a list of per-request results."""

    def deco(fn: Callable) -> Callable:
        fn._b9_batch = {"max_size": max_size, "wait_ms": wait_ms}
        return fn

    return deco
