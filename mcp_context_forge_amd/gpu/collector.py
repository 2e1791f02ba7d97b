"""Adaptive micro-batch collector for interactive transports.

SURVEY.md §7 "hard parts": batching an interactive RPC path without killing
p50. Requests submitted from concurrent HTTP handlers are coalesced into one
GPU batch when either `max_batch` requests are pending or `window_us` has
elapsed since the first pending request — under load the window never waits;
at low QPS a lone request pays at most the window.
"""

from __future__ import annotations

import asyncio
from typing import Awaitable, Callable, List, Optional


class BatchCollector:
    def __init__(self, process: Callable[[List[bytes]], Awaitable[List[Optional[bytes]]]],
                 max_batch: int = 8192, window_us: int = 500):
        self.process = process
        self.max_batch = max_batch
        self.window_s = window_us / 1e6
        self._pending: List[tuple[bytes, asyncio.Future]] = []
        self._flush_task: Optional[asyncio.Task] = None
        self._lock = asyncio.Lock()

    async def submit(self, raw: bytes) -> Optional[bytes]:
        loop = asyncio.get_running_loop()
        fut: asyncio.Future = loop.create_future()
        async with self._lock:
            self._pending.append((raw, fut))
            if len(self._pending) >= self.max_batch:
                batch = self._take()
                asyncio.create_task(self._run(batch))
            elif self._flush_task is None or self._flush_task.done():
                self._flush_task = asyncio.create_task(self._delayed_flush())
        return await fut

    def _take(self):
        batch = self._pending
        self._pending = []
        return batch

    async def _delayed_flush(self) -> None:
        await asyncio.sleep(self.window_s)
        async with self._lock:
            batch = self._take()
        if batch:
            await self._run(batch)

    async def _run(self, batch) -> None:
        raws = [raw for raw, _ in batch]
        try:
            outs = await self.process(raws)
        except Exception as exc:
            for _, fut in batch:
                if not fut.done():
                    fut.set_exception(exc)
            return
        for (_, fut), out in zip(batch, outs):
            if not fut.done():
                fut.set_result(out)
