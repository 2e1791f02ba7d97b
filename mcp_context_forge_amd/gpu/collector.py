"""Adaptive micro-batch collector for interactive transports.

SURVEY.md §7 "hard parts": batching an interactive RPC path without killing
p50. Serial-consumer design: one worker loop waits for the first pending
request, lingers `window_us`, then drains EVERYTHING pending into one batch
and processes it; requests arriving while a batch is in flight accumulate
for the next batch (natural backpressure → batches grow with load, latency
stays ≈ window + one batch time at low load).
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
        self._wakeup: Optional[asyncio.Event] = None
        self._worker: Optional[asyncio.Task] = None
        self.batches = 0
        self.max_seen = 0

    def _ensure_worker(self) -> None:
        if self._worker is None or self._worker.done():
            self._wakeup = asyncio.Event()
            self._worker = asyncio.create_task(self._run_loop())

    async def submit(self, raw: bytes, user: Optional[str] = None) -> Optional[bytes]:
        loop = asyncio.get_running_loop()
        fut: asyncio.Future = loop.create_future()
        self._ensure_worker()
        self._pending.append((raw, fut, user))
        self._wakeup.set()
        return await fut

    async def submit_many(self, raws: List[bytes],
                          users: Optional[List[Optional[str]]] = None) -> List[Optional[bytes]]:
        """Submit a pre-formed group (e.g. one edge frame) with ONE future for
        the whole group — the owner loop pays O(frames), not O(requests)."""
        if not raws:
            return []
        loop = asyncio.get_running_loop()
        fut: asyncio.Future = loop.create_future()
        self._ensure_worker()
        self._pending.append((raws, fut, users))
        self._wakeup.set()
        return await fut

    async def _run_loop(self) -> None:
        while True:
            if not self._pending:
                self._wakeup.clear()
                await self._wakeup.wait()
            # linger so concurrent submitters can pile on
            if self.window_s > 0:
                await asyncio.sleep(self.window_s)
            batch, self._pending = self._pending[: self.max_batch], self._pending[self.max_batch:]
            if not batch:
                continue
            self.batches += 1
            # flatten: an entry is a single bytes or a list (one frame);
            # per-request user identity rides along so plugin context,
            # metrics attribution and cache tenancy survive micro-batching
            raws: List[bytes] = []
            users: List[Optional[str]] = []
            spans: List[tuple] = []
            for item, fut, u in batch:
                if isinstance(item, list):
                    spans.append((len(raws), len(item), fut, True))
                    raws.extend(item)
                    users.extend(u if isinstance(u, list) else [u] * len(item))
                else:
                    spans.append((len(raws), 1, fut, False))
                    raws.append(item)
                    users.append(u)
            self.max_seen = max(self.max_seen, len(raws))
            try:
                outs = await self.process(raws, users=users)
            except Exception as exc:
                for _, _n, fut, _f in spans:
                    if not fut.done():
                        fut.set_exception(exc)
                continue
            for start, n, fut, is_frame in spans:
                if not fut.done():
                    fut.set_result(outs[start:start + n] if is_frame else outs[start])
