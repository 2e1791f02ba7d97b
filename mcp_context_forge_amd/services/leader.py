"""Leader election for background singletons.

Reference analog: gateway_service.py:1254-1261 (Redis SET NX health-check
leader), heartbeat :5231, follower TTL re-election :5272. Two regimes here:

* **Collective world (torchrun, world>1):** rank 0 is leader by
  construction. This is deliberate, not a gap: the ranks are collectively
  coupled (RCCL/gloo collectives + the bus pump), so a dead rank 0 stalls
  the next collective on every rank and the job restarts as a unit —
  TTL takeover inside a dead world has nothing to take over.

* **Shared-DB multi-process (N gateway processes, one SQLite/Postgres,
  no collectives):** the reference's Redis lease maps to a DB row lease —
  `DbLeaderElector` below. Acquire = atomic claim of an expired/own row;
  heartbeat renews at ttl/3; a crashed leader's lease expires and any
  follower takes over on its next tick (the reference's :5272 semantics).
"""

from __future__ import annotations

import asyncio
import logging
import os
import socket
import time
import uuid
from typing import Optional

from sqlalchemy import text

logger = logging.getLogger(__name__)


class DbLeaderElector:
    def __init__(self, db, name: str = "gateway-singletons", ttl_s: float = 15.0,
                 holder_id: Optional[str] = None):
        self.db = db
        self.name = name
        self.ttl_s = ttl_s
        self.holder_id = holder_id or f"{socket.gethostname()}:{os.getpid()}:{uuid.uuid4().hex[:6]}"
        self._leader = False
        self._task: Optional[asyncio.Task] = None
        self._stop: Optional[asyncio.Event] = None
        self.acquisitions = 0

    @property
    def is_leader(self) -> bool:
        return self._leader

    def try_acquire(self) -> bool:
        """One election round: claim the lease if free/expired/ours."""
        now = time.time()
        try:
            with self.db.session() as s:
                row = s.execute(text("SELECT holder, expires_at FROM leader_leases WHERE name = :n"),
                                {"n": self.name}).first()
                if row is None:
                    s.execute(text("INSERT INTO leader_leases (name, holder, expires_at) "
                                   "VALUES (:n, :h, :e)"),
                              {"n": self.name, "h": self.holder_id, "e": now + self.ttl_s})
                    won = True
                elif row[0] == self.holder_id or (row[1] or 0) < now:
                    res = s.execute(text(
                        "UPDATE leader_leases SET holder = :h, expires_at = :e "
                        "WHERE name = :n AND (holder = :h OR expires_at < :now)"),
                        {"n": self.name, "h": self.holder_id, "e": now + self.ttl_s, "now": now})
                    won = res.rowcount > 0
                else:
                    won = False
        except Exception:  # racing insert → lost this round
            won = False
        if won and not self._leader:
            self.acquisitions += 1
            logger.info("leader lease %r acquired by %s", self.name, self.holder_id)
        if not won and self._leader:
            logger.warning("leader lease %r LOST by %s", self.name, self.holder_id)
        self._leader = won
        return won

    def release(self) -> None:
        if not self._leader:
            return
        try:
            with self.db.session() as s:
                s.execute(text("DELETE FROM leader_leases WHERE name = :n AND holder = :h"),
                          {"n": self.name, "h": self.holder_id})
        except Exception:  # pragma: no cover
            pass
        self._leader = False

    async def start(self) -> None:
        """Heartbeat loop: renew at ttl/3 (reference heartbeat :5231)."""
        if self._task is not None:
            return
        self._stop = asyncio.Event()
        self.try_acquire()

        async def loop() -> None:
            while not self._stop.is_set():
                try:
                    await asyncio.wait_for(self._stop.wait(), timeout=self.ttl_s / 3.0)
                    return
                except asyncio.TimeoutError:
                    pass
                try:
                    self.try_acquire()
                except Exception:  # pragma: no cover
                    logger.exception("leader heartbeat error")

        self._task = asyncio.create_task(loop())

    async def stop(self) -> None:
        if self._stop is not None:
            self._stop.set()
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
            self._task = None
        self.release()
