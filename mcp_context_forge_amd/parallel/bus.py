"""Cross-rank request/response bus over collectives.

Reference analog: services/session_affinity.py — gunicorn workers forward
requests to the worker owning an MCP session over Redis pub/sub
(forward_request_to_owner :747, rpc listener :893, executor :950). Here the
same role is played by a fixed-cadence **exchange pump**: every rank joins
an `all_to_all` round every `cadence_us`, shipping pending forwards and
returning responses. On GPUs the rounds ride RCCL over xGMI; on CPU (tests)
the same code runs over gloo. Collectives require symmetric participation,
so the pump runs even when idle (empty rounds are a ~50 µs all_to_all of
zero-length buckets).

Session→rank ownership mirrors the reference's session registry ownership:
`owner_rank = stable_hash(session_id) % world` by default, or explicit
claims recorded in the owner map (reference: register_session_owner :664).
"""

from __future__ import annotations

import asyncio
import itertools
import pickle
import threading
import time
from typing import Any, Awaitable, Callable, Dict, List, Optional, Tuple

from . import collectives


def stable_hash(s: str) -> int:
    h = 1469598103934665603
    for b in s.encode():
        h = ((h ^ b) * 1099511628211) & ((1 << 63) - 1)
    return h


class RcclBus:
    """Request/response forwarding between gateway ranks.

    handler(payload) -> response payload, executed on the destination rank.
    `submit(dest_rank, payload)` returns a future resolved when the response
    round-trips. The pump thread drives the collective rounds; the asyncio
    side only touches thread-safe queues.
    """

    def __init__(self, handler: Callable[[Any], Awaitable[Any]],
                 loop: Optional[asyncio.AbstractEventLoop] = None,
                 cadence_us: int = 1000):
        self.rank, self.world = collectives.rank_world()
        self.group = collectives.new_bus_group() if self.world > 1 else None
        self.handler = handler
        self.loop = loop
        self.cadence_s = cadence_us / 1e6
        self._pending: List[Tuple[int, int, Any]] = []   # (dest, msg_id, payload)
        self._futures: Dict[int, asyncio.Future] = {}
        self._ids = itertools.count(1)
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.rounds = 0
        self.forwarded = 0
        self.handled = 0
        self._publish_q: List[Any] = []
        # broadcast subscriber (registry/plugin invalidation channel —
        # reference: signed Redis pub/sub, plugins/__init__.py:46-100)
        self.on_publish: Optional[Callable[[int, Any], None]] = None

    # -- public API (asyncio side) ------------------------------------------
    def owner_of(self, session_id: str) -> int:
        return stable_hash(session_id) % self.world

    def publish(self, payload: Any) -> None:
        """Deliver `payload` to every OTHER rank's on_publish next round."""
        with self._lock:
            self._publish_q.append(payload)

    async def submit(self, dest: int, payload: Any) -> Any:
        """Forward `payload` to `dest`; resolves with the handler's response."""
        if dest == self.rank:
            return await self.handler(payload)
        loop = asyncio.get_running_loop()
        fut: asyncio.Future = loop.create_future()
        msg_id = next(self._ids)
        with self._lock:
            self._futures[msg_id] = fut
            self._pending.append((dest, msg_id, payload))
        return await fut

    def start(self) -> None:
        if self.world <= 1 or self._thread is not None:
            return
        if self.loop is None:
            self.loop = asyncio.get_event_loop()
        self._thread = threading.Thread(target=self._pump, name="rccl-bus", daemon=True)
        self._thread.start()

    def stop(self) -> None:
        """Coordinated shutdown: the pump keeps joining rounds (so peers are
        never left blocking in a collective) until EVERY rank has signaled
        stop in the same round; then all exit together."""
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=30)
            self._thread = None

    # -- pump (collective side; its own thread, symmetric across ranks) ------
    def _pump(self) -> None:
        while True:
            time.sleep(self.cadence_s)
            all_stopping = self._round()
            if all_stopping:
                return

    def _round(self) -> bool:
        # 1) ship pending requests + queued responses (+ stop signal to all)
        with self._lock:
            outgoing, self._pending = self._pending, []
        stopping = self._stop.is_set()
        with self._lock:
            pubs, self._publish_q = self._publish_q, []
        buckets: List[List[Tuple[str, int, int, Any]]] = [[] for _ in range(self.world)]
        for d in range(self.world):
            buckets[d].append(("ctl", self.rank, 0, {"stopping": stopping}))
            if d != self.rank:
                for pub in pubs:
                    buckets[d].append(("pub", self.rank, 0, pub))
        for dest, msg_id, payload in outgoing:
            buckets[dest].append(("req", self.rank, msg_id, payload))
        for dest, msg_id, payload in self._take_responses():
            buckets[dest].append(("resp", self.rank, msg_id, payload))
        received = collectives.all_to_all_objects(buckets, group=self.group)
        self.rounds += 1

        n_stopping = 0
        for src_list in received:
            for kind, src, msg_id, payload in src_list:
                if kind == "ctl":
                    if payload.get("stopping"):
                        n_stopping += 1
                elif kind == "pub":
                    if self.on_publish is not None and self.loop is not None:
                        self.loop.call_soon_threadsafe(self.on_publish, src, payload)
                elif kind == "req":
                    self.forwarded += 1
                    self._execute(src, msg_id, payload)
                else:
                    fut = self._futures.pop(msg_id, None)
                    if fut is not None and self.loop is not None:
                        self.loop.call_soon_threadsafe(
                            lambda f=fut, p=payload: (not f.done()) and f.set_result(p))
        return n_stopping == self.world and not outgoing

    _responses: List[Tuple[int, int, Any]] = None  # (dest, msg_id, payload)

    def _take_responses(self) -> List[Tuple[int, int, Any]]:
        if self._responses is None:
            self._responses = []
        with self._lock:
            out, self._responses = self._responses, []
        return out

    def _execute(self, src: int, msg_id: int, payload: Any) -> None:
        """Run the handler on the asyncio loop; queue the response for the
        next round (reference: _execute_forwarded_request :950)."""
        if self.loop is None:
            return

        def done_cb(fut: "asyncio.Future") -> None:
            try:
                result = fut.result()
            except Exception as exc:  # pragma: no cover - handler errors ride back
                result = {"__bus_error__": str(exc)}
            with self._lock:
                if self._responses is None:
                    self._responses = []
                self._responses.append((src, msg_id, result))
            self.handled += 1

        def schedule() -> None:
            task = asyncio.ensure_future(self.handler(payload))
            task.add_done_callback(done_cb)

        self.loop.call_soon_threadsafe(schedule)

    def stats(self) -> dict:
        return {"rank": self.rank, "world": self.world, "rounds": self.rounds,
                "forwarded": self.forwarded, "handled": self.handled,
                "pending": len(self._pending)}
