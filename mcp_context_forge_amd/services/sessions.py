"""Transport session registry + resumable event store.

Reference analogs: cache/session_registry.py (memory/redis/database backends,
broadcast :1271, respond loop :1499), transports/streamablehttp_transport.py
InMemoryEventStore (:467, replay :615), transports/redis_event_store.py.

MI355X mapping: the Redis backend's cross-worker role is
replaced by rank-ownership — each session is owned by exactly one GPU rank
and cross-rank messages ride the RCCL control channel (parallel/). The
`database` backend is kept for durability/multi-node parity; `memory` is the
single-rank fast path.
"""

from __future__ import annotations

import asyncio
import itertools
import time
import uuid
from collections import deque
from dataclasses import dataclass, field
from typing import Any, AsyncIterator, Deque, Dict, List, Optional, Tuple


@dataclass
class SessionEvent:
    event_id: str
    message: dict


@dataclass
class Session:
    session_id: str
    transport: str = "streamablehttp"
    server_id: Optional[str] = None
    user: Optional[str] = None
    owner_rank: int = 0
    created_at: float = field(default_factory=time.monotonic)
    last_accessed: float = field(default_factory=time.monotonic)
    queue: "asyncio.Queue[dict]" = field(default_factory=asyncio.Queue)
    initialized: bool = False
    protocol_version: Optional[str] = None


class EventStore:
    """Per-session resumable event log with Last-Event-ID replay
    (reference: InMemoryEventStore.replay_events_after, streamablehttp_transport.py:615)."""

    def __init__(self, max_events_per_session: int = 512):
        self.max_events = max_events_per_session
        self._events: Dict[str, Deque[SessionEvent]] = {}
        self._counter = itertools.count(1)

    def store(self, session_id: str, message: dict) -> str:
        eid = f"{session_id}-{next(self._counter)}"
        dq = self._events.setdefault(session_id, deque(maxlen=self.max_events))
        dq.append(SessionEvent(eid, message))
        return eid

    def replay_after(self, session_id: str, last_event_id: Optional[str]) -> List[SessionEvent]:
        dq = self._events.get(session_id)
        if not dq:
            return []
        if last_event_id is None:
            return list(dq)
        out: List[SessionEvent] = []
        seen = False
        for ev in dq:
            if seen:
                out.append(ev)
            elif ev.event_id == last_event_id:
                seen = True
        return out if seen else list(dq)

    def drop(self, session_id: str) -> None:
        self._events.pop(session_id, None)


class SessionRegistry:
    """Session registry: memory backend with optional DB durability
    (reference: cache/session_registry.py:105-215 memory/database backends;
    the redis backend's cross-worker role is rank-ownership here).

    With a `db` attached, create/remove/touch mirror into mcp_sessions so a
    restarted gateway can tell a resumed session id (client may replay with
    Last-Event-ID) from an unknown one."""

    def __init__(self, ttl_s: float = 3600.0, rank: int = 0, event_store: Optional[EventStore] = None,
                 world_size: int = 1, db=None):
        self.ttl_s = ttl_s
        self.rank = rank
        self.world_size = max(1, world_size)
        self._sessions: Dict[str, Session] = {}
        self.event_store = event_store or EventStore()
        self.db = db
        self.resumable: Dict[str, dict] = {}
        if db is not None:
            self._load_persisted()

    def _load_persisted(self) -> None:
        import datetime as _dt

        from sqlalchemy import delete, select

        from ..db.models import DbSessionRecord, utcnow

        cutoff = utcnow() - _dt.timedelta(seconds=self.ttl_s)
        with self.db.session() as s:
            s.execute(delete(DbSessionRecord).where(DbSessionRecord.last_accessed < cutoff))
            for r in s.execute(select(DbSessionRecord)).scalars():
                self.resumable[r.session_id] = {"transport": r.transport,
                                                "server_id": r.server_id, "user": r.user_email}

    def _persist(self, sess: "Session") -> None:
        if self.db is None:
            return
        from ..db.models import DbSessionRecord

        with self.db.session() as s:
            s.merge(DbSessionRecord(session_id=sess.session_id, transport=sess.transport,
                                    owner_rank=sess.owner_rank, server_id=sess.server_id,
                                    user_email=sess.user))

    def _unpersist(self, session_id: str) -> None:
        if self.db is None:
            return
        from sqlalchemy import delete

        from ..db.models import DbSessionRecord

        with self.db.session() as s:
            s.execute(delete(DbSessionRecord).where(DbSessionRecord.session_id == session_id))

    def create(self, transport: str = "streamablehttp", server_id: Optional[str] = None,
               user: Optional[str] = None, session_id: Optional[str] = None) -> Session:
        sid = session_id
        if sid is None:
            # multi-rank: pick an id whose stable hash maps to THIS rank, so
            # any rank can route a session to its owner without shared state
            # (reference analog: session_affinity register_session_owner :664)
            from ..parallel.bus import stable_hash

            while True:
                sid = uuid.uuid4().hex
                if self.world_size == 1 or stable_hash(sid) % self.world_size == self.rank:
                    break
        sess = Session(session_id=sid, transport=transport, server_id=server_id, user=user, owner_rank=self.rank)
        self._sessions[sid] = sess
        self._persist(sess)
        return sess

    def get(self, session_id: str) -> Optional[Session]:
        sess = self._sessions.get(session_id)
        if sess:
            sess.last_accessed = time.monotonic()
        return sess

    def remove(self, session_id: str) -> None:
        self._sessions.pop(session_id, None)
        self.resumable.pop(session_id, None)
        self.event_store.drop(session_id)
        self._unpersist(session_id)

    def count(self) -> int:
        return len(self._sessions)

    def session_ids(self) -> list:
        """Public snapshot of live session ids (stable across backends —
        callers must not reach into the internal map)."""
        return list(self._sessions.keys())

    async def broadcast_all(self, message: dict) -> int:
        """Deliver a message to every live session; returns delivery count
        (reference: notification fan-out to all listeners)."""
        n = 0
        for sid in self.session_ids():
            if await self.broadcast(sid, message):
                n += 1
        return n

    async def broadcast(self, session_id: str, message: dict) -> bool:
        """Deliver a message to the transport holding `session_id`
        (reference: session_registry.broadcast :1271)."""
        sess = self._sessions.get(session_id)
        if sess is None:
            return False
        self.event_store.store(session_id, message)
        await sess.queue.put(message)
        return True

    async def respond_stream(self, session_id: str, keepalive_s: float = 30.0) -> AsyncIterator[Tuple[Optional[str], dict]]:
        """Yield (event_id, message) for an SSE/streamable GET consumer
        (reference: respond loop session_registry.py:1499)."""
        sess = self._sessions.get(session_id)
        if sess is None:
            return
        while True:
            try:
                message = await asyncio.wait_for(sess.queue.get(), timeout=keepalive_s)
            except asyncio.TimeoutError:
                yield None, {"type": "keepalive"}
                continue
            if message.get("__close__"):
                return
            yield message.get("__event_id__"), message

    def cleanup_expired(self) -> int:
        """Stale-session reaper (reference: session_registry.py:1879-2034)."""
        now = time.monotonic()
        stale = [sid for sid, s in self._sessions.items() if now - s.last_accessed > self.ttl_s]
        for sid in stale:
            self.remove(sid)
        return len(stale)

    def resume(self, session_id: str, transport: str = "streamablehttp") -> Optional["Session"]:
        """Re-materialize a persisted session after a restart (reference:
        database session backend — the client resumes with its old id and
        replays from Last-Event-ID; server-side queue starts fresh)."""
        meta = self.resumable.pop(session_id, None)
        if meta is None:
            return None
        sess = Session(session_id=session_id, transport=meta.get("transport") or transport,
                       server_id=meta.get("server_id"), user=meta.get("user"),
                       owner_rank=self.rank)
        sess.initialized = True
        self._sessions[session_id] = sess
        self._persist(sess)
        return sess


class CancellationService:
    """In-flight request registry + MCP `notifications/cancelled` handling
    (reference: routers/cancellation_router.py + session cancel service).

    tools/call requests arriving on a session are wrapped in a task and
    registered under (session_id, request_id); a cancellation notification
    cancels the task and the request produces no response (MCP semantics).
    """

    def __init__(self):
        self._inflight: Dict[Tuple[str, str], "asyncio.Task"] = {}
        self.cancelled = 0

    @staticmethod
    def _key(session_id: str, rid: Any) -> Tuple[str, str]:
        return (session_id, str(rid))

    def register(self, session_id: str, rid: Any, task: "asyncio.Task") -> None:
        self._inflight[self._key(session_id, rid)] = task

    def unregister(self, session_id: str, rid: Any) -> None:
        self._inflight.pop(self._key(session_id, rid), None)

    def cancel(self, session_id: str, rid: Any) -> bool:
        task = self._inflight.pop(self._key(session_id, rid), None)
        if task is None or task.done():
            return False
        task.cancel()
        self.cancelled += 1
        return True

    def inflight_count(self) -> int:
        return len(self._inflight)


class ElicitationService:
    """Server→client mid-call input requests (reference: session elicitation
    service; MCP `elicitation/create`). The request rides the session's
    message stream; the client's JSON-RPC *response* resolves the future."""

    def __init__(self, sessions: "SessionRegistry"):
        self.sessions = sessions
        self._pending: Dict[str, "asyncio.Future"] = {}
        self._counter = itertools.count(1)

    async def elicit(self, session_id: str, message: str, requested_schema: Optional[dict] = None,
                     timeout_s: float = 60.0) -> Any:
        eid = f"elicit-{next(self._counter)}"
        fut: "asyncio.Future" = asyncio.get_running_loop().create_future()
        self._pending[eid] = fut
        ok = await self.sessions.broadcast(session_id, {
            "jsonrpc": "2.0", "id": eid, "method": "elicitation/create",
            "params": {"message": message,
                       "requestedSchema": requested_schema or {"type": "object"}}})
        if not ok:
            self._pending.pop(eid, None)
            raise RuntimeError(f"no live session {session_id!r} for elicitation")
        try:
            return await asyncio.wait_for(fut, timeout_s)
        finally:
            self._pending.pop(eid, None)

    def resolve(self, eid: Any, result: Any, error: Any = None) -> bool:
        fut = self._pending.get(str(eid))
        if fut is None or fut.done():
            return False
        if error is not None:
            fut.set_exception(RuntimeError(str(error)))
        else:
            fut.set_result(result)
        return True

    def pending_count(self) -> int:
        return len(self._pending)
