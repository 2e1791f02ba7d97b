"""NativeEdge — Python side of the C++ epoll HTTP edge (ops/csrc/edge.cpp).

Reference analog: the Rust mcp_runtime sidecar's role split (README.md:15-16
"Python stays authoritative for auth/RBAC"): the C++ threads own sockets,
HTTP parsing, response framing and the hot auth memo; THIS loop is the
Python authority — it authenticates cache misses, runs the engine's batched
GPU pipeline over each drained batch, and serves control-plane (non-/rpc)
requests through the regular ASGI app.

One asyncio task pulls batches with edge.poll (GIL released while waiting),
processes them, and completes them back to C++; while Python is busy with
batch N the C++ side accumulates batch N+1 — the micro-batch window emerges
from load instead of a timer, same serial-consumer shape as
gpu/collector.py.
"""

from __future__ import annotations

import asyncio
import importlib.util
import logging
import struct
import time
from typing import Dict, List, Optional, Tuple

logger = logging.getLogger(__name__)

_mod = None


def get_edge_module():
    global _mod
    if _mod is None:
        from ..ops.build import EDGE, build_edge

        build_edge(verbose=False)
        spec = importlib.util.spec_from_file_location("forge_edge", EDGE)
        _mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(_mod)
    return _mod


# streaming endpoints cannot ride the buffered cold path — clients get a
# redirect to the control-plane port (uvicorn) which streams natively
_STREAM_GET_SUFFIXES = ("/mcp", "/sse", "/events")


class NativeEdge:
    def __init__(self, engine, app=None, auth=None, port: int = 4446,
                 threads: int = 4, control_url: Optional[str] = None,
                 depth: int = 2):
        self.engine = engine
        self.app = app            # FastAPI app for the cold path (optional)
        self.auth = auth          # AuthService; falls back to app.state.auth
        self.port = port
        self.threads = threads
        self.control_url = control_url
        # batches processed concurrently: with 2 in flight, batch B's C++
        # envelope/decide host stages overlap batch A's GPU sync (the
        # pipeline's _gpu_lock serializes only the device sections)
        self.depth = max(1, depth)
        self.edge = get_edge_module()
        self.handle = 0
        self._task: Optional[asyncio.Task] = None
        self._running = False
        self._auth_epoch = -1
        # python-side auth memo for server-scoped tokens (not C++-cacheable:
        # their requests must carry server_id through a per-request path)
        self._scoped_memo: Dict[bytes, Tuple[object, float]] = {}
        self.batches = 0
        self.requests = 0
        self.max_batch = 0

    # ------------------------------------------------------------------
    async def start(self) -> None:
        s = self.engine.settings
        if self.auth is None and self.app is not None:
            self.auth = self.app.state.auth
        self.handle = self.edge.start(self.port, self.threads,
                                      s.max_request_body_bytes,
                                      bool(s.auth_required))
        self._running = True
        self._task = asyncio.create_task(self._loop())
        logger.info("native edge listening on :%d (%d threads)", self.port, self.threads)

    async def stop(self) -> None:
        self._running = False
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
            self._task = None
        if self.handle:
            await asyncio.to_thread(self.edge.stop, self.handle)
            self.handle = 0

    def stats(self) -> dict:
        out = {"batches": self.batches, "requests": self.requests, "max_batch": self.max_batch}
        if self.handle:
            out.update(self.edge.stats(self.handle))
        return out

    # ------------------------------------------------------------------
    def _check_revocations(self) -> None:
        if self.auth is None:
            return
        epoch = self.auth.revocation_epoch
        if epoch != self._auth_epoch:
            self._auth_epoch = epoch
            self._scoped_memo.clear()
            if self.handle:
                self.edge.auth_clear(self.handle)

    def _authenticate(self, authz: Optional[bytes]):
        """Resolve an Authorization header; returns AuthContext or None.
        Positive unscoped results are pushed into the C++ cache so the next
        request from this credential never reaches Python."""
        from ..auth.service import AuthError

        if self.auth is None:
            return None
        ent = self._scoped_memo.get(authz or b"")
        if ent is not None and ent[1] > time.monotonic():
            return ent[0]
        try:
            ctx = self.auth.authenticate(authz.decode() if authz else None)
        except AuthError:
            if authz and self.handle:
                self.edge.auth_put(self.handle, authz, None)  # negative (TTL'd)
            return None
        if authz and self.handle and ctx.server_id is None:
            self.edge.auth_put(self.handle, authz, ctx.user)
        else:
            if len(self._scoped_memo) > 2048:
                self._scoped_memo.clear()
            self._scoped_memo[authz or b""] = (ctx, time.monotonic() + 30.0)
        return ctx

    # ------------------------------------------------------------------
    async def _loop(self) -> None:
        s = self.engine.settings
        wait_us = 50_000
        linger_us = s.gpu_batch_window_us
        max_n = s.gpu_batch_max_requests
        sem = asyncio.Semaphore(self.depth)

        async def run_one(batch):
            try:
                await self._handle_batch(*batch)
            except Exception:
                logger.exception("native edge batch error")
            finally:
                sem.release()

        while self._running:
            try:
                await sem.acquire()
                ids, kinds, bodies, users, authzs, meta = await asyncio.to_thread(
                    self.edge.poll, self.handle, wait_us, linger_us, max_n)
                n = len(bodies)
                if n == 0:
                    sem.release()
                    continue
                self._check_revocations()
                self.batches += 1
                self.requests += n
                self.max_batch = max(self.max_batch, n)
                asyncio.ensure_future(run_one((ids, kinds, bodies, users, authzs, meta)))
            except asyncio.CancelledError:
                return
            except Exception:
                logger.exception("native edge loop error")
                sem.release()
                await asyncio.sleep(0.05)

    async def _handle_batch(self, ids: bytes, kinds: bytes, bodies, users, authzs, meta) -> None:
        n = len(bodies)
        statuses = bytearray(n * 2)
        outs: List[Optional[bytes]] = [None] * n

        hot_idx: List[int] = []
        hot_users: List[Optional[str]] = []
        scoped: List[Tuple[int, object]] = []   # (idx, ctx) server-scoped tokens
        cold: List[int] = []
        for i in range(n):
            if kinds[i] == 0:
                u = users[i]
                if u is None and authzs[i] is not None:
                    ctx = self._authenticate(authzs[i])
                    if ctx is None:
                        struct.pack_into("<H", statuses, i * 2, 401)
                        outs[i] = b'{"detail":"Not authenticated"}'
                        continue
                    if ctx.server_id is not None:
                        scoped.append((i, ctx))
                        continue
                    u = ctx.user
                hot_idx.append(i)
                hot_users.append(u)
            else:
                cold.append(i)

        if hot_idx:
            # usage accounting for the C++-authenticated hot lane (one
            # increment per request; credential granularity is the user —
            # the native cache intentionally does not carry token ids)
            for u in hot_users:
                self.engine.token_usage.record(f"user:{u}" if u else "anonymous", u)
            raws = [bodies[i] for i in hot_idx]
            results = await self.engine.process_rpc_batch(raws, users=hot_users)
            for i, r in zip(hot_idx, results):
                if r is None:
                    struct.pack_into("<H", statuses, i * 2, 202)
                    outs[i] = b""
                else:
                    struct.pack_into("<H", statuses, i * 2, 200)
                    outs[i] = r if isinstance(r, bytes) else bytes(r)

        for i, ctx in scoped:
            r = await self.engine.handle_rpc_bytes(bodies[i], user=ctx.user, server_id=ctx.server_id)
            struct.pack_into("<H", statuses, i * 2, 200 if r is not None else 202)
            outs[i] = r or b""

        # complete the hot lane first — cold requests may take arbitrary time
        if hot_idx or scoped or any(outs[i] is not None for i in range(n)):
            hot_ids = b"".join(ids[i * 8:(i + 1) * 8] for i in range(n) if outs[i] is not None)
            hot_sts = b"".join(statuses[i * 2:(i + 1) * 2] for i in range(n) if outs[i] is not None)
            hot_bodies = [outs[i] for i in range(n) if outs[i] is not None]
            await asyncio.to_thread(self.edge.complete, self.handle, hot_ids, bytes(hot_sts), hot_bodies)

        for i in cold:
            (rid,) = struct.unpack_from("<Q", ids, i * 8)
            asyncio.ensure_future(self._cold_one(rid, meta[i], bodies[i], authzs[i]))

    # ------------------------------------------------------------------
    async def _cold_one(self, rid: int, m, body: bytes, authz: Optional[bytes]) -> None:
        """Serve a control-plane request through the ASGI app and write the
        raw HTTP response back on the native connection."""
        try:
            raw = await self._run_asgi(m, body, authz)
        except Exception:
            logger.exception("cold path error")
            payload = b'{"detail":"internal error"}'
            raw = (b"HTTP/1.1 500 Internal Server Error\r\nContent-Type: application/json\r\n"
                   + b"Content-Length: " + str(len(payload)).encode() + b"\r\n\r\n" + payload)
        await asyncio.to_thread(self.edge.complete_raw, self.handle, rid, raw)

    async def _run_asgi(self, m, body: bytes, authz: Optional[bytes]) -> bytes:
        method, target, headers_blob = m
        path, _, query = target.partition("?")
        if self.app is None:
            payload = b'{"detail":"control plane not mounted on the native edge"}'
            return (b"HTTP/1.1 404 Not Found\r\nContent-Type: application/json\r\nContent-Length: "
                    + str(len(payload)).encode() + b"\r\n\r\n" + payload)
        if method == "GET" and (path.endswith(_STREAM_GET_SUFFIXES) or "/sse" in path):
            # streaming endpoints live on the control-plane port
            loc = (self.control_url or "") + target
            if self.control_url:
                return (b"HTTP/1.1 307 Temporary Redirect\r\nLocation: " + loc.encode()
                        + b"\r\nContent-Length: 0\r\n\r\n")
            payload = b'{"detail":"streaming endpoints are served by the control-plane port"}'
            return (b"HTTP/1.1 501 Not Implemented\r\nContent-Type: application/json\r\nContent-Length: "
                    + str(len(payload)).encode() + b"\r\n\r\n" + payload)
        # parse header lines (skip request line) into ASGI header pairs
        headers: List[Tuple[bytes, bytes]] = []
        for line in headers_blob.split(b"\r\n")[1:]:
            k, sep, v = line.partition(b":")
            if sep:
                headers.append((k.strip().lower(), v.strip()))
        scope = {
            "type": "http", "asgi": {"version": "3.0"}, "http_version": "1.1",
            "method": method, "scheme": "http", "path": path, "raw_path": path.encode(),
            "query_string": query.encode(), "headers": headers,
            "client": ("127.0.0.1", 0), "server": ("127.0.0.1", self.port),
        }
        sent = [False]

        async def receive():
            if not sent[0]:
                sent[0] = True
                return {"type": "http.request", "body": body, "more_body": False}
            return {"type": "http.disconnect"}

        status = [500]
        resp_headers: List[Tuple[bytes, bytes]] = []
        chunks: List[bytes] = []

        async def send(msg):
            if msg["type"] == "http.response.start":
                status[0] = msg["status"]
                resp_headers.extend(msg.get("headers") or [])
            elif msg["type"] == "http.response.body":
                chunks.append(msg.get("body", b""))

        await asyncio.wait_for(self.app(scope, receive, send), timeout=60.0)
        payload = b"".join(chunks)
        head = [b"HTTP/1.1 " + str(status[0]).encode() + b" X"]
        has_len = False
        for k, v in resp_headers:
            if k.lower() == b"content-length":
                has_len = True
            if k.lower() in (b"transfer-encoding", b"connection"):
                continue
            head.append(k + b": " + v)
        if not has_len:
            head.append(b"Content-Length: " + str(len(payload)).encode())
        return b"\r\n".join(head) + b"\r\n\r\n" + payload
