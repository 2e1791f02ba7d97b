"""HTTP edge: FastAPI application.

Reference analogs: mcpgateway/main.py app assembly (:2096), middleware stack
(:3282-3474, ordering per docs/docs/architecture/middleware-ordering.md),
REST CRUD routers (:3557), JSON-RPC POST /rpc (:7896), protocol endpoints
(:3962-4082), per-server SSE pair (:4478/:4597), streamable-HTTP /mcp
(transports/streamablehttp_transport.py), WebSocket /ws (:11866),
well-known endpoints (routers/well_known.py), admin API (admin.py).

The /rpc and /mcp POST paths feed the GPU micro-batch collector when the
HIP pipeline is attached; everything else is standard control-plane work.
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid
from typing import Any, Dict, List, Optional

from fastapi import Depends, FastAPI, HTTPException, Query, Request, Response, WebSocket, WebSocketDisconnect
from fastapi.responses import HTMLResponse, JSONResponse, PlainTextResponse, StreamingResponse
from starlette.middleware.base import BaseHTTPMiddleware
from starlette.middleware.cors import CORSMiddleware
from starlette.middleware.gzip import GZipMiddleware

from ..auth.service import AuthContext, AuthError, AuthService, PermissionError_
from ..config import Settings
from ..engine import GatewayEngine
from ..gpu.collector import BatchCollector
from ..protocol import jsonrpc
from ..protocol.mcp import PROTOCOL_VERSION, initialize_result
from ..registry.registry import ConflictError, NotFoundError
from ..services.gateway_service import GatewayConnectionError
from ..services.llm_proxy import LLMProxyError
from ..utils import TokenBucket

# ---------------------------------------------------------------------------
# middleware (reference execution order, trimmed to the load-bearing set)
# ---------------------------------------------------------------------------


class SecurityHeadersMiddleware(BaseHTTPMiddleware):
    """reference: middleware/security_headers.py"""

    async def dispatch(self, request, call_next):
        resp = await call_next(request)
        resp.headers.setdefault("X-Content-Type-Options", "nosniff")
        resp.headers.setdefault("X-Frame-Options", "DENY")
        resp.headers.setdefault("Referrer-Policy", "no-referrer")
        resp.headers.setdefault("Content-Security-Policy", "default-src 'self'; script-src 'self' 'unsafe-inline'; style-src 'self' 'unsafe-inline'")
        return resp


class HeaderSizeMiddleware(BaseHTTPMiddleware):
    """reference: middleware HeaderSizeMiddleware"""

    def __init__(self, app, max_bytes: int = 16384):
        super().__init__(app)
        self.max_bytes = max_bytes

    async def dispatch(self, request, call_next):
        total = sum(len(k) + len(v) for k, v in request.headers.raw)
        if total > self.max_bytes:
            return JSONResponse({"detail": "headers too large"}, status_code=431)
        return await call_next(request)


class RateLimitMiddleware(BaseHTTPMiddleware):
    """Token-bucket per client IP (reference: middleware/rate_limit_middleware.py)."""

    def __init__(self, app, rpm: int, burst: int):
        super().__init__(app)
        self.rpm = rpm
        self.burst = burst
        self.buckets: Dict[str, TokenBucket] = {}

    async def dispatch(self, request, call_next):
        client = request.client.host if request.client else "?"
        bucket = self.buckets.get(client)
        if bucket is None:
            bucket = self.buckets.setdefault(client, TokenBucket(self.rpm, self.burst))
        if not bucket.allow():
            return JSONResponse({"detail": "rate limit exceeded"}, status_code=429)
        return await call_next(request)


class SseAwareGZipMiddleware(GZipMiddleware):
    """Compression that leaves streaming endpoints alone (reference:
    SSE-aware compression layer in the middleware stack §2.4)."""

    _STREAM_PREFIXES = ("/servers", "/mcp", "/reverse-proxy", "/ws")

    async def __call__(self, scope, receive, send):
        if scope["type"] == "http" and any(
                scope["path"].startswith(p) for p in self._STREAM_PREFIXES):
            await self.app(scope, receive, send)
            return
        await super().__call__(scope, receive, send)


class CorrelationIDMiddleware(BaseHTTPMiddleware):
    """reference: CorrelationIDMiddleware — propagate/emit X-Correlation-ID."""

    async def dispatch(self, request, call_next):
        cid = request.headers.get("x-correlation-id") or uuid.uuid4().hex[:16]
        resp = await call_next(request)
        resp.headers["X-Correlation-ID"] = cid
        return resp


class BodyLimitMiddleware(BaseHTTPMiddleware):
    """reference: ValidationMiddleware body-size guard."""

    def __init__(self, app, max_bytes: int):
        super().__init__(app)
        self.max_bytes = max_bytes

    async def dispatch(self, request, call_next):
        cl = request.headers.get("content-length")
        if cl and cl.isdigit() and int(cl) > self.max_bytes:
            return JSONResponse({"detail": "request body too large"}, status_code=413)
        return await call_next(request)


class ApiVersionAlias:
    """Versioned API facade (reference: api/v1/__init__.py:355 + the legacy
    alias shim wired at main.py:12811): every route is also reachable under
    /v1/… — one implementation surface, a stable versioned path for
    clients. GET /v1 returns the version index."""

    PREFIX = "/v1"

    def __init__(self, app):
        self.app = app

    async def __call__(self, scope, receive, send):
        if scope["type"] == "http":
            p = scope.get("path", "")
            if p in ("/v1", "/v1/"):
                body = (b'{"version":"v1","status":"stable",'
                        b'"note":"all unversioned routes are aliased under /v1"}')
                await send({"type": "http.response.start", "status": 200,
                            "headers": [(b"content-type", b"application/json"),
                                        (b"content-length", str(len(body)).encode())]})
                await send({"type": "http.response.body", "body": body})
                return
            if p.startswith("/v1/"):
                scope = dict(scope)
                scope["path"] = p[len(self.PREFIX):]
                scope["raw_path"] = scope["path"].encode()
        await self.app(scope, receive, send)


class RpcFastPath:
    """Pure-ASGI fast lane for POST /rpc — bypasses FastAPI routing/DI and
    the BaseHTTPMiddleware stack for the hot endpoint, exactly the role the
    reference gives its Rust edge runtime for /mcp (crates/mcp_runtime:
    nginx routes the hot path to the sidecar, Python keeps the rest).
    Auth contexts are memoized (reference: cache/auth_cache.py)."""

    def __init__(self, app, fastapi_app):
        self.app = app
        self.fastapi = fastapi_app
        self._auth_cache: Dict[bytes, tuple] = {}
        self._auth_epoch = -1

    def _auth(self, authz: Optional[bytes]):
        import time as _t

        auth = self.fastapi.state.auth
        epoch = auth.revocation_epoch
        if epoch != self._auth_epoch:
            # a token was revoked somewhere: drop every memoized context so
            # revoked credentials stop working immediately, not after the TTL
            self._auth_cache.clear()
            self._auth_epoch = epoch
        ent = self._auth_cache.get(authz)
        now = _t.monotonic()
        if ent is not None and ent[1] > now:
            return ent[0]
        try:
            ctx = auth.authenticate(authz.decode() if authz else None)
        except AuthError:
            return None
        if len(self._auth_cache) > 4096:
            self._auth_cache.clear()
        self._auth_cache[authz] = (ctx, now + 30.0)  # 30 s TTL (reference auth_cache)
        return ctx

    async def __call__(self, scope, receive, send):
        if scope["type"] != "http" or scope["method"] != "POST" or scope["path"] != "/rpc":
            await self.app(scope, receive, send)
            return
        headers = {k: v for k, v in scope["headers"]}
        st = self.fastapi.state
        if st.fastpath_passthrough and any(h in headers for h in st.fastpath_passthrough):
            await self.app(scope, receive, send)  # passthrough headers need the full path
            return
        cl = headers.get(b"content-length")
        if cl and cl.isdigit() and int(cl) > st.engine.settings.max_request_body_bytes:
            await self._respond(send, 413, b'{"detail":"request body too large"}')
            return
        ctx = self._auth(headers.get(b"authorization"))
        if ctx is None:
            await self._respond(send, 401, b'{"detail":"Not authenticated"}')
            return
        max_body = st.engine.settings.max_request_body_bytes
        body = b""
        while True:
            msg = await receive()
            body += msg.get("body", b"")
            if len(body) > max_body:
                # chunked uploads (no Content-Length) must hit the same guard
                await self._respond(send, 413, b'{"detail":"request body too large"}')
                return
            if not msg.get("more_body", False):
                break
        st.engine.token_usage.record(ctx.credential or ctx.auth_method, ctx.user)
        collector = st.collector
        if collector is not None and ctx.server_id is None:
            out = await collector.submit(body, user=ctx.user)
        else:
            out = await st.engine.handle_rpc_bytes(body, user=ctx.user, server_id=ctx.server_id)
        if out is None:
            await self._respond(send, 202, b"")
        else:
            await self._respond(send, 200, out)

    @staticmethod
    async def _respond(send, status: int, body: bytes):
        await send({"type": "http.response.start", "status": status,
                    "headers": [(b"content-type", b"application/json"),
                                (b"content-length", str(len(body)).encode())]})
        await send({"type": "http.response.body", "body": body})


# ---------------------------------------------------------------------------


def build_app(engine: GatewayEngine, auth: Optional[AuthService] = None) -> FastAPI:
    settings = engine.settings
    auth = auth or AuthService(engine.db, settings, token_blocklist=engine.token_blocklist)
    auth.bootstrap_admin()

    from contextlib import asynccontextmanager

    @asynccontextmanager
    async def lifespan(app_: FastAPI):
        # reference: main.py lifespan (:1443) — service startup, GPU attach
        await engine.startup()
        if engine.enable_gpu():
            app_.state.collector = BatchCollector(engine.process_rpc_batch,
                                                  max_batch=settings.gpu_batch_max_requests,
                                                  window_us=settings.gpu_batch_window_us)
        owner_server = None
        if settings.edge_socket:
            # owner side of the multi-worker edge (transports/edge.py)
            from .edge import GpuOwnerServer

            owner_server = GpuOwnerServer(engine, app_.state.collector, settings.edge_socket)
            await owner_server.start()
            app_.state.owner_server = owner_server
        native_edge = None
        if settings.native_edge_port:
            # C++ epoll front door: hot /rpc batches + cold control-plane
            # passthrough to this app (transports/native_edge.py)
            import os as _os

            from .native_edge import NativeEdge

            threads = settings.native_edge_threads or min(8, max(2, (_os.cpu_count() or 8) // 2))
            native_edge = NativeEdge(engine, app=app_, auth=auth,
                                     port=settings.native_edge_port, threads=threads,
                                     control_url=f"http://{settings.host}:{settings.port}")
            await native_edge.start()
            app_.state.native_edge = native_edge
        yield
        if native_edge is not None:
            await native_edge.stop()
        if owner_server is not None:
            await owner_server.stop()
        await engine.shutdown()

    app = FastAPI(title=settings.app_name, version=settings.version, lifespan=lifespan,
                  docs_url="/docs" if settings.docs_enabled else None, redoc_url=None)
    app.state.engine = engine
    app.state.auth = auth
    app.state.collector = None
    app.state.started = time.time()

    # middleware registration — innermost added first (starlette wraps outward);
    # effective order matches reference §2.4 (CORS outermost → ... → handlers)
    app.add_middleware(SseAwareGZipMiddleware, minimum_size=1024)
    app.add_middleware(CorrelationIDMiddleware)
    app.add_middleware(BodyLimitMiddleware, max_bytes=settings.max_request_body_bytes)
    if settings.rate_limit_enabled:
        app.add_middleware(RateLimitMiddleware, rpm=settings.rate_limit_requests_per_minute,
                           burst=settings.rate_limit_burst)
    app.add_middleware(HeaderSizeMiddleware, max_bytes=settings.max_header_bytes)
    if settings.security_headers_enabled:
        app.add_middleware(SecurityHeadersMiddleware)
    app.add_middleware(CORSMiddleware, allow_origins=settings.cors_allow_origins,
                       allow_methods=["*"], allow_headers=["*"])
    app.state.fastpath_passthrough = [h.lower().encode() for h in settings.passthrough_headers]
    if not settings.rate_limit_enabled:
        # hot-lane /rpc (outermost; skipped when rate limiting must apply)
        app.add_middleware(RpcFastPath, fastapi_app=app)
    # /v1 alias outermost so versioned paths hit the fast lane too
    app.add_middleware(ApiVersionAlias)

    # -- auth dependency -------------------------------------------------------
    def _server_scope_ok(path: str, sid: str) -> bool:
        """Token-scoping enforcement (reference: middleware/token_scoping.py,
        1,497 LoC): a server-scoped token may only touch ITS server's
        surface — the RPC/MCP data plane (where the scope filters tool
        visibility) and that server's own endpoints. Everything else
        (registry CRUD, admin, other servers) is 403."""
        if path in ("/rpc", "/mcp", "/health", "/healthz", "/version") or \
                path.startswith("/.well-known"):
            return True
        if path.startswith(f"/servers/{sid}/") or path == f"/servers/{sid}":
            return True
        return False

    async def get_auth(request: Request) -> AuthContext:
        try:
            ctx = auth.authenticate(request.headers.get("authorization"))
        except AuthError as exc:
            raise HTTPException(status_code=exc.status, detail=str(exc),
                                headers={"WWW-Authenticate": "Bearer"}) from exc
        if ctx.server_id is not None and not _server_scope_ok(request.url.path, ctx.server_id):
            raise HTTPException(status_code=403,
                                detail=f"token is scoped to server {ctx.server_id}")
        engine.token_usage.record(ctx.credential or ctx.auth_method, ctx.user)
        return ctx

    def require(permission: str):
        async def dep(request: Request, ctx: AuthContext = Depends(get_auth)) -> AuthContext:
            try:
                auth.require_permission(ctx, permission)
            except PermissionError_ as exc:
                raise HTTPException(status_code=403, detail=str(exc)) from exc
            return ctx

        return dep

    async def rpc_bytes(raw: bytes, ctx: AuthContext, server_id: Optional[str] = None) -> Optional[bytes]:
        sid = server_id or ctx.server_id  # token-scoped server (reference: token_scoping.py)
        if app.state.collector is not None and server_id is None and ctx.server_id is None:
            return await app.state.collector.submit(raw, user=ctx.user)
        return await engine.handle_rpc_bytes(raw, user=ctx.user, server_id=sid)

    # -- health / version / metrics (reference: main.py /health /version, prometheus) --
    @app.get("/health")
    async def health():
        return engine.health()

    @app.get("/healthz")
    async def healthz():
        return {"status": "ok"}

    @app.get("/ready")
    async def ready():
        return {"ready": True}

    @app.get("/version")
    async def version(ctx: AuthContext = Depends(get_auth)):
        return engine.version_info()

    @app.get("/metrics")
    async def metrics():
        snap = engine.metrics.snapshot()
        lines = ["# TYPE mcpgateway_tool_invocations_total counter"]
        for k, v in snap["counters"].items():
            lines.append(f"mcpgateway_{k} {v}")
        if engine.gpu_pipeline is not None:
            st = engine.gpu_pipeline.stats()
            for k in ("requests", "fast_path", "slow_path", "blocked", "cache_hits"):
                lines.append(f"mcpgateway_gpu_{k} {st[k]}")
        return PlainTextResponse("\n".join(lines) + "\n")

    # -- protocol endpoints (reference: main.py:3962-4082) -----------------------
    @app.post("/protocol/initialize")
    async def protocol_initialize(request: Request, ctx: AuthContext = Depends(get_auth)):
        body = await request.json()
        return initialize_result(body.get("protocolVersion") if isinstance(body, dict) else None)

    @app.post("/protocol/ping")
    async def protocol_ping(request: Request, ctx: AuthContext = Depends(get_auth)):
        body = await request.json()
        rid = body.get("id") if isinstance(body, dict) else None
        return {"jsonrpc": "2.0", "id": rid, "result": {}}

    # -- JSON-RPC endpoint (THE hot path; reference: main.py:7896) ----------------
    passthrough = [h.lower() for h in settings.passthrough_headers]

    @app.post("/rpc")
    async def handle_rpc(request: Request, ctx: AuthContext = Depends(get_auth)):
        raw = await request.body()
        # passthrough headers ride to upstreams (reference: passthrough_headers.py)
        fwd = {h: request.headers[h] for h in passthrough if h in request.headers}
        if fwd and (app.state.collector is None or ctx.server_id is not None):
            out = await engine.handle_rpc_bytes(raw, user=ctx.user, server_id=ctx.server_id, headers=fwd)
        else:
            out = await rpc_bytes(raw, ctx)
        if out is None:
            return Response(status_code=202)
        return Response(content=out, media_type="application/json")

    # -- streamable HTTP /mcp (reference: transports/streamablehttp_transport.py) --
    async def mcp_post(request: Request, ctx: AuthContext, server_id: Optional[str] = None):
        # protocol-version negotiation guard (reference:
        # middleware/protocol_version.py MCPProtocolVersionMiddleware):
        # a client-declared MCP-Protocol-Version we do not speak is a 400
        from ..protocol.mcp import SUPPORTED_PROTOCOL_VERSIONS

        pv = request.headers.get("mcp-protocol-version")
        if pv and pv not in SUPPORTED_PROTOCOL_VERSIONS:
            return JSONResponse(
                {"detail": f"unsupported MCP protocol version {pv!r}; "
                           f"supported: {list(SUPPORTED_PROTOCOL_VERSIONS)}"},
                status_code=400)
        raw = await request.body()
        session_id = request.headers.get("mcp-session-id")
        sess = engine.sessions.get(session_id) if session_id else None
        headers = {}
        try:
            req = jsonrpc.parse_request_bytes(raw)
        except jsonrpc.JSONRPCError as exc:
            return Response(content=jsonrpc.JSONRPCResponse(id=None, error=exc).to_bytes(),
                            media_type="application/json")
        if req.method == "initialize":
            sess = engine.sessions.create(transport="streamablehttp", server_id=server_id, user=ctx.user)
            headers["mcp-session-id"] = sess.session_id
        elif sess is None and session_id:
            # restart continuity: a DB-persisted session resumes in place
            # (reference: database session backend; client replays from
            # Last-Event-ID on its GET stream)
            sess = engine.sessions.resume(session_id)
            if sess is None:
                if engine.bus is not None and engine.forward_rpc is not None:
                    owner = engine.bus.owner_of(session_id)
                    if owner != engine.rank:
                        out = await engine.forward_rpc(owner, raw, user=ctx.user, server_id=server_id)
                        if out is None:
                            return Response(status_code=202)
                        return Response(content=out, media_type="application/json")
                return JSONResponse({"detail": "session not found"}, status_code=404)
        resp = await engine.handle_rpc(req, user=ctx.user, server_id=server_id or ctx.server_id, session=sess)
        if resp is None:
            return Response(status_code=202, headers=headers)
        return Response(content=resp.to_bytes(), media_type="application/json", headers=headers)

    @app.post("/mcp")
    async def mcp_endpoint(request: Request, ctx: AuthContext = Depends(get_auth)):
        return await mcp_post(request, ctx)

    @app.post("/servers/{server_id}/mcp")
    async def mcp_server_endpoint(server_id: str, request: Request, ctx: AuthContext = Depends(get_auth)):
        return await mcp_post(request, ctx, server_id=server_id)

    @app.get("/mcp")
    async def mcp_get(request: Request, ctx: AuthContext = Depends(get_auth)):
        session_id = request.headers.get("mcp-session-id")
        sess = engine.sessions.get(session_id) if session_id else None
        if sess is None:
            return JSONResponse({"detail": "session not found"}, status_code=404)
        last_event_id = request.headers.get("last-event-id")

        async def stream():
            # Last-Event-ID replay (reference: InMemoryEventStore.replay_events_after :615)
            for ev in engine.sessions.event_store.replay_after(sess.session_id, last_event_id):
                yield f"id: {ev.event_id}\nevent: message\ndata: {json.dumps(ev.message)}\n\n"
            async for eid, message in engine.sessions.respond_stream(sess.session_id,
                                                                     keepalive_s=engine.settings.sse_keepalive_interval):
                if message.get("type") == "keepalive":
                    yield ": keepalive\n\n"
                else:
                    prefix = f"id: {eid}\n" if eid else ""
                    yield f"{prefix}event: message\ndata: {json.dumps(message)}\n\n"

        return StreamingResponse(stream(), media_type="text/event-stream")

    @app.delete("/mcp")
    async def mcp_delete(request: Request, ctx: AuthContext = Depends(get_auth)):
        session_id = request.headers.get("mcp-session-id")
        if session_id:
            engine.sessions.remove(session_id)
        return Response(status_code=204)

    # -- legacy SSE pair (reference: main.py:4478 sse_endpoint / :4597 message_endpoint) --
    @app.get("/servers/{server_id}/sse")
    async def sse_endpoint(server_id: str, request: Request, ctx: AuthContext = Depends(get_auth)):
        sess = engine.sessions.create(transport="sse", server_id=server_id, user=ctx.user)
        endpoint_url = f"/servers/{server_id}/message?session_id={sess.session_id}"

        async def stream():
            yield f"event: endpoint\ndata: {endpoint_url}\n\n"
            async for eid, message in engine.sessions.respond_stream(sess.session_id,
                                                                     keepalive_s=engine.settings.sse_keepalive_interval):
                if message.get("type") == "keepalive":
                    yield ": keepalive\n\n"
                else:
                    yield f"event: message\ndata: {json.dumps(message)}\n\n"

        return StreamingResponse(stream(), media_type="text/event-stream")

    @app.post("/servers/{server_id}/message")
    async def message_endpoint(server_id: str, request: Request, session_id: str = Query(...),
                               ctx: AuthContext = Depends(get_auth)):
        raw = await request.body()
        sess = engine.sessions.get(session_id)
        if sess is None:
            return JSONResponse({"detail": "session not found"}, status_code=404)
        out = await engine.handle_rpc_bytes(raw, user=ctx.user, server_id=server_id, session=sess)
        if out is not None:
            await engine.sessions.broadcast(session_id, json.loads(out))
        return JSONResponse({"status": "accepted"}, status_code=202)

    # -- WebSocket /ws (reference: main.py:11866) ---------------------------------
    @app.websocket("/ws")
    async def websocket_endpoint(ws: WebSocket):
        try:
            ctx = auth.authenticate(ws.headers.get("authorization"))
        except AuthError:
            await ws.close(code=1008)
            return
        await ws.accept()
        sess = engine.sessions.create(transport="websocket", user=ctx.user)
        try:
            while True:
                raw = await ws.receive_text()
                out = await engine.handle_rpc_bytes(raw.encode(), user=ctx.user, session=sess)
                if out is not None:
                    await ws.send_text(out.decode())
        except WebSocketDisconnect:
            pass
        finally:
            engine.sessions.remove(sess.session_id)

    # -- entity CRUD (reference: main.py:3557 routers) ----------------------------
    ENTITY_ROUTES = {
        "tools": "tool", "gateways": "gateway", "servers": "server",
        "resources": "resource", "prompts": "prompt", "a2a": "a2a_agent",
    }

    def _register_crud(plural: str, kind: str) -> None:
        perm = {"a2a": "tools"}.get(plural, plural)

        @app.get(f"/{plural}", name=f"list_{plural}")
        async def list_entities(include_inactive: bool = False, limit: int = 0, cursor: int = 0,
                                ctx: AuthContext = Depends(require(f"{perm}.read"))):
            # cursor pagination (reference: utils/pagination.py nextCursor)
            items = engine.registry.list(kind, include_disabled=include_inactive)
            if limit <= 0:
                return items
            page = items[cursor:cursor + limit]
            from fastapi.encoders import jsonable_encoder

            next_cursor = cursor + limit if cursor + limit < len(items) else None
            return JSONResponse(jsonable_encoder({"items": page, "nextCursor": next_cursor,
                                                  "total": len(items)}),
                                headers={"X-Total-Count": str(len(items))})

        @app.get(f"/{plural}/{{entity_id}}", name=f"get_{plural}")
        async def get_entity(entity_id: str, ctx: AuthContext = Depends(require(f"{perm}.read"))):
            try:
                return engine.registry.get(kind, entity_id)
            except NotFoundError as exc:
                raise HTTPException(404, str(exc)) from exc

        @app.post(f"/{plural}", status_code=201, name=f"create_{plural}")
        async def create_entity(request: Request, ctx: AuthContext = Depends(require(f"{perm}.create"))):
            body = await request.json()
            # typed request validation (reference: schemas.py models) —
            # field-level 422s before anything touches the registry
            from pydantic import ValidationError

            from ..protocol.schemas import validate_create

            try:
                body = validate_create(kind, body if isinstance(body, dict) else {})
            except ValidationError as exc:
                raise HTTPException(422, detail=json.loads(exc.json())) from exc
            try:
                if kind == "gateway":
                    # defer=true → async lifecycle: 201 with a `pending` row;
                    # the lifecycle loop initializes it with backoff
                    # (reference: _register_gateway_pending :1564)
                    return await engine.gateway_service.register_gateway(
                        name=body["name"], url=body.get("url", ""),
                        transport=body.get("transport", "streamablehttp"),
                        description=body.get("description", ""),
                        auth_type=body.get("auth_type"), auth_value=body.get("auth_value"),
                        tags=body.get("tags"), defer=bool(body.get("defer", False)))
                if kind == "tool":
                    body.setdefault("original_name", body.get("name"))
                    body.setdefault("integration_type", "REST" if body.get("url") else "LOCAL")
                if kind == "a2a_agent":
                    from ..utils import slugify

                    body.setdefault("slug", slugify(body["name"]))
                ent = engine.registry.create(kind, **body)
                engine.audit.record(ctx.user, "create", kind, ent.get("id"))
                await engine.notify_list_changed(kind)
                engine.invalidate_peers(kind)
                return ent
            except ConflictError as exc:
                raise HTTPException(409, str(exc)) from exc
            except GatewayConnectionError as exc:
                raise HTTPException(502, str(exc)) from exc
            except (TypeError, KeyError) as exc:
                raise HTTPException(422, f"invalid fields: {exc}") from exc

        @app.put(f"/{plural}/{{entity_id}}", name=f"update_{plural}")
        async def update_entity(entity_id: str, request: Request,
                                ctx: AuthContext = Depends(require(f"{perm}.update"))):
            body = await request.json()
            try:
                return engine.registry.update(kind, entity_id, **body)
            except NotFoundError as exc:
                raise HTTPException(404, str(exc)) from exc
            except TypeError as exc:
                raise HTTPException(422, str(exc)) from exc

        @app.delete(f"/{plural}/{{entity_id}}", status_code=204, name=f"delete_{plural}")
        async def delete_entity(entity_id: str, ctx: AuthContext = Depends(require(f"{perm}.delete"))):
            try:
                if kind == "gateway":
                    await engine.gateway_service.delete_gateway(entity_id)
                else:
                    engine.registry.delete(kind, entity_id)
                engine.audit.record(ctx.user, "delete", kind, entity_id)
                await engine.notify_list_changed(kind)
                engine.invalidate_peers(kind)
            except NotFoundError as exc:
                raise HTTPException(404, str(exc)) from exc
            return Response(status_code=204)

        @app.post(f"/{plural}/{{entity_id}}/toggle", name=f"toggle_{plural}")
        async def toggle_entity(entity_id: str, activate: bool = True,
                                ctx: AuthContext = Depends(require(f"{perm}.update"))):
            try:
                return engine.registry.set_enabled(kind, entity_id, activate)
            except NotFoundError as exc:
                raise HTTPException(404, str(exc)) from exc

    for plural, kind in ENTITY_ROUTES.items():
        _register_crud(plural, kind)

    @app.get("/search")
    async def global_search(q: str, kinds: str = "", limit: int = 50,
                            ctx: AuthContext = Depends(require("tools.read"))):
        """Global entity search (reference: routers/search.py): substring
        match over name/description/tags across registry kinds."""
        klist = [k.strip() for k in kinds.split(",") if k.strip()] or             ["tool", "gateway", "server", "resource", "prompt", "a2a_agent"]
        ql = q.lower()
        out = []
        for kind in klist:
            if kind not in ("tool", "gateway", "server", "resource", "prompt", "a2a_agent"):
                continue
            for ent in engine.registry.list(kind):
                hay = " ".join([str(ent.get("name") or ent.get("uri") or ""),
                                str(ent.get("description") or ""),
                                " ".join(ent.get("tags") or [])]).lower()
                if ql in hay:
                    out.append({"kind": kind, "id": ent.get("id"),
                                "name": ent.get("name") or ent.get("uri"),
                                "description": (ent.get("description") or "")[:200]})
                    if len(out) >= limit:
                        return out
        return out

    # -- per-tool plugin bindings (reference: routers/tool_plugin_bindings.py) --
    @app.get("/tools/{tool_name}/plugin-bindings")
    async def list_plugin_bindings(tool_name: str, ctx: AuthContext = Depends(require("tools.read"))):
        return [b for b in engine.registry.list("plugin_binding") if b["tool_name"] == tool_name]

    @app.put("/tools/{tool_name}/plugin-bindings/{plugin_name}")
    async def set_plugin_binding(tool_name: str, plugin_name: str, request: Request,
                                 ctx: AuthContext = Depends(require("tools.update"))):
        body = await request.json() if (await request.body()) else {}
        mode = body.get("mode")
        if mode is not None and mode not in ("enforce", "enforce_ignore_error", "permissive", "disabled"):
            raise HTTPException(422, f"invalid mode {mode!r}")
        if engine.plugins.get(plugin_name) is None:
            raise HTTPException(404, f"plugin {plugin_name!r} not registered")
        ent = engine.set_plugin_binding(tool_name, plugin_name, mode=mode, config=body.get("config"))
        engine.audit.record(ctx.user, "bind", "plugin_binding", ent.get("id"))
        return ent

    @app.delete("/tools/{tool_name}/plugin-bindings/{plugin_name}", status_code=204)
    async def delete_plugin_binding(tool_name: str, plugin_name: str,
                                    ctx: AuthContext = Depends(require("tools.update"))):
        try:
            engine.delete_plugin_binding(tool_name, plugin_name)
        except NotFoundError as exc:
            raise HTTPException(404, str(exc)) from exc
        return Response(status_code=204)

    # agent-hook bindings share the same keyed map (ctx.name = agent name)
    # (reference: routers/a2a_agent_plugin_bindings.py)
    app.add_api_route("/a2a/{tool_name}/plugin-bindings", list_plugin_bindings, methods=["GET"])
    app.add_api_route("/a2a/{tool_name}/plugin-bindings/{plugin_name}", set_plugin_binding, methods=["PUT"])
    app.add_api_route("/a2a/{tool_name}/plugin-bindings/{plugin_name}", delete_plugin_binding,
                      methods=["DELETE"], status_code=204)

    @app.post("/gateways/{gateway_id}/refresh")
    async def refresh_gateway(gateway_id: str, ctx: AuthContext = Depends(require("gateways.update"))):
        try:
            return await engine.gateway_service.refresh_gateway(gateway_id)
        except NotFoundError as exc:
            raise HTTPException(404, str(exc)) from exc
        except Exception as exc:
            raise HTTPException(502, str(exc)) from exc

    @app.post("/gateways/{gateway_id}/oauth/authorize")
    async def gateway_oauth_authorize(gateway_id: str, request: Request,
                                      ctx: AuthContext = Depends(require("gateways.update"))):
        """Start the upstream authorization-code flow: returns the provider
        authorize URL + signed state for the operator's browser
        (reference: oauth_router initiate)."""
        from ..auth.oauth import OAuthError

        body = {}
        if await request.body():
            body = await request.json()
        try:
            return engine.gateway_service.begin_upstream_authorization(
                gateway_id, redirect_uri=body.get("redirect_uri"))
        except NotFoundError as exc:
            raise HTTPException(404, str(exc)) from exc
        except OAuthError as exc:
            raise HTTPException(422, str(exc)) from exc

    @app.get("/oauth/upstream/callback")
    async def gateway_oauth_callback(code: str, state: str):
        """Provider redirect target: exchanges the code, stores tokens
        sealed (reference: oauth_router callback). Unauthenticated by
        design — the signed state is the proof of initiation."""
        from ..auth.oauth import OAuthError

        try:
            gid = await engine.gateway_service.complete_upstream_authorization(code, state)
        except OAuthError as exc:
            raise HTTPException(400, str(exc)) from exc
        return {"status": "authorized", "gateway_id": gid}

    @app.post("/gateways/{gateway_id}/retry")
    async def retry_gateway(gateway_id: str, ctx: AuthContext = Depends(require("gateways.update"))):
        """Put a `failed` lifecycle row back into `pending` (operator action)."""
        try:
            return await engine.gateway_service.retry_failed_gateway(gateway_id)
        except NotFoundError as exc:
            raise HTTPException(404, str(exc)) from exc

    # -- export / import (reference: services/export_service.py:268) --------------
    @app.get("/export")
    async def export_config(ctx: AuthContext = Depends(require("admin.export"))):
        return engine.registry.export_configuration()

    @app.post("/import")
    async def import_config(request: Request, conflict_strategy: str = "update",
                            ctx: AuthContext = Depends(require("admin.import"))):
        body = await request.json()
        out = engine.registry.import_configuration(body, conflict_strategy)
        engine.sync_plugin_bindings()   # imported bindings reach the manager
        engine.invalidate_peers("import")
        return out

    # -- auth endpoints (reference: routers/auth.py, routers/tokens.py) ------------
    @app.post("/auth/login")
    async def login(request: Request):
        body = await request.json()
        ctx = auth.verify_user(body.get("email", ""), body.get("password", ""))
        if ctx is None:
            raise HTTPException(401, "invalid credentials")
        from ..auth import jwt as jwt_mod

        token = jwt_mod.create_token({"sub": ctx.user, "admin": ctx.is_admin}, settings.jwt_secret_key,
                                     expires_minutes=settings.token_expiry,
                                     audience=settings.jwt_audience, issuer=settings.jwt_issuer)
        return {"access_token": token, "token_type": "bearer"}

    @app.post("/tokens", status_code=201)
    async def create_token_ep(request: Request, ctx: AuthContext = Depends(get_auth)):
        body = await request.json()
        raw = auth.create_api_token(ctx.user, body.get("name", "token"), scopes=body.get("scopes"),
                                    server_id=body.get("server_id"),
                                    expires_minutes=body.get("expires_minutes"))
        return {"token": raw}

    @app.get("/tokens")
    async def list_tokens_ep(ctx: AuthContext = Depends(get_auth)):
        return auth.list_api_tokens(ctx.user)

    @app.delete("/tokens/{token_id}", status_code=204)
    async def revoke_token_ep(token_id: str, ctx: AuthContext = Depends(get_auth)):
        auth.revoke_api_token(token_id)
        return Response(status_code=204)

    # -- A2A invoke (reference: routers a2a + a2a_service.invoke_agent :1997) ------
    @app.post("/a2a/{agent_name}/invoke")
    async def a2a_invoke(agent_name: str, request: Request,
                         ctx: AuthContext = Depends(require("tools.invoke"))):
        from ..services.a2a_service import A2AError

        body = await request.json()
        message = body.get("message") if isinstance(body, dict) else None
        if not isinstance(message, str):
            message = json.dumps(body, default=str)
        try:
            return await engine.a2a_service.invoke_agent(
                agent_name, message, user=ctx.user, context=body.get("context"),
                hop_count=int(request.headers.get("x-a2a-hop-count", "0")))
        except A2AError as exc:
            status = {-32602: 404, -32003: 403, -32002: 502}.get(exc.code, 502)
            raise HTTPException(status, str(exc)) from exc

    # -- LLM proxy (reference: llm_proxy_service :103,442; routers/llm_*) ---------
    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request, ctx: AuthContext = Depends(get_auth)):
        from ..services.llm_proxy import LLMProxyError

        body = await request.json()
        try:
            if body.get("stream"):
                stream = engine.llm_proxy.chat_completions_stream(body)
                return StreamingResponse(stream, media_type="text/event-stream")
            return await engine.llm_proxy.chat_completions(body)
        except LLMProxyError as exc:
            raise HTTPException(exc.status, str(exc)) from exc

    @app.post("/chat")
    async def llm_chat(request: Request, ctx: AuthContext = Depends(require("tools.invoke"))):
        """Built-in LLM chat with tool-calling over the registry
        (reference: routers/llmchat_router.py + mcp_client_chat_service)."""
        body = await request.json()
        messages = body.get("messages")
        if not isinstance(messages, list) or not messages:
            raise HTTPException(422, "missing messages")
        try:
            return await engine.chat.chat(messages, model=body.get("model"),
                                          provider=body.get("provider"), user=ctx.user,
                                          server_id=ctx.server_id,
                                          max_rounds=body.get("max_rounds"))
        except LLMProxyError as exc:
            raise HTTPException(exc.status, str(exc)) from exc

    @app.get("/llm/providers")
    async def list_llm_providers(ctx: AuthContext = Depends(require("admin.read"))):
        return engine.llm_proxy.registry.list()

    @app.post("/llm/providers", status_code=201)
    async def add_llm_provider(request: Request, ctx: AuthContext = Depends(require("admin.update"))):
        body = await request.json()
        return engine.llm_proxy.registry.register(
            name=body["name"], base_url=body["base_url"], api_key=body.get("api_key"),
            models=body.get("models"), default_model=body.get("default_model"))

    # -- reverse proxy tunnel (reference: mcpgateway/reverse_proxy.py) -------------
    from .reverse_proxy import TunnelRegistry, TunnelUpstream

    app.state.tunnels = TunnelRegistry()

    @app.post("/reverse-proxy/register", status_code=201)
    async def rp_register(request: Request, ctx: AuthContext = Depends(require("gateways.create"))):
        body = await request.json()
        name = body.get("name") or f"tunnel-{uuid.uuid4().hex[:6]}"
        tunnel = app.state.tunnels.create(name)
        return {"tunnel_id": tunnel.tunnel_id, "name": name}

    @app.get("/reverse-proxy/stream")
    async def rp_stream(tunnel_id: str, ctx: AuthContext = Depends(require("gateways.create"))):
        tunnel = app.state.tunnels.get(tunnel_id)
        if tunnel is None:
            raise HTTPException(404, "tunnel not found")
        tunnel.connected = True

        async def gen():
            # once the client is pumping, register the tunneled upstream as a
            # federated gateway (sync happens over the tunnel itself)
            async def do_register():
                try:
                    await engine.gateway_service.register_gateway(
                        name=tunnel.name, url=f"tunnel://{tunnel.tunnel_id}",
                        client=TunnelUpstream(tunnel))
                except Exception:
                    pass

            reg_task = asyncio.create_task(do_register())
            try:
                while True:
                    try:
                        body = await asyncio.wait_for(tunnel.requests.get(),
                                                      timeout=engine.settings.sse_keepalive_interval)
                    except asyncio.TimeoutError:
                        yield ": keepalive\n\n"
                        continue
                    yield f"data: {json.dumps(body, separators=(',', ':'))}\n\n"
            finally:
                reg_task.cancel()
                tunnel.connected = False

        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.post("/reverse-proxy/respond")
    async def rp_respond(tunnel_id: str, request: Request,
                         ctx: AuthContext = Depends(require("gateways.create"))):
        tunnel = app.state.tunnels.get(tunnel_id)
        if tunnel is None:
            raise HTTPException(404, "tunnel not found")
        body = await request.json()
        ok = tunnel.resolve(body.get("id"), body)
        return {"resolved": ok}

    @app.delete("/reverse-proxy/{tunnel_id}", status_code=204)
    async def rp_delete(tunnel_id: str, ctx: AuthContext = Depends(require("gateways.create"))):
        tunnel = app.state.tunnels.get(tunnel_id)
        if tunnel is not None:
            gw = engine.registry.find("gateway", tunnel.name)
            if gw is not None:
                await engine.gateway_service.delete_gateway(gw["id"])
            app.state.tunnels.remove(tunnel_id)
        return Response(status_code=204)

    # -- catalog (reference: services/catalog_service.py) --------------------------
    @app.get("/tags")
    async def list_tags(kinds: str = "", include_entities: bool = False,
                        ctx: AuthContext = Depends(require("tools.read"))):
        """Cross-entity tag aggregation (reference: services/tag_service.py)."""
        klist = [k.strip() for k in kinds.split(",") if k.strip()] or None
        return engine.tags.list_tags(klist, include_entities=include_entities)

    @app.get("/tags/{tag}/entities")
    async def tag_entities(tag: str, kinds: str = "",
                           ctx: AuthContext = Depends(require("tools.read"))):
        klist = [k.strip() for k in kinds.split(",") if k.strip()] or None
        return engine.tags.entities_for_tag(tag, klist)

    @app.get("/catalog")
    async def catalog_list(category: Optional[str] = None, ctx: AuthContext = Depends(require("tools.read"))):
        return engine.catalog.list(category)

    @app.post("/catalog/{catalog_id}/register", status_code=201)
    async def catalog_register(catalog_id: str, ctx: AuthContext = Depends(require("gateways.create"))):
        try:
            return await engine.catalog.register_from_catalog(catalog_id, engine.gateway_service)
        except KeyError as exc:
            raise HTTPException(404, str(exc)) from exc
        except GatewayConnectionError as exc:
            raise HTTPException(502, str(exc)) from exc

    # -- user management with password policy (reference: email_auth + password_policy)
    @app.post("/auth/register", status_code=201)
    async def register_user(request: Request, ctx: AuthContext = Depends(require("admin.update"))):
        body = await request.json()
        errs = engine.password_policy.validate(body.get("password", ""))
        if errs:
            raise HTTPException(422, "; ".join(errs))
        auth.create_user(body["email"], body["password"], body.get("full_name", ""),
                         bool(body.get("is_admin", False)))
        engine.audit.record(ctx.user, "create", "user", body["email"])
        return {"email": body["email"]}

    # -- SSO (reference: services/sso_service.py; routers/sso.py) ------------------
    from ..auth.sso import SSOError, SSOService

    sso = SSOService(auth, settings)
    app.state.sso = sso

    @app.post("/auth/sso/providers", status_code=201)
    async def sso_add_provider(request: Request, ctx: AuthContext = Depends(require("admin.update"))):
        body = await request.json()
        try:
            return sso.register_provider(body["name"], body["client_id"], body["client_secret"],
                                         preset=body.get("preset", "oidc"),
                                         **{k: v for k, v in body.items()
                                            if k in ("authorize_url", "token_url", "userinfo_url",
                                                     "email_field", "scopes")})
        except SSOError as exc:
            raise HTTPException(exc.status, str(exc)) from exc

    @app.get("/auth/sso/providers")
    async def sso_list_providers():
        return [{"name": n} for n in sso.providers]

    @app.get("/auth/sso/{provider}/login")
    async def sso_login(provider: str, request: Request):
        redirect_uri = str(request.base_url).rstrip("/") + f"/auth/sso/{provider}/callback"
        try:
            url = sso.login_url(provider, redirect_uri)
        except SSOError as exc:
            raise HTTPException(exc.status, str(exc)) from exc
        from fastapi.responses import RedirectResponse

        return RedirectResponse(url, status_code=302)

    @app.get("/auth/sso/{provider}/callback")
    async def sso_callback(provider: str, code: str, state: str, request: Request):
        redirect_uri = str(request.base_url).rstrip("/") + f"/auth/sso/{provider}/callback"
        try:
            return await sso.handle_callback(provider, code, state, redirect_uri)
        except SSOError as exc:
            raise HTTPException(exc.status, str(exc)) from exc

    # -- well-known (reference: routers/well_known.py RFC 9728) --------------------
    @app.get("/.well-known/oauth-protected-resource")
    async def oauth_protected_resource(request: Request):
        base = str(request.base_url).rstrip("/")
        return {"resource": f"{base}/mcp", "authorization_servers": [],
                "bearer_methods_supported": ["header"], "resource_name": settings.app_name}

    @app.get("/.well-known/security.txt")
    async def security_txt():
        return PlainTextResponse("Contact: mailto:security@example.com\n")

    # -- admin API (reference: mcpgateway/admin.py, trimmed) -----------------------
    @app.get("/admin/stats")
    async def admin_stats(ctx: AuthContext = Depends(require("admin.read"))):
        out = {
            "uptime_s": round(time.time() - app.state.started, 1),
            "entities": {k: len(engine.registry.list(k)) for k in
                         ("tool", "gateway", "server", "resource", "prompt", "a2a_agent")},
            "sessions": engine.sessions.count(),
            "metrics": engine.metrics.snapshot(),
            "plugins": engine.plugins.stats(),
        }
        if engine.gpu_pipeline is not None:
            out["gpu"] = engine.gpu_pipeline.stats()
        return out

    @app.get("/admin/traces")
    async def admin_traces(limit: int = 100, ctx: AuthContext = Depends(require("admin.read"))):
        engine.observability.flush()
        return engine.observability.query_traces(limit)

    @app.get("/admin/siem/export")
    async def siem_export(limit: int = 1000, ctx: AuthContext = Depends(require("admin.read"))):
        """Security events as NDJSON (reference: siem_export_service)."""
        return PlainTextResponse(engine.siem.export_jsonl(limit=limit),
                                 media_type="application/x-ndjson")

    @app.post("/admin/siem/push")
    async def siem_push(request: Request, ctx: AuthContext = Depends(require("admin.write"))):
        body = await request.json()
        url = body.get("url")
        if not url:
            raise HTTPException(422, "missing url")
        n = await engine.siem.push_webhook(url, limit=int(body.get("limit", 1000)))
        return {"pushed": n}

    @app.get("/admin/classification")
    async def admin_classification(ctx: AuthContext = Depends(require("admin.read"))):
        """Server/gateway capability classification (reference: server_classification)."""
        return engine.classification.classify_all()

    @app.get("/admin/compliance/report")
    async def compliance_report(ctx: AuthContext = Depends(require("admin.read"))):
        return engine.compliance.report()

    @app.get("/admin/logs")
    async def admin_logs(level: str = "", logger_name: str = "", q: str = "",
                         limit: int = 200, ctx: AuthContext = Depends(require("admin.read"))):
        """Structured-log search (reference: routers/log_search.py +
        log_storage_service): filter persisted logs by level/logger/text."""
        from sqlalchemy import select

        from ..db.models import DbStructuredLog

        stmt = select(DbStructuredLog).order_by(DbStructuredLog.id.desc()).limit(min(limit, 2000))
        if level:
            stmt = stmt.where(DbStructuredLog.level == level.upper())
        if logger_name:
            stmt = stmt.where(DbStructuredLog.logger.like(f"%{logger_name}%"))
        if q:
            stmt = stmt.where(DbStructuredLog.message.like(f"%{q}%"))
        with engine.db.session() as s_:
            return [{"timestamp": r.timestamp.isoformat(), "level": r.level,
                     "logger": r.logger, "message": r.message, "context": r.context}
                    for r in s_.execute(stmt).scalars()]

    @app.get("/admin/audit")
    async def admin_audit(limit: int = 100, ctx: AuthContext = Depends(require("admin.read"))):
        return engine.audit.query(limit)

    @app.get("/admin/token-usage")
    async def admin_token_usage(limit: int = 200,
                                ctx: AuthContext = Depends(require("admin.read"))):
        """Per-credential hourly usage (reference: TokenUsageLog +
        TokenUsageMiddleware)."""
        return engine.token_usage.query(limit)

    @app.get("/admin/metrics/rollups")
    async def admin_rollups(ctx: AuthContext = Depends(require("admin.read"))):
        # roll raw rows into hourly aggregates and return them
        from sqlalchemy import select

        from ..db.models import DbMetricRollup
        from ..services.metrics import rollup_hourly

        engine.metrics.flush()
        rollup_hourly(engine.db)
        with engine.db.session() as s_:
            rows = s_.execute(select(DbMetricRollup).order_by(DbMetricRollup.hour.desc()).limit(200)).scalars().all()
            return [{"entity_id": r.entity_id, "hour": str(r.hour), "count": r.count,
                     "error_count": r.error_count, "avg_ms": (r.total_ms / r.count) if r.count else 0}
                    for r in rows]

    @app.get("/admin/support-bundle")
    async def admin_support_bundle(ctx: AuthContext = Depends(require("admin.read"))):
        return engine.support_bundle.collect()

    @app.get("/admin/performance")
    async def admin_performance(ctx: AuthContext = Depends(require("admin.read"))):
        engine.performance.snapshot()
        return engine.performance.history()

    @app.get("/admin/runtime")
    async def admin_runtime_get(ctx: AuthContext = Depends(require("admin.read"))):
        """Runtime data-plane state (reference: runtime_state.py shadow↔edge +
        runtime_admin_router PATCH): which path serves /rpc right now."""
        col = app.state.collector
        return {"mode": "gpu-batched" if (engine.gpu_pipeline is not None and col is not None)
                else ("gpu-direct" if engine.gpu_pipeline is not None else "cpu"),
                "gpu_pipeline": engine.gpu_pipeline is not None,
                "collector": None if col is None else
                {"window_us": int(col.window_s * 1e6), "max_batch": col.max_batch,
                 "batches": col.batches, "max_seen": col.max_seen}}

    @app.patch("/admin/runtime")
    async def admin_runtime_patch(request: Request, ctx: AuthContext = Depends(require("admin.write"))):
        """Runtime-flippable serving mode: batching window/size, or detach
        the collector entirely (per-request mode) without a restart."""
        body = await request.json()
        col = app.state.collector
        if "window_us" in body and col is not None:
            col.window_s = max(0, int(body["window_us"])) / 1e6
        if "max_batch" in body and col is not None:
            col.max_batch = max(1, int(body["max_batch"]))
        if "timing" in body and engine.gpu_pipeline is not None:
            engine.gpu_pipeline.timing_enabled = bool(body["timing"])
            if not body["timing"]:
                engine.gpu_pipeline.timing.clear()
        if body.get("mode") == "cpu" and engine.gpu_pipeline is not None:
            app.state.collector = None     # per-request CPU chain
        elif body.get("mode") == "gpu-batched" and engine.gpu_pipeline is not None and col is None:
            app.state.collector = BatchCollector(engine.process_rpc_batch,
                                                 settings.gpu_batch_max_requests,
                                                 settings.gpu_batch_window_us)
        engine.audit.record(ctx.user, "update", "runtime", None, detail=body)
        return await admin_runtime_get(ctx)

    @app.post("/toolops/{tool_name}/test")
    async def toolops_test(tool_name: str, count: int = 3,
                           ctx: AuthContext = Depends(require("tools.invoke"))):
        try:
            return await engine.toolops.run_tests(tool_name, count)
        except KeyError as exc:
            raise HTTPException(404, str(exc)) from exc

    @app.get("/admin/plugins")
    async def admin_plugins(ctx: AuthContext = Depends(require("admin.read"))):
        return [{"name": p.name, "mode": p.mode.value, "priority": p.priority,
                 "hooks": [h.value for h in p.hooks], "gpu_capable": p.gpu_capable}
                for p in engine.plugins.plugins]

    @app.post("/admin/plugins/{name}/mode")
    async def admin_plugin_mode(name: str, mode: str, ctx: AuthContext = Depends(require("admin.update"))):
        from ..plugins.framework import PluginMode

        p = engine.plugins.get(name)
        if p is None:
            raise HTTPException(404, f"plugin {name} not found")
        p.mode = PluginMode(mode)
        engine.plugins.bump()
        return {"name": name, "mode": p.mode.value}

    @app.get("/admin", response_class=HTMLResponse)
    async def admin_ui(ctx: AuthContext = Depends(require("admin.read"))):
        from ..admin.ui import render_admin_page

        return render_admin_page(engine)

    # -- admin UI partials (reference: admin.py HTMX partial endpoints) -----
    @app.get("/admin/ui/{partial}", response_class=HTMLResponse)
    async def admin_ui_partial(partial: str, q: str = "", level: str = "",
                               ctx: AuthContext = Depends(require("admin.read"))):
        from ..admin import ui as adm

        kind_map = {v: k for k, v in adm._ENTITY_PLURAL.items()}
        if partial == "dashboard":
            return adm.partial_dashboard(engine)
        if partial in kind_map:
            return adm.partial_entities(engine, kind_map[partial])
        if partial == "plugins":
            return adm.partial_plugins(engine)
        if partial == "bindings":
            return adm.partial_bindings(engine)
        if partial == "metrics":
            return adm.partial_metrics(engine)
        if partial == "logs":
            return adm.partial_logs(engine, q=q, level=level)
        if partial == "traces":
            return adm.partial_traces(engine)
        if partial == "audit":
            return adm.partial_audit(engine)
        if partial == "tokens":
            return adm.partial_tokens(engine, auth)
        if partial == "runtime":
            ne = getattr(app.state, "native_edge", None)
            return adm.partial_runtime(engine, collector=app.state.collector,
                                       edge_stats=ne.stats() if ne else None)
        raise HTTPException(404, f"unknown partial {partial!r}")

    @app.post("/admin/ui/bind")
    async def admin_ui_bind(request: Request, ctx: AuthContext = Depends(require("tools.update"))):
        body = await request.json()
        if engine.plugins.get(body.get("plugin_name", "")) is None:
            raise HTTPException(404, f"plugin {body.get('plugin_name')!r} not registered")
        engine.set_plugin_binding(body["tool_name"], body["plugin_name"],
                                  mode=body.get("mode") or None)
        engine.audit.record(ctx.user, "bind", "plugin_binding",
                            f"{body['tool_name']}::{body['plugin_name']}")
        return {"status": "bound"}

    @app.post("/admin/ui/runtime")
    async def admin_ui_runtime(request: Request, ctx: AuthContext = Depends(require("admin.write"))):
        body = await request.json()
        col = app.state.collector
        if col is not None:
            if body.get("window_us"):
                col.window_s = max(0, int(body["window_us"])) / 1e6
            if body.get("max_batch"):
                col.max_batch = max(1, int(body["max_batch"]))
        engine.audit.record(ctx.user, "update", "runtime", None, detail=body)
        return {"status": "applied"}

    return app


def serve(engine: Optional[GatewayEngine] = None, **uvicorn_kwargs) -> None:
    """CLI entrypoint (reference: mcpgateway/cli.py uvicorn wrapper)."""
    import uvicorn

    engine = engine or GatewayEngine()
    app = build_app(engine)
    settings = engine.settings
    uvicorn.run(app, host=settings.host, port=settings.port, **uvicorn_kwargs)
