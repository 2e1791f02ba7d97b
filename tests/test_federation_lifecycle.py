"""Async federation lifecycle (reference: gateway_service.py:4077-4362) and
the SSE upstream transport client (reference: connect_to_sse_server :6900).

The SSE tier federates one in-proc forge into another THROUGH the real SSE
pair endpoints (GET /servers/{id}/sse + POST /servers/{id}/message) over
httpx ASGITransport — wire framing, endpoint event, id-correlated message
frames all exercised."""

import asyncio
import base64
import json

import httpx
import pytest

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.services.gateway_service import GatewayService
from mcp_context_forge_amd.services.upstream import InProcUpstream, SseUpstreamClient, UpstreamError

BASIC = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}


class FlakyUpstream(InProcUpstream):
    """Fails `fail_times` initializations, then behaves."""

    def __init__(self, fail_times: int, exc: Exception = None):
        super().__init__("flaky")
        self.fail_times = fail_times
        self.attempts = 0
        self.exc = exc or UpstreamError("upstream flaky unreachable: connect refused")

        async def echo(args):
            return args

        self.add_tool("echo", echo, "echo")

    async def initialize(self):
        self.attempts += 1
        if self.attempts <= self.fail_times:
            raise self.exc
        return await super().initialize()


def _engine(**kw):
    s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=False,
                 plugins_enabled=False, gateway_retry_base_s=0.02, gateway_retry_cap_s=0.1,
                 gateway_lifecycle_tick_s=0.02, gateway_max_retries=4, **kw)
    return GatewayEngine(s)


def test_deferred_registration_backoff_then_active(run):
    async def go():
        e = _engine()
        up = FlakyUpstream(fail_times=2)
        gw = await e.gateway_service.register_gateway(
            name="peer", url="inproc://peer", client=up, owner_rank=0, defer=True)
        assert gw["status"] == "pending"
        # lifecycle loop: fail → backoff → fail → backoff → active
        for _ in range(300):
            gw = e.registry.get("gateway", gw["id"])
            if gw["status"] == "active":
                break
            await asyncio.sleep(0.02)
        assert gw["status"] == "active", gw
        assert up.attempts == 3
        assert gw["retry_count"] == 0 and gw["last_error"] is None
        # tools were synced on activation
        assert e.registry.find("tool", "peer-echo") is not None
        await e.shutdown()

    run(go())


def test_deferred_registration_failure_classification_and_terminal(run):
    async def go():
        e = _engine()
        up = FlakyUpstream(fail_times=99)
        gw = await e.gateway_service.register_gateway(
            name="dead", url="inproc://dead", client=up, owner_rank=0, defer=True)
        for _ in range(400):
            gw = e.registry.get("gateway", gw["id"])
            if gw["status"] == "failed":
                break
            await asyncio.sleep(0.02)
        assert gw["status"] == "failed", gw
        assert gw["retry_count"] == 4          # gateway_max_retries
        assert gw["failure_class"] == "connect_error"
        assert "unreachable" in gw["last_error"]
        assert gw["reachable"] is False
        # operator retry puts it back into the lifecycle; upstream recovered
        up.fail_times = 0
        gw = await e.gateway_service.retry_failed_gateway(gw["id"])
        assert gw["status"] == "pending"
        for _ in range(300):
            gw = e.registry.get("gateway", gw["id"])
            if gw["status"] == "active":
                break
            await asyncio.sleep(0.02)
        assert gw["status"] == "active"
        await e.shutdown()

    run(go())


def test_classification_table():
    cls = GatewayService.classify_failure
    assert cls(asyncio.TimeoutError()) == "timeout"
    assert cls(UpstreamError("upstream x unreachable: connect refused")) == "connect_error"
    assert cls(UpstreamError("upstream x HTTP 401")) == "auth_error"
    assert cls(UpstreamError("upstream x HTTP 503")) == "http_error"
    assert cls(UpstreamError("upstream x: empty response")) == "protocol_error"
    assert cls(ValueError("boom")) == "internal_error"


def test_deferred_delete(run):
    async def go():
        e = _engine()
        up = FlakyUpstream(fail_times=0)
        gw = await e.gateway_service.register_gateway(
            name="togo", url="inproc://togo", client=up, owner_rank=0)
        assert e.registry.find("tool", "togo-echo") is not None
        await e.gateway_service.delete_gateway(gw["id"], defer=True)
        assert e.registry.get("gateway", gw["id"])["status"] == "deleting"
        for _ in range(200):
            if e.registry.find("gateway", "togo") is None:
                break
            await asyncio.sleep(0.02)
        assert e.registry.find("gateway", "togo") is None
        assert e.registry.find("tool", "togo-echo") is None
        await e.shutdown()

    run(go())


def test_deferred_registration_via_http_route(run):
    from mcp_context_forge_amd.transports.http_app import build_app

    async def go():
        s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True,
                     plugins_enabled=False, gpu_enabled=False,
                     gateway_retry_base_s=0.05, gateway_lifecycle_tick_s=0.05)
        e = GatewayEngine(s)
        app = build_app(e)
        async with app.router.lifespan_context(app):
            async with httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                         base_url="http://gw") as c:
                r = await c.post("/gateways", headers=BASIC,
                                 json={"name": "slow-peer", "url": "http://127.0.0.1:1/mcp",
                                       "defer": True})
                assert r.status_code == 201
                assert r.json()["status"] == "pending"
                # row is queryable while pending; lifecycle keeps retrying
                r2 = await c.get(f"/gateways/{r.json()['id']}", headers=BASIC)
                assert r2.json()["status"] in ("pending", "initializing")

    run(go())


# ---------------------------------------------------------------- SSE client


from contextlib import asynccontextmanager


@asynccontextmanager
async def _sse_backend():
    """A forge instance acting as the upstream, serving its SSE pair on a
    real loopback port (SSE streams are infinite — an in-process ASGI
    transport would buffer them forever)."""
    import socket

    import uvicorn

    from mcp_context_forge_amd.transports.http_app import build_app

    s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=False,
                 plugins_enabled=False, gpu_enabled=False)
    e = GatewayEngine(s)

    async def multiply(args):
        return {"product": args.get("a", 0) * args.get("b", 0)}

    e.tool_service.register_local_tool("multiply", multiply, "Multiply")
    app = build_app(e)
    sock = socket.socket()
    sock.bind(("127.0.0.1", 0))
    port = sock.getsockname()[1]
    sock.close()
    server = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port,
                                           log_level="error", lifespan="on"))
    task = asyncio.create_task(server.serve())
    for _ in range(200):
        if server.started:
            break
        await asyncio.sleep(0.05)
    try:
        yield f"http://127.0.0.1:{port}"
    finally:
        server.should_exit = True
        await asyncio.wait_for(task, timeout=10)


def test_sse_upstream_client_roundtrip(run):
    async def go():
        async with _sse_backend() as base:
            client = SseUpstreamClient(f"{base}/servers/srv1/sse", timeout=10.0)
            init = await client.initialize()
            assert init.get("protocolVersion")
            tools = await client.list_tools()
            assert any(t["name"] == "multiply" for t in tools)
            result = await client.call_tool("multiply", {"a": 6, "b": 7})
            assert result["structuredContent"] == {"product": 42}
            assert await client.ping() is True
            await client.aclose()

    run(go())


def test_sse_federation_end_to_end(run):
    """Gateway A federates gateway B over the SSE transport: register with
    transport=sse (no injected client — _make_client builds the real SSE
    client), tools sync through the stream, tools/call round-trips."""

    async def go():
        async with _sse_backend() as base:
            a = _engine()
            gw = await a.gateway_service.register_gateway(
                name="sse-peer", url=f"{base}/servers/srv1/sse",
                transport="sse", owner_rank=0)
            assert gw["status"] == "active"
            tool = a.registry.find("tool", "sse-peer-multiply")
            assert tool is not None
            out = await a.tool_service.invoke_tool("sse-peer-multiply", {"a": 3, "b": 5})
            assert out["structuredContent"] == {"product": 15}
            await a.shutdown()

    run(go())


def test_make_client_selects_sse_transport():
    e = _engine()
    gws = e.gateway_service
    client = gws._make_client({"url": "http://x/servers/1/sse", "transport": "sse",
                               "auth_type": None, "auth_value": None})
    assert isinstance(client, SseUpstreamClient)
    from mcp_context_forge_amd.services.upstream import HttpUpstreamClient

    client2 = gws._make_client({"url": "http://x/mcp", "transport": "streamablehttp",
                                "auth_type": None, "auth_value": None})
    assert isinstance(client2, HttpUpstreamClient)
    assert not isinstance(client2, SseUpstreamClient)


def test_health_loop_failure_and_enabled_filter(run):
    """Mutation survivors pinned: a failing ping marks failures (the
    except-path must set ok=False) and disabled gateways are skipped."""

    async def go():
        e = _engine()
        up = FlakyUpstream(fail_times=0)
        gw = await e.gateway_service.register_gateway(
            name="hc", url="inproc://hc", client=up, owner_rank=0)

        async def bad_ping():
            raise RuntimeError("down")

        up.ping = bad_ping
        res = await e.gateway_service.check_health_once()
        assert res.get(gw["id"]) is False
        assert e.registry.get("gateway", gw["id"])["consecutive_failures"] >= 1
        # disabled gateways are not checked at all
        e.registry.set_enabled("gateway", gw["id"], False)
        res2 = await e.gateway_service.check_health_once()
        assert gw["id"] not in res2
        await e.shutdown()

    run(go())


def test_backoff_schedules_future_retry(run):
    """next_retry_at must land in the FUTURE after a failed attempt."""
    import time as _t

    async def go():
        e = _engine()
        up = FlakyUpstream(fail_times=99)
        gw = await e.gateway_service.register_gateway(
            name="fut", url="inproc://fut", client=up, owner_rank=0, defer=True)
        for _ in range(200):
            gw = e.registry.get("gateway", gw["id"])
            if (gw.get("retry_count") or 0) >= 1 and gw["status"] == "pending":
                break
            await asyncio.sleep(0.01)
        assert gw["next_retry_at"] is not None and gw["next_retry_at"] > _t.time() - 0.001
        # strictly scheduled AFTER the failure that produced it
        assert gw["next_retry_at"] > float(gw["updated_at"].timestamp()) - 60  # sanity
        await e.shutdown()

    run(go())


def test_make_client_bearer_header(run):
    async def go():
        e = _engine()
        gws = e.gateway_service
        sealed = gws.crypto.seal("tok-123")
        client = gws._make_client({"url": "http://x/mcp", "transport": "streamablehttp",
                                   "auth_type": "bearer", "auth_value": sealed})
        assert client.base_headers.get("authorization") == "Bearer tok-123"
        await e.shutdown()

    run(go())
