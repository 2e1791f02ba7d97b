"""Live-socket tier: real uvicorn server on 127.0.0.1, streaming transports.

Reference analog: tests/live_gateway/ (compose stack) — here a one-process
uvicorn bound to an ephemeral port, covering what ASGITransport can't:
SSE streaming, /mcp GET with Last-Event-ID resume, and WebSocket JSON-RPC.
"""

import asyncio
import base64
import json
import socket

import httpx
import pytest
import uvicorn

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.transports.http_app import build_app

ADMIN = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


async def _start_server(app, port):
    config = uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error", lifespan="on")
    server = uvicorn.Server(config)
    task = asyncio.create_task(server.serve())
    for _ in range(100):
        if server.started:
            break
        await asyncio.sleep(0.05)
    assert server.started
    return server, task


@pytest.fixture()
def live():
    s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True, gpu_enabled=False)
    engine = GatewayEngine(s)

    async def echo(args):
        return args

    engine.tool_service.register_local_tool("echo", echo)
    app = build_app(engine)
    port = _free_port()
    return app, engine, port


def test_live_sse_pair_and_mcp_stream(live):
    app, engine, port = live
    base = f"http://127.0.0.1:{port}"

    async def go():
        server, task = await _start_server(app, port)
        try:
            async with httpx.AsyncClient(base_url=base, timeout=10.0) as c:
                # --- plain rpc over the socket ---
                r = await c.post("/rpc", headers=ADMIN,
                                 json={"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                                       "params": {"name": "echo", "arguments": {"k": 1}}})
                assert r.json()["result"]["structuredContent"] == {"k": 1}

                # --- legacy SSE pair ---
                srv = engine.registry.create("server", name="live-sse")

                async with c.stream("GET", f"/servers/{srv['id']}/sse", headers=ADMIN) as stream:
                    it = stream.aiter_lines()
                    endpoint_url = None
                    async for line in it:
                        if line.startswith("data:"):
                            endpoint_url = line[5:].strip()
                            break
                    assert endpoint_url and "session_id=" in endpoint_url
                    r = await c.post(endpoint_url, headers=ADMIN,
                                     json={"jsonrpc": "2.0", "id": 9, "method": "ping"})
                    assert r.status_code == 202
                    async for line in it:
                        if line.startswith("data:"):
                            msg = json.loads(line[5:])
                            assert msg["id"] == 9 and msg["result"] == {}
                            break

                # --- streamable /mcp with resume ---
                r = await c.post("/mcp", headers=ADMIN,
                                 json={"jsonrpc": "2.0", "id": 1, "method": "initialize",
                                       "params": {"protocolVersion": "2025-11-25"}})
                sid = r.headers["mcp-session-id"]
                # push two server events, consume stream, then resume after the first
                await engine.sessions.broadcast(sid, {"jsonrpc": "2.0", "method": "notifications/message", "params": {"n": 1}})
                await engine.sessions.broadcast(sid, {"jsonrpc": "2.0", "method": "notifications/message", "params": {"n": 2}})
                seen = []
                first_id = None
                async with c.stream("GET", "/mcp", headers={**ADMIN, "mcp-session-id": sid}) as stream:
                    async for line in stream.aiter_lines():
                        if line.startswith("id:"):
                            if first_id is None:
                                first_id = line[3:].strip()
                        if line.startswith("data:"):
                            seen.append(json.loads(line[5:])["params"]["n"])
                            if len(seen) == 2:
                                break
                assert seen == [1, 2] and first_id
                # resume with Last-Event-ID → replay only the second event
                async with c.stream("GET", "/mcp",
                                    headers={**ADMIN, "mcp-session-id": sid,
                                             "last-event-id": first_id}) as stream:
                    async for line in stream.aiter_lines():
                        if line.startswith("data:"):
                            assert json.loads(line[5:])["params"]["n"] == 2
                            break
        finally:
            server.should_exit = True
            await asyncio.wait_for(task, timeout=10)

    asyncio.run(go())


def test_live_websocket(live):
    app, engine, port = live
    # starlette TestClient drives the ws endpoint in-process (no socket needed)
    from starlette.testclient import TestClient

    with TestClient(app) as tc:
        with tc.websocket_connect("/ws", headers=ADMIN) as ws:
            ws.send_text(json.dumps({"jsonrpc": "2.0", "id": 3, "method": "tools/call",
                                     "params": {"name": "echo", "arguments": {"w": 1}}}))
            out = json.loads(ws.receive_text())
            assert out["result"]["structuredContent"] == {"w": 1}
            ws.send_text(json.dumps({"jsonrpc": "2.0", "id": 4, "method": "ping"}))
            assert json.loads(ws.receive_text())["result"] == {}
