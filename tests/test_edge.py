"""Multi-worker HTTP edge (reference analog: gunicorn workers + sidecar):
owner app + unix-socket batch server + a worker shell forwarding /rpc and
proxying the control plane."""

import asyncio
import base64
import json
import socket

import httpx
import pytest
import uvicorn

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.transports.edge import build_worker_app
from mcp_context_forge_amd.transports.http_app import build_app

ADMIN = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


async def _serve(app, port):
    config = uvicorn.Config(app, host="127.0.0.1", port=port, log_level="error", lifespan="on")
    server = uvicorn.Server(config)
    task = asyncio.create_task(server.serve())
    for _ in range(100):
        if server.started:
            break
        await asyncio.sleep(0.05)
    assert server.started
    return server, task


def test_edge_worker_forwarding(tmp_path):
    async def go():
        sock_path = str(tmp_path / "edge.sock")
        owner_port = _free_port()
        worker_port = _free_port()

        engine = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                        auth_required=True, gpu_enabled=False,
                                        edge_socket=sock_path))

        async def echo(args):
            return args

        engine.tool_service.register_local_tool("echo", echo)
        owner_app = build_app(engine)
        owner_srv, owner_task = await _serve(owner_app, owner_port)

        worker_app = build_worker_app(sock_path, f"http://127.0.0.1:{owner_port}")
        worker_srv, worker_task = await _serve(worker_app, worker_port)
        try:
            async with httpx.AsyncClient(base_url=f"http://127.0.0.1:{worker_port}", timeout=15) as c:
                # hot path over the unix socket
                r = await c.post("/rpc", headers=ADMIN,
                                 json={"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                                       "params": {"name": "echo", "arguments": {"w": 1}}})
                assert r.status_code == 200, r.text
                assert r.json()["result"]["structuredContent"] == {"w": 1}
                # unauthenticated rejected at the worker
                r = await c.post("/rpc", json={"jsonrpc": "2.0", "id": 2, "method": "ping"})
                assert r.status_code == 401
                # notification → 202 through the frame protocol
                r = await c.post("/rpc", headers=ADMIN,
                                 json={"jsonrpc": "2.0", "method": "notifications/initialized"})
                assert r.status_code == 202
                # control plane proxied to the owner
                r = await c.get("/tools", headers=ADMIN)
                assert r.status_code == 200 and r.json()[0]["name"] == "echo"
                r = await c.get("/healthz")
                assert r.json()["role"] == "worker"
                # burst through the frame path (coalescing)
                outs = await asyncio.gather(*(
                    c.post("/rpc", headers=ADMIN,
                           json={"jsonrpc": "2.0", "id": i, "method": "tools/call",
                                 "params": {"name": "echo", "arguments": {"i": i}}})
                    for i in range(50)))
                assert all(o.status_code == 200 for o in outs)
                got = sorted(o.json()["result"]["structuredContent"]["i"] for o in outs)
                assert got == list(range(50))
            assert owner_app.state.owner_server.requests >= 52
        finally:
            worker_srv.should_exit = True
            await asyncio.wait_for(worker_task, timeout=10)
            owner_srv.should_exit = True
            await asyncio.wait_for(owner_task, timeout=10)

    asyncio.run(go())
