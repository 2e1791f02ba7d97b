"""OAuth client-credentials upstream auth + reverse-proxy tunnel E2E
(reference analogs: oauth_manager flow, mcpgateway/reverse_proxy.py)."""

import asyncio
import base64
import json
import socket

import httpx
import pytest
import uvicorn
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from mcp_context_forge_amd.auth.oauth import ClientCredentialsProvider, OAuthError
from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.transports.http_app import build_app
from mcp_context_forge_amd.transports.reverse_proxy import ReverseProxyClient

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
    return server, task


def test_client_credentials_flow_and_401_retry():
    issued = {"n": 0}
    auth_srv = FastAPI()

    @auth_srv.post("/token")
    async def token(request: Request):
        from urllib.parse import parse_qs

        form = {k: v[0] for k, v in parse_qs((await request.body()).decode()).items()}
        assert form["grant_type"] == "client_credentials"
        if form["client_id"] != "cid" or form["client_secret"] != "sec":
            return JSONResponse({"error": "invalid_client"}, status_code=401)
        issued["n"] += 1
        return {"access_token": f"tok-{issued['n']}", "expires_in": 3600, "token_type": "Bearer"}

    # upstream MCP server that requires the CURRENT token and rejects tok-1
    # after the first call (forces the 401 re-exchange retry)
    upstream = FastAPI()
    state = {"revoked": set()}

    @upstream.post("/mcp")
    async def mcp(request: Request):
        tok = (request.headers.get("authorization") or "").removeprefix("Bearer ")
        if not tok.startswith("tok-") or tok in state["revoked"]:
            return JSONResponse({"detail": "unauthorized"}, status_code=401)
        body = await request.json()
        method = body.get("method")
        rid = body.get("id")
        if method == "initialize":
            return {"jsonrpc": "2.0", "id": rid,
                    "result": {"protocolVersion": "2025-11-25", "capabilities": {},
                               "serverInfo": {"name": "oauth-up", "version": "0"}}}
        if method == "tools/list":
            return {"jsonrpc": "2.0", "id": rid,
                    "result": {"tools": [{"name": "secure_echo", "inputSchema": {"type": "object"}}]}}
        if method == "tools/call":
            # revoke the current token after serving → next call must re-exchange
            state["revoked"].add(tok)
            return {"jsonrpc": "2.0", "id": rid,
                    "result": {"content": [{"type": "text", "text": "ok"}],
                               "structuredContent": {"token_used": tok}, "isError": False}}
        if rid is None:
            return JSONResponse(status_code=202, content=None)
        return {"jsonrpc": "2.0", "id": rid, "result": {}}

    async def go():
        p1, p2 = _free_port(), _free_port()
        s1, t1 = await _serve(auth_srv, p1)
        s2, t2 = await _serve(upstream, p2)
        engine = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False, auth_required=False))
        try:
            # direct provider behavior
            prov = ClientCredentialsProvider(f"http://127.0.0.1:{p1}/token", "cid", "sec")
            tok = await prov.get_token()
            assert tok == "tok-1"
            assert await prov.get_token() == "tok-1"  # cached
            prov.invalidate()
            assert await prov.get_token() == "tok-2"
            bad = ClientCredentialsProvider(f"http://127.0.0.1:{p1}/token", "cid", "wrong")
            with pytest.raises(OAuthError):
                await bad.get_token()

            # gateway federates the oauth-protected upstream
            gw = await engine.gateway_service.register_gateway(
                name="oauth-up", url=f"http://127.0.0.1:{p2}/mcp", auth_type="oauth",
                auth_value=json.dumps({"token_url": f"http://127.0.0.1:{p1}/token",
                                       "client_id": "cid", "client_secret": "sec"}))
            assert gw["status"] == "active"
            out = await engine.handle_rpc_bytes(json.dumps({
                "jsonrpc": "2.0", "id": 1, "method": "tools/call",
                "params": {"name": "oauth-up-secure_echo", "arguments": {}}}).encode())
            used1 = json.loads(out)["result"]["structuredContent"]["token_used"]
            # second call: old token revoked upstream → 401 → re-exchange → success
            out = await engine.handle_rpc_bytes(json.dumps({
                "jsonrpc": "2.0", "id": 2, "method": "tools/call",
                "params": {"name": "oauth-up-secure_echo", "arguments": {}}}).encode())
            used2 = json.loads(out)["result"]["structuredContent"]["token_used"]
            assert used1 != used2
        finally:
            await engine.shutdown()
            s1.should_exit = True
            s2.should_exit = True
            await asyncio.wait_for(t1, timeout=10)
            await asyncio.wait_for(t2, timeout=10)

    asyncio.run(go())


def test_reverse_proxy_tunnel_e2e():
    """Firewalled local engine exposes a tool through the public gateway
    via the SSE+POST tunnel."""

    async def go():
        # public gateway
        gw_engine = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                           auth_required=True, gpu_enabled=False))
        gw_app = build_app(gw_engine)
        port = _free_port()
        server, task = await _serve(gw_app, port)

        # "firewalled" local engine with a tool (never listens on any port)
        local = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                       auth_required=False, plugins_enabled=False))

        async def secret_tool(args):
            return {"from_behind_nat": True, **args}

        local.tool_service.register_local_tool("secret_tool", secret_tool, "NAT'd tool")

        session = local.sessions.create(transport="stdio")

        async def forward(raw: bytes):
            return await local.handle_rpc_bytes(raw, session=session)

        rp = ReverseProxyClient(f"http://127.0.0.1:{port}", "natbox", forward,
                                token=ADMIN["Authorization"])
        try:
            await rp.register()
            serve_task = asyncio.create_task(rp.serve())
            # wait for the gateway to sync the tunneled tools
            for _ in range(100):
                if gw_engine.registry.find("tool", "natbox-secret_tool"):
                    break
                await asyncio.sleep(0.05)
            tool = gw_engine.registry.find("tool", "natbox-secret_tool")
            assert tool is not None, "tunnel sync did not register tools"

            async with httpx.AsyncClient(base_url=f"http://127.0.0.1:{port}", timeout=10) as c:
                r = await c.post("/rpc", headers=ADMIN,
                                 json={"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                                       "params": {"name": "natbox-secret_tool",
                                                  "arguments": {"q": 7}}})
                res = r.json()
                assert res["result"]["structuredContent"] == {"from_behind_nat": True, "q": 7}, res
            serve_task.cancel()
        finally:
            await rp.aclose()
            await local.shutdown()
            server.should_exit = True
            await asyncio.wait_for(task, timeout=10)

    asyncio.run(go())
