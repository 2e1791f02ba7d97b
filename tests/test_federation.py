"""Federation E2E: gateway federating another live gateway over streamable
HTTP (reference analog: gateway_service tests + live_gateway tier; the
upstream here is a second instance of THIS gateway — self-federation)."""

import asyncio
import base64
import json
import socket

import httpx
import pytest
import uvicorn

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.services.upstream import HttpUpstreamClient, UpstreamError, make_fake_time_upstream
from mcp_context_forge_amd.transports.http_app import build_app

ADMIN = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}


def _free_port() -> int:
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


def test_gateway_federates_live_upstream():
    async def go():
        # upstream gateway: auth off for its /mcp (the peer connects anonymously)
        up_engine = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                           auth_required=False, gpu_enabled=False))

        async def weather(args):
            return {"temp_c": 21, "city": args.get("city", "?")}

        up_engine.tool_service.register_local_tool(
            "weather", weather, "Weather lookup",
            input_schema={"type": "object", "properties": {"city": {"type": "string"}}, "required": ["city"]})
        up_engine.registry.create("resource", uri="mem://up/info", name="info", content="upstream resource")
        up_engine.registry.create("prompt", name="greet", template="Hi {{ name }}")
        up_app = build_app(up_engine)
        up_port = _free_port()
        up_server, up_task = await _serve(up_app, up_port)

        main = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                      auth_required=False, gpu_enabled=False,
                                      health_check_timeout=2, unhealthy_threshold=1))
        try:
            gw = await main.gateway_service.register_gateway(
                name="peer-one", url=f"http://127.0.0.1:{up_port}/mcp")
            assert gw["status"] == "active" and gw["reachable"]
            # tools synced with qualified names (reference: _update_or_create_tools :5648)
            tool = main.registry.find("tool", "peer-one-weather")
            assert tool is not None and tool["integration_type"] == "MCP"
            assert tool["input_schema"]["required"] == ["city"]
            # resource + prompt synced
            assert main.registry.find("resource", "mem://up/info") is not None
            assert main.registry.find("prompt", "peer-one-greet") is not None

            # invoke through the full engine path
            out = await main.handle_rpc_bytes(json.dumps({
                "jsonrpc": "2.0", "id": 1, "method": "tools/call",
                "params": {"name": "peer-one-weather", "arguments": {"city": "Berlin"}}}).encode())
            res = json.loads(out)
            assert res["result"]["structuredContent"] == {"temp_c": 21, "city": "Berlin"}

            # schema enforced against the synced upstream schema
            out = await main.handle_rpc_bytes(json.dumps({
                "jsonrpc": "2.0", "id": 2, "method": "tools/call",
                "params": {"name": "peer-one-weather", "arguments": {}}}).encode())
            assert json.loads(out)["error"]["code"] == -32003

            # health check green
            results = await main.gateway_service.check_health_once()
            assert results[gw["id"]] is True

            # upstream goes away → gateway + tools marked unreachable
            up_server.should_exit = True
            await asyncio.wait_for(up_task, timeout=10)
            results = await main.gateway_service.check_health_once()
            assert results[gw["id"]] is False
            gw2 = main.registry.get("gateway", gw["id"])
            assert gw2["reachable"] is False and gw2["status"] == "unreachable"
            tool2 = main.registry.find("tool", "peer-one-weather")
            assert tool2["reachable"] is False
            out = await main.handle_rpc_bytes(json.dumps({
                "jsonrpc": "2.0", "id": 3, "method": "tools/call",
                "params": {"name": "peer-one-weather", "arguments": {"city": "X"}}}).encode())
            assert json.loads(out)["error"]["code"] == -32002

            # upstream returns → reactivation on next health pass
            up_server2, up_task2 = await _serve(up_app, up_port)
            try:
                results = await main.gateway_service.check_health_once()
                assert results[gw["id"]] is True
                assert main.registry.get("gateway", gw["id"])["reachable"] is True
            finally:
                up_server2.should_exit = True
                await asyncio.wait_for(up_task2, timeout=10)
        finally:
            await main.shutdown()
            await up_engine.shutdown()

    asyncio.run(go())


def test_register_unreachable_gateway_fails():
    async def go():
        main = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                      auth_required=False, federation_timeout=2,
                                      federation_sync_timeout=3))
        from mcp_context_forge_amd.services.gateway_service import GatewayConnectionError

        with pytest.raises(GatewayConnectionError):
            await main.gateway_service.register_gateway(name="ghost", url="http://127.0.0.1:1/mcp")
        gw = main.registry.find("gateway", "ghost")
        assert gw is not None and gw["status"] == "unreachable"
        await main.shutdown()

    asyncio.run(go())


def test_gateway_refresh_picks_up_new_tools(run):
    async def go():
        main = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                      auth_required=False))
        up = make_fake_time_upstream("dyn")
        gw = await main.gateway_service.register_gateway(name="dyn", url="inproc://dyn", client=up)
        assert main.registry.find("tool", "dyn-convert_time") is not None

        async def extra(args):
            return {"ok": True}

        up.add_tool("extra_tool", extra, "added later")
        await main.gateway_service.refresh_gateway(gw["id"])
        assert main.registry.find("tool", "dyn-extra_tool") is not None
        # removing a tool upstream prunes it locally
        del up._tools["extra_tool"]
        await main.gateway_service.refresh_gateway(gw["id"])
        assert main.registry.find("tool", "dyn-extra_tool") is None
        await main.shutdown()

    run(go())


def test_delete_gateway_removes_tools(run):
    async def go():
        main = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False, auth_required=False))
        up = make_fake_time_upstream("gone")
        gw = await main.gateway_service.register_gateway(name="gone", url="inproc://gone", client=up)
        assert main.registry.find("tool", "gone-echo") is not None
        await main.gateway_service.delete_gateway(gw["id"])
        assert main.registry.find("tool", "gone-echo") is None
        assert main.registry.find("gateway", "gone") is None
        await main.shutdown()

    run(go())
