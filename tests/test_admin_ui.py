"""Admin UI tier (reference analog: tests/playwright/test_admin_ui.py —
here httpx-DOM: fetch shell + partials, assert rendered state, drive the
same REST actions the buttons wire to and assert the re-rendered partial
reflects them)."""

import base64
import json
import re

import httpx
import pytest

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.transports.http_app import build_app

ADMIN = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}


@pytest.fixture()
def ui_client():
    from contextlib import asynccontextmanager

    s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True,
                 plugins_enabled=True, gpu_enabled=False)
    engine = GatewayEngine(s)

    async def echo(args):
        return args

    engine.tool_service.register_local_tool("ui-echo", echo, "Echo for UI tests")
    app = build_app(engine)

    @asynccontextmanager
    async def client():
        async with app.router.lifespan_context(app):
            async with httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                         base_url="http://gw", headers=ADMIN) as c:
                yield c

    yield client, engine, app


def test_shell_and_all_partials_render(ui_client, run):
    from mcp_context_forge_amd.admin.ui import TABS

    client, engine, app = ui_client

    async def go():
        async with client() as c:
            r = await c.get("/admin")
            assert r.status_code == 200
            assert "loadTab" in r.text and "hx-get" in r.text or "bindHx" in r.text
            for tab in TABS:
                rp = await c.get(f"/admin/ui/{tab}")
                assert rp.status_code == 200, (tab, rp.text[:200])
                assert "<" in rp.text  # HTML fragment, not JSON
            r404 = await c.get("/admin/ui/bogus")
            assert r404.status_code == 404

    run(go())


def test_partials_require_auth(ui_client, run):
    client, engine, app = ui_client

    async def go():
        async with client() as c:
            r = await c.get("/admin/ui/dashboard", headers={"Authorization": ""})
            assert r.status_code == 401

    run(go())


def test_tools_partial_actions_roundtrip(ui_client, run):
    client, engine, app = ui_client

    async def go():
        async with client() as c:
            rp = await c.get("/admin/ui/tools")
            assert "ui-echo" in rp.text
            assert "hx-post" in rp.text and "toggle" in rp.text
            tid = engine.registry.find("tool", "ui-echo")["id"]
            # drive the exact URL the disable button carries
            m = re.search(r"hx-post='(/tools/[^']+/toggle\?activate=false)'", rp.text)
            assert m, rp.text[:500]
            r2 = await c.post(m.group(1))
            assert r2.status_code == 200
            rp2 = await c.get("/admin/ui/tools")
            # disabled pill now rendered for the tool row
            assert re.search(r"ui-echo.*?pill bad", rp2.text, re.S)
            await c.post(f"/tools/{tid}/toggle?activate=true")

    run(go())


def test_gateway_create_form_and_lifecycle_column(ui_client, run):
    client, engine, app = ui_client

    async def go():
        async with client() as c:
            rp = await c.get("/admin/ui/gateways")
            assert "Register gateway" in rp.text and "hx-post='/gateways'" in rp.text
            # the form posts defer=true → pending row appears with status pill
            r = await c.post("/gateways", json={"name": "ui-peer", "url": "http://127.0.0.1:1/mcp",
                                                "defer": True})
            assert r.status_code == 201
            rp2 = await c.get("/admin/ui/gateways")
            assert "ui-peer" in rp2.text
            assert "pill" in rp2.text

    run(go())


def test_plugins_partial_mode_flip(ui_client, run):
    client, engine, app = ui_client

    async def go():
        async with client() as c:
            rp = await c.get("/admin/ui/plugins")
            assert "deny_filter" in rp.text
            r = await c.post("/admin/plugins/deny_filter/mode?mode=permissive")
            assert r.status_code == 200
            rp2 = await c.get("/admin/ui/plugins")
            assert re.search(r"deny_filter.*?permissive", rp2.text, re.S)
            await c.post("/admin/plugins/deny_filter/mode?mode=enforce")

    run(go())


def test_bindings_partial_bind_unbind(ui_client, run):
    client, engine, app = ui_client

    async def go():
        async with client() as c:
            r = await c.post("/admin/ui/bind", json={"tool_name": "ui-echo",
                                                     "plugin_name": "deny_filter",
                                                     "mode": "disabled"})
            assert r.status_code == 200
            rp = await c.get("/admin/ui/bindings")
            assert "ui-echo" in rp.text and "deny_filter" in rp.text
            r2 = await c.delete("/tools/ui-echo/plugin-bindings/deny_filter")
            assert r2.status_code == 204
            rp2 = await c.get("/admin/ui/bindings")
            assert "ui-echo" not in rp2.text

    run(go())


def test_metrics_logs_traces_audit_partials_show_activity(ui_client, run):
    client, engine, app = ui_client

    async def go():
        async with client() as c:
            # generate some traffic first
            for i in range(3):
                r = await c.post("/rpc", json={"jsonrpc": "2.0", "id": i, "method": "tools/call",
                                               "params": {"name": "ui-echo",
                                                          "arguments": {"x": i}}})
                assert r.status_code == 200
            rp = await c.get("/admin/ui/metrics")
            assert "rollups" in rp.text.lower()
            rp = await c.get("/admin/ui/traces")
            assert "tools/call" in rp.text or "none" in rp.text
            # an audited action
            await c.post("/gateways", json={"name": "aud-peer", "url": "x", "defer": True})
            rp = await c.get("/admin/ui/audit")
            assert "create" in rp.text or "none" in rp.text
            rp = await c.get("/admin/ui/logs?q=&level=ERROR")
            assert rp.status_code == 200

    run(go())


def test_runtime_partial_apply(ui_client, run):
    client, engine, app = ui_client

    async def go():
        async with client() as c:
            rp = await c.get("/admin/ui/runtime")
            assert "Runtime" in rp.text and "rank0-by-construction" in rp.text
            r = await c.post("/admin/ui/runtime", json={"window_us": 750})
            assert r.status_code == 200

    run(go())


def test_tokens_partial_revoke_flow(ui_client, run):
    client, engine, app = ui_client

    async def go():
        async with client() as c:
            app.state.auth.create_api_token("admin@example.com", "ui-token")
            rp = await c.get("/admin/ui/tokens")
            assert "ui-token" in rp.text
            m = re.search(r"hx-delete='(/tokens/[^']+)'", rp.text)
            assert m
            r = await c.delete(m.group(1))
            assert r.status_code == 204
            rp2 = await c.get("/admin/ui/tokens")
            assert "revoked" in rp2.text

    run(go())
