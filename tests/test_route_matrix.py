"""Route-matrix sweep (VERDICT round-1 item 9: per-route CRUD/RBAC/error
coverage, reference analog: the unit-test mass over main.py routers).

Parametrized across every registry kind and representative error cases:
404s, 409 conflicts, 422 validation, RBAC denial for a viewer-role user,
unauthenticated 401s, toggle/pagination behavior, and the JSON-RPC method
error matrix."""

import base64
import json

import httpx
import pytest

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.transports.http_app import build_app

ADMIN = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}

KINDS = [
    ("tools", {"name": "mx-tool", "integration_type": "LOCAL"}),
    ("gateways", {"name": "mx-gw", "url": "http://127.0.0.1:1/mcp", "defer": True}),
    ("servers", {"name": "mx-server"}),
    ("resources", {"uri": "res://mx/1", "name": "mx-res", "content": "hello"}),
    ("prompts", {"name": "mx-prompt", "template": "Hi {{name}}"}),
    ("a2a", {"name": "mx-agent", "endpoint_url": "inproc://mx-agent"}),
]


@pytest.fixture()
def matrix_app():
    from contextlib import asynccontextmanager

    s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True,
                 plugins_enabled=False, gpu_enabled=False)
    engine = GatewayEngine(s)

    async def echo(args):
        return args

    engine.tool_service.register_local_tool("matrix-echo", echo, "t")
    app = build_app(engine)
    # a viewer-role (read-only) user
    app.state.auth.create_user("viewer@x.io", "viewpass", "Viewer")
    viewer = {"Authorization": "Basic " + base64.b64encode(b"viewer@x.io:viewpass").decode()}

    @asynccontextmanager
    async def client():
        async with app.router.lifespan_context(app):
            async with httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                         base_url="http://gw") as c:
                yield c

    return client, engine, app, viewer


# ---------------------------------------------------------------- CRUD errors


@pytest.mark.parametrize("plural,body", KINDS)
def test_get_unknown_404(matrix_app, run, plural, body):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.get(f"/{plural}/does-not-exist", headers=ADMIN)
            assert r.status_code == 404

    run(go())


@pytest.mark.parametrize("plural,body", KINDS)
def test_update_unknown_404(matrix_app, run, plural, body):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.put(f"/{plural}/does-not-exist", headers=ADMIN,
                            json={"description": "x"})
            assert r.status_code == 404

    run(go())


@pytest.mark.parametrize("plural,body", KINDS)
def test_delete_unknown_404(matrix_app, run, plural, body):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.delete(f"/{plural}/does-not-exist", headers=ADMIN)
            assert r.status_code == 404

    run(go())


@pytest.mark.parametrize("plural,body", KINDS)
def test_create_roundtrip_conflict_toggle_delete(matrix_app, run, plural, body):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            body2 = {**body}
            for k in ("name", "uri"):
                if k in body2:
                    body2[k] = body2[k] + "-rt"
            r = await c.post(f"/{plural}", headers=ADMIN, json=body2)
            assert r.status_code == 201, (plural, r.text)
            ent = r.json()
            # duplicate name → 409
            r2 = await c.post(f"/{plural}", headers=ADMIN, json=body2)
            assert r2.status_code == 409, (plural, r2.status_code, r2.text)
            # toggle off and on
            r3 = await c.post(f"/{plural}/{ent['id']}/toggle?activate=false", headers=ADMIN)
            assert r3.status_code == 200 and r3.json()["enabled"] is False
            r4 = await c.post(f"/{plural}/{ent['id']}/toggle?activate=true", headers=ADMIN)
            assert r4.json()["enabled"] is True
            # delete → gone
            r5 = await c.delete(f"/{plural}/{ent['id']}", headers=ADMIN)
            assert r5.status_code == 204
            r6 = await c.get(f"/{plural}/{ent['id']}", headers=ADMIN)
            assert r6.status_code == 404

    run(go())


@pytest.mark.parametrize("plural,body", KINDS)
def test_list_and_pagination(matrix_app, run, plural, body):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.get(f"/{plural}", headers=ADMIN)
            assert r.status_code == 200 and isinstance(r.json(), list)
            r = await c.get(f"/{plural}?limit=1&cursor=0", headers=ADMIN)
            assert r.status_code == 200
            page = r.json()
            assert "items" in page and "nextCursor" in page
            assert "X-Total-Count" in r.headers

    run(go())


# ---------------------------------------------------------------- RBAC


@pytest.mark.parametrize("plural,body", KINDS)
def test_viewer_cannot_mutate(matrix_app, run, plural, body):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.post(f"/{plural}", headers=viewer, json={**body, "name": "nope"})
            assert r.status_code == 403, (plural, r.status_code)
            r = await c.delete(f"/{plural}/anything", headers=viewer)
            assert r.status_code == 403

    run(go())


@pytest.mark.parametrize("plural,body", KINDS)
def test_viewer_read_matches_role(matrix_app, run, plural, body):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.get(f"/{plural}", headers=viewer)
            if plural == "gateways":
                # federation config is not in the viewer role (DEFAULT_PERMISSIONS)
                assert r.status_code == 403, (plural, r.status_code)
            else:
                assert r.status_code == 200, (plural, r.status_code, r.text[:200])

    run(go())


@pytest.mark.parametrize("path", ["/admin/stats", "/admin/logs", "/admin/traces",
                                  "/admin/audit", "/admin/metrics/rollups",
                                  "/admin/plugins", "/admin/runtime", "/admin"])
def test_admin_surface_requires_admin(matrix_app, run, path):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.get(path, headers=viewer)
            assert r.status_code == 403, (path, r.status_code)
            r = await c.get(path)
            assert r.status_code == 401
            r = await c.get(path, headers=ADMIN)
            assert r.status_code == 200

    run(go())


# ---------------------------------------------------------------- RPC matrix


RPC_ERRORS = [
    ({"jsonrpc": "2.0", "id": 1, "method": "no/such/method"}, -32601),
    ({"jsonrpc": "2.0", "id": 2, "method": "tools/call", "params": {}}, -32602),
    ({"jsonrpc": "2.0", "id": 3, "method": "tools/call",
      "params": {"name": "ghost-tool", "arguments": {}}}, -32602),
    ({"jsonrpc": "2.0", "id": 4, "method": "prompts/get", "params": {}}, -32602),
    ({"jsonrpc": "2.0", "id": 5, "method": "resources/read", "params": {}}, -32602),
    ({"jsonrpc": "1.0", "id": 6, "method": "ping"}, -32600),
    ({"id": 7, "method": "ping"}, -32600),
]


@pytest.mark.parametrize("body,code", RPC_ERRORS)
def test_rpc_error_matrix(matrix_app, run, body, code):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.post("/rpc", headers=ADMIN, json=body)
            assert r.status_code == 200
            assert r.json()["error"]["code"] == code, r.json()

    run(go())


@pytest.mark.parametrize("method,expect_key", [
    ("tools/list", "tools"), ("resources/list", "resources"),
    ("prompts/list", "prompts"), ("resources/templates/list", "resourceTemplates"),
    ("roots/list", "roots"),
])
def test_rpc_list_methods(matrix_app, run, method, expect_key):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.post("/rpc", headers=ADMIN,
                             json={"jsonrpc": "2.0", "id": 1, "method": method})
            assert r.status_code == 200
            assert expect_key in r.json()["result"]

    run(go())


def test_rpc_parse_error_and_notification(matrix_app, run):
    client, engine, app, viewer = matrix_app

    async def go():
        async with client() as c:
            r = await c.post("/rpc", headers=ADMIN, content=b"{nope")
            assert r.json()["error"]["code"] == -32700
            r = await c.post("/rpc", headers=ADMIN,
                             json={"jsonrpc": "2.0", "method": "notifications/initialized"})
            assert r.status_code == 202

    run(go())
