"""Server-scoped token enforcement swept across EVERY HTTP route
(VERDICT round-1 item 5: "tests proving server-scoped tokens are enforced
on all routes"; reference: middleware/token_scoping.py).

The sweep enumerates the app's route table programmatically, fills path
parameters with dummies, and asserts: a token scoped to server S may reach
only the data plane (/rpc, /mcp, health/version/well-known) and S's own
endpoints — every other route answers 403 BEFORE any handler logic runs."""

import base64

import httpx
import pytest
from fastapi.routing import APIRoute

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.transports.http_app import build_app

ADMIN = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}


@pytest.fixture()
def scoped_setup():
    from contextlib import asynccontextmanager

    s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True,
                 plugins_enabled=False, gpu_enabled=False)
    engine = GatewayEngine(s)

    async def echo(args):
        return args

    tool = engine.tool_service.register_local_tool("scoped-echo", echo, "t")
    srv = engine.registry.create("server", name="srv-a", associated_tools=[tool["id"]])
    other = engine.registry.create("server", name="srv-b", associated_tools=[])
    app = build_app(engine)
    token = app.state.auth.create_api_token("admin@example.com", "scoped",
                                            server_id=srv["id"])

    @asynccontextmanager
    async def client():
        async with app.router.lifespan_context(app):
            async with httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                         base_url="http://gw") as c:
                yield c

    yield client, engine, app, token, srv, other


def _fill(path: str, sid: str, dummy: str = "zzz-dummy") -> str:
    out = []
    for seg in path.split("/"):
        if seg.startswith("{") and seg.endswith("}"):
            out.append(sid if "server_id" in seg else dummy)
        else:
            out.append(seg)
    return "/".join(out)


ALWAYS_OPEN_PREFIXES = ("/.well-known",)
DATA_PLANE = {"/rpc", "/mcp", "/health", "/healthz", "/version"}


def test_scope_sweep_covers_every_route(scoped_setup, run):
    client, engine, app, token, srv, other = scoped_setup
    hdr = {"Authorization": f"Bearer {token}"}

    routes = [(r.path, sorted(m for m in r.methods if m not in ("HEAD", "OPTIONS")))
              for r in app.routes if isinstance(r, APIRoute)]
    assert len(routes) >= 70, f"route table shrank? {len(routes)}"

    async def go():
        async with client() as c:
            checked = blocked = allowed = public = 0
            for path, methods in routes:
                for method in methods:
                    if method == "GET" and (path.endswith("/sse") or path.endswith("/mcp")
                                            or path == "/mcp" or "stream" in path
                                            or "tunnel" in path or "events" in path):
                        continue  # infinite streams: ASGITransport would block
                    url = _fill(path, srv["id"])
                    body = {} if method in ("POST", "PUT", "PATCH") else None
                    # probe: routes that do not 401 unauthenticated are
                    # public by design (health, login, oauth callback, …) —
                    # the scope gate does not apply to them
                    probe = await c.request(method, url, json=body)
                    if probe.status_code != 401:
                        public += 1
                        continue
                    r = await c.request(method, url, headers=hdr, json=body)
                    checked += 1
                    in_scope = (url in DATA_PLANE or url.startswith(ALWAYS_OPEN_PREFIXES)
                                or url == f"/servers/{srv['id']}"
                                or url.startswith(f"/servers/{srv['id']}/"))
                    if in_scope:
                        assert r.status_code != 403 or "scoped" not in r.text, (method, url, r.text)
                        allowed += 1
                    else:
                        # 403 from the scope gate; a handful of routes run
                        # their own stricter auth and 401 the bearer first —
                        # either way the scoped token never reaches the handler
                        assert r.status_code in (401, 403), (method, url, r.status_code, r.text[:200])
                        if r.status_code == 403:
                            assert "scoped to server" in r.text, (method, url, r.text[:200])
                        blocked += 1
            # the sweep must actually have exercised a broad surface
            assert checked >= 80, (checked, public)
            assert blocked >= 60, (checked, blocked, allowed, public)

    run(go())


def test_scoped_token_data_plane_works_and_filters(scoped_setup, run):
    client, engine, app, token, srv, other = scoped_setup
    hdr = {"Authorization": f"Bearer {token}"}

    async def go():
        async with client() as c:
            # tools/list through /rpc is filtered to the scoped server
            r = await c.post("/rpc", headers=hdr,
                             json={"jsonrpc": "2.0", "id": 1, "method": "tools/list"})
            assert r.status_code == 200
            names = [t["name"] for t in r.json()["result"]["tools"]]
            assert names == ["scoped-echo"]
            # tools/call within scope works
            r = await c.post("/rpc", headers=hdr,
                             json={"jsonrpc": "2.0", "id": 2, "method": "tools/call",
                                   "params": {"name": "scoped-echo", "arguments": {"a": 1}}})
            assert r.status_code == 200 and "result" in r.json()
            # the scoped server's own SSE message endpoint is reachable (404
            # for a bogus session proves we got PAST the scope gate)
            r = await c.post(f"/servers/{srv['id']}/message?session_id=nope", headers=hdr,
                             content=b"{}")
            assert r.status_code == 404
            # another server's surface is NOT
            r = await c.post(f"/servers/{other['id']}/message?session_id=nope", headers=hdr,
                             content=b"{}")
            assert r.status_code == 403
            # registry CRUD is NOT
            r = await c.get("/tools", headers=hdr)
            assert r.status_code == 403
            # an unscoped admin token still has the full surface
            r = await c.get("/tools", headers=ADMIN)
            assert r.status_code == 200

    run(go())
