"""Native C++ edge (ops/csrc/edge.cpp + transports/native_edge.py) over real
TCP sockets — CPU tier (no GPU: the engine runs the per-request reference
path; the socket loop, HTTP parsing, auth memo and batch plumbing are
identical to the GPU deployment)."""

import asyncio
import base64
import json
import socket
import subprocess
import sys

import httpx
import pytest

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.transports.http_app import build_app

BASIC = "Basic " + base64.b64encode(b"admin:changeme").decode()


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture()
def edge_app():
    from contextlib import asynccontextmanager

    port = free_port()
    s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True,
                 plugins_enabled=True, gpu_enabled=False, native_edge_port=port,
                 native_edge_threads=2, max_request_body_bytes=64 * 1024)
    engine = GatewayEngine(s)

    async def echo(args):
        return args

    engine.tool_service.register_local_tool("echo", echo, "Echo tool")
    app = build_app(engine)

    @asynccontextmanager
    async def running():
        async with app.router.lifespan_context(app):
            yield f"http://127.0.0.1:{port}"

    yield running, engine, app


def _rpc(i, name="echo", args=None):
    return {"jsonrpc": "2.0", "id": i, "method": "tools/call",
            "params": {"name": name, "arguments": args or {"msg": f"m{i}"}}}


def test_edge_rpc_roundtrip_and_auth(edge_app, run):
    running, engine, app = edge_app

    async def go():
        async with running() as base:
            async with httpx.AsyncClient(base_url=base, timeout=10.0) as c:
                # no auth → 401 straight from C++
                r = await c.post("/rpc", json=_rpc(1))
                assert r.status_code == 401
                # authenticated ping
                r = await c.post("/rpc", json={"jsonrpc": "2.0", "id": 1, "method": "ping"},
                                 headers={"Authorization": BASIC})
                assert r.status_code == 200 and r.json()["result"] == {}
                # tools/call roundtrip; repeat to hit the C++ auth cache
                for i in range(5):
                    r = await c.post("/rpc", json=_rpc(i + 2), headers={"Authorization": BASIC})
                    assert r.status_code == 200
                    assert r.json()["result"]["content"][0]["text"]
                ne = app.state.native_edge
                st = ne.stats()
                assert st["hot"] >= 6
                assert st["direct_401"] >= 1

    run(go())


def test_edge_health_and_cold_path(edge_app, run):
    running, engine, app = edge_app

    async def go():
        async with running() as base:
            async with httpx.AsyncClient(base_url=base, timeout=10.0) as c:
                r = await c.get("/healthz")   # served inline by C++
                assert r.status_code == 200 and r.json()["edge"] == "native"
                # cold path: control-plane GET through the ASGI app
                r = await c.get("/version", headers={"Authorization": BASIC})
                assert r.status_code == 200
                assert r.json()["name"] == "mcp-context-forge-amd"
                # cold POST too (tools CRUD)
                r = await c.get("/tools", headers={"Authorization": BASIC})
                assert r.status_code == 200

    run(go())


def test_edge_oversize_body_and_keepalive(edge_app, run):
    running, engine, app = edge_app

    async def go():
        async with running() as base:
            async with httpx.AsyncClient(base_url=base, timeout=10.0) as c:
                big = _rpc(1, args={"pad": "x" * (128 * 1024)})
                r = await c.post("/rpc", json=big, headers={"Authorization": BASIC})
                assert r.status_code == 413
            # keep-alive: many requests over ONE raw connection
            reader, writer = await asyncio.open_connection("127.0.0.1", int(base.rsplit(":", 1)[1]))
            for i in range(10):
                body = json.dumps(_rpc(i)).encode()
                req = (f"POST /rpc HTTP/1.1\r\nHost: x\r\nAuthorization: {BASIC}\r\n"
                       f"Content-Type: application/json\r\nContent-Length: {len(body)}\r\n\r\n"
                       ).encode() + body
                writer.write(req)
                await writer.drain()
                # read status line + headers
                head = await reader.readuntil(b"\r\n\r\n")
                assert b"200 OK" in head.split(b"\r\n")[0]
                cl = [ln for ln in head.split(b"\r\n") if ln.lower().startswith(b"content-length")]
                n = int(cl[0].split(b":")[1])
                payload = await reader.readexactly(n)
                assert json.loads(payload)["id"] == i
            writer.close()

    run(go())


def test_edge_revocation_flushes_native_cache(edge_app, run):
    running, engine, app = edge_app

    async def go():
        async with running() as base:
            async with httpx.AsyncClient(base_url=base, timeout=10.0) as c:
                raw = app.state.auth.create_api_token("admin@example.com", "edge-tok")
                hdr = {"Authorization": f"Bearer {raw}"}
                body = {"jsonrpc": "2.0", "id": 1, "method": "ping"}
                r = await c.post("/rpc", json=body, headers=hdr)
                assert r.status_code == 200
                r = await c.post("/rpc", json=body, headers=hdr)  # C++ cache hit
                assert r.status_code == 200
                tid = app.state.auth.list_api_tokens("admin@example.com")[-1]["id"]
                assert app.state.auth.revoke_api_token(tid)
                # the edge loop flushes the native cache on the next batch;
                # poll until the revocation lands (bounded)
                for _ in range(50):
                    r = await c.post("/rpc", json=body, headers=hdr)
                    if r.status_code == 401:
                        break
                    await asyncio.sleep(0.02)
                assert r.status_code == 401

    run(go())


def test_edge_under_forge_hey(edge_app, run):
    """Drive the native edge with the C++ load generator (the bench rig)."""
    from mcp_context_forge_amd.ops.build import HEY, build_hey

    build_hey(verbose=False)
    running, engine, app = edge_app

    async def go():
        async with running() as base:
            port = base.rsplit(":", 1)[1]
            payload = json.dumps(_rpc(7))
            proc = await asyncio.create_subprocess_exec(
                str(HEY), "--host", "127.0.0.1", "--port", port, "--path", "/rpc",
                "--connections", "8", "--threads", "2",
                "--requests-per-step", "200", "--warmup", "1", "--steps", "2",
                "--payload", payload, "--auth", BASIC,
                stdin=asyncio.subprocess.PIPE, stdout=asyncio.subprocess.PIPE)
            line = await asyncio.wait_for(proc.stdout.readline(), timeout=30)
            assert line.strip() == b"WARM"
            proc.stdin.write(b"GO\n")
            await proc.stdin.drain()
            out = await asyncio.wait_for(proc.stdout.readline(), timeout=60)
            await proc.wait()
            res = json.loads(out)
            assert res["requests"] == 400
            assert res["errors"] == 0
            assert res["non200"] == 0
            assert res["rps"] > 0
            assert res["p50_ms"] > 0

    run(go())


@pytest.mark.gpu
def test_edge_with_gpu_pipeline_under_load(run):
    """GPU tier: the C++ edge feeding the GPU batch pipeline end to end,
    driven by the native load generator — the flagship serving shape."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no ROCm device")
    from mcp_context_forge_amd.ops.build import HEY, build_hey

    build_hey(verbose=False)
    port = free_port()
    s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True,
                 plugins_enabled=True, gpu_enabled=True, native_edge_port=port,
                 native_edge_threads=4)
    engine = GatewayEngine(s)

    async def go():
        from mcp_context_forge_amd.services.upstream import NativeInProcUpstream

        await engine.gateway_service.register_gateway(
            name="nt", url="inproc://nt", client=NativeInProcUpstream(), owner_rank=0)
        app = build_app(engine)
        async with app.router.lifespan_context(app):
            assert engine.gpu_pipeline is not None, "GPU pipeline must attach"
            tok = app.state.auth.create_api_token("admin@example.com", "edge-gpu")
            payload = json.dumps({"jsonrpc": "2.0", "id": 1, "method": "tools/call",
                                  "params": {"name": "nt-convert_time",
                                             "arguments": {"time": "2026-01-01T00:00:00Z",
                                                           "source_timezone": "UTC",
                                                           "target_timezone": "UTC"}}})
            proc = await asyncio.create_subprocess_exec(
                str(HEY), "--host", "127.0.0.1", "--port", str(port), "--path", "/rpc",
                "--connections", "64", "--threads", "2",
                "--requests-per-step", "2000", "--warmup", "1", "--steps", "2",
                "--payload", payload, "--auth", f"Bearer {tok}",
                stdin=asyncio.subprocess.PIPE, stdout=asyncio.subprocess.PIPE)
            line = await asyncio.wait_for(proc.stdout.readline(), timeout=60)
            assert line.strip() == b"WARM"
            proc.stdin.write(b"GO\n")
            await proc.stdin.drain()
            out = await asyncio.wait_for(proc.stdout.readline(), timeout=120)
            await proc.wait()
            res = json.loads(out)
            assert res["errors"] == 0 and res["non200"] == 0, res
            assert res["requests"] == 4000
            st = engine.gpu_pipeline.stats()
            assert st["requests"] >= 4000  # batches actually rode the GPU path

    run(go())


def test_edge_adversarial_wire_forms(edge_app, run):
    """Raw-socket adversarial input against the C++ parser: split headers
    across packets, two pipelined requests in one send, and garbage bytes
    — the edge must answer correctly or close cleanly, never hang."""
    running, engine, app = edge_app

    async def go():
        async with running() as base:
            port = int(base.rsplit(":", 1)[1])
            body = json.dumps(_rpc(1)).encode()

            def talk(chunks, expect_json=True, timeout=5.0):
                with socket.create_connection(("127.0.0.1", port), timeout=timeout) as c:
                    c.settimeout(timeout)
                    for ch in chunks:
                        c.sendall(ch)
                    data = b""
                    try:
                        while b"\r\n\r\n" not in data or (expect_json and not data.endswith(b"}")):
                            part = c.recv(65536)
                            if not part:
                                break
                            data += part
                    except socket.timeout:
                        pass
                    return data

            req = (f"POST /rpc HTTP/1.1\r\nHost: x\r\nAuthorization: {BASIC}\r\n"
                   f"Content-Type: application/json\r\nContent-Length: {len(body)}\r\n"
                   f"\r\n").encode() + body

            # (a) header split into tiny packets
            out = await asyncio.to_thread(talk, [req[:17], req[17:60], req[60:]])
            assert b"200" in out.split(b"\r\n", 1)[0] and b'"result"' in out, out[:200]

            # (b) two pipelined requests in ONE send → two responses
            def talk2():
                with socket.create_connection(("127.0.0.1", port), timeout=5.0) as c:
                    c.settimeout(5.0)
                    c.sendall(req + req)
                    data = b""
                    try:
                        while data.count(b'"result"') < 2:
                            part = c.recv(65536)
                            if not part:
                                break
                            data += part
                    except socket.timeout:
                        pass
                    return data

            out2 = await asyncio.to_thread(talk2)
            assert out2.count(b'"result"') == 2, out2[:300]

            # (c) garbage request line → error response or clean close, no hang
            out3 = await asyncio.to_thread(talk, [b"\x00\xffGARBAGE\r\n\r\n"], False)
            assert out3 == b"" or b"400" in out3 or b"HTTP/1.1" in out3

            # (d) the edge still serves normal traffic afterwards
            out4 = await asyncio.to_thread(talk, [req])
            assert b'"result"' in out4

    run(go())
