"""stdio transport + CLI bridges (reference analogs: stdio_transport,
wrapper.py, translate.py) via real subprocess pipes."""

import asyncio
import json
import subprocess
import sys

import pytest

STDIO_ENV = {"DATABASE_URL": "sqlite://", "FORGE_FEDERATION_ENABLED": "false",
             "FORGE_AUTH_REQUIRED": "false", "FORGE_PLUGINS_ENABLED": "false",
             "PYTHONPATH": "/root/repo"}


def test_stdio_server_subprocess():
    import os

    env = dict(os.environ, **STDIO_ENV)
    proc = subprocess.Popen([sys.executable, "-m", "mcp_context_forge_amd", "stdio"],
                            stdin=subprocess.PIPE, stdout=subprocess.PIPE, env=env, cwd="/root/repo")
    try:
        def rpc(obj):
            proc.stdin.write(json.dumps(obj).encode() + b"\n")
            proc.stdin.flush()
            line = proc.stdout.readline()
            return json.loads(line) if line else None

        out = rpc({"jsonrpc": "2.0", "id": 1, "method": "initialize",
                   "params": {"protocolVersion": "2025-11-25"}})
        assert out["result"]["serverInfo"]["name"] == "mcp-context-forge-amd"
        out = rpc({"jsonrpc": "2.0", "id": 2, "method": "tools/list"})
        assert out["result"]["tools"] == []
        out = rpc({"jsonrpc": "2.0", "id": 3, "method": "ping"})
        assert out["result"] == {}
    finally:
        proc.stdin.close()
        proc.wait(timeout=10)


def test_stdio_subprocess_manager_and_translate_app(run):
    """StdIOEndpoint analog drives a child stdio MCP server; translate app
    bridges it to HTTP."""
    import os

    from mcp_context_forge_amd.transports.stdio import StdioSubprocess, build_translate_app

    async def go():
        env_args = [sys.executable, "-m", "mcp_context_forge_amd", "stdio"]
        os.environ.update(STDIO_ENV)
        sp = StdioSubprocess(env_args)
        await sp.start()
        try:
            out = await sp.request("initialize", {"protocolVersion": "2025-11-25"})
            assert out["result"]["protocolVersion"] == "2025-11-25"
            out = await sp.request("ping")
            assert out["result"] == {}

            import httpx

            app = build_translate_app(sp)
            transport = httpx.ASGITransport(app=app)
            async with httpx.AsyncClient(transport=transport, base_url="http://t") as c:
                r = await c.post("/mcp", json={"jsonrpc": "2.0", "id": 9, "method": "tools/list"})
                assert r.json()["result"]["tools"] == []
                r = await c.post("/mcp", json={"jsonrpc": "2.0", "method": "notifications/initialized"})
                assert r.status_code == 202
        finally:
            await sp.stop()

    run(go())


def test_cli_token_mint():
    import os

    env = dict(os.environ, PYTHONPATH="/root/repo")
    out = subprocess.run([sys.executable, "-m", "mcp_context_forge_amd", "token",
                          "--user", "x@y.com", "--admin"],
                         capture_output=True, text=True, env=env, cwd="/root/repo", timeout=60)
    assert out.returncode == 0
    token = out.stdout.strip()
    from mcp_context_forge_amd.auth import jwt as jwt_mod
    from mcp_context_forge_amd.config import Settings

    claims = jwt_mod.decode_token(token, Settings().jwt_secret_key)
    assert claims["sub"] == "x@y.com" and claims["admin"] is True


def test_plugin_scaffold_cli(tmp_path):
    """cforge-analog plugin bootstrap (reference: tools/cli.py): generated
    skeletons import cleanly and register as plugins."""
    import importlib.util
    import subprocess
    import sys

    env = {"PYTHONPATH": "/root/repo", "PATH": "/usr/bin:/bin"}
    r = subprocess.run([sys.executable, "-m", "mcp_context_forge_amd", "plugin-scaffold",
                        "my_guard", "--dir", str(tmp_path)],
                       capture_output=True, text=True, env=env, timeout=120)
    assert r.returncode == 0, r.stderr
    spec = importlib.util.spec_from_file_location("my_guard", tmp_path / "my_guard.py")
    m = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(m)
    p = m.MyGuardPlugin()
    assert p.name == "my_guard" and len(p.hooks) == 2

    r2 = subprocess.run([sys.executable, "-m", "mcp_context_forge_amd", "plugin-scaffold",
                         "av", "--dir", str(tmp_path), "--external"],
                        capture_output=True, text=True, env=env, timeout=120)
    assert r2.returncode == 0, r2.stderr
    spec2 = importlib.util.spec_from_file_location("av_service", tmp_path / "av_service.py")
    m2 = importlib.util.module_from_spec(spec2)
    spec2.loader.exec_module(m2)
    assert callable(m2.app)
