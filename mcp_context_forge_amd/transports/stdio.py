"""stdio transport + bridges.

Reference analogs:
  transports/stdio_transport.py (:47)  — gateway engine as a stdio MCP server
  mcpgateway/wrapper.py (:5-13)        — the *gateway itself* exposed as a stdio
                                         MCP server for clients like Claude
                                         Desktop, proxying to a remote gateway
  mcpgateway/translate.py              — stdio↔streamable-HTTP bridge with an
                                         embedded FastAPI (StdIOEndpoint :310)

All three speak newline-delimited JSON-RPC on stdio.
"""

from __future__ import annotations

import asyncio
import json
import sys
from typing import Any, Dict, List, Optional

import fastapi

from ..engine import GatewayEngine


async def _stdio_streams():
    loop = asyncio.get_running_loop()
    reader = asyncio.StreamReader()
    await loop.connect_read_pipe(lambda: asyncio.StreamReaderProtocol(reader), sys.stdin)
    w_transport, w_protocol = await loop.connect_write_pipe(asyncio.streams.FlowControlMixin, sys.stdout)
    writer = asyncio.StreamWriter(w_transport, w_protocol, None, loop)
    return reader, writer


class StdioServer:
    """Serve the engine over stdio (newline-delimited JSON-RPC)."""

    def __init__(self, engine: GatewayEngine, user: str = "stdio"):
        self.engine = engine
        self.user = user

    async def serve(self, reader=None, writer=None) -> None:
        if reader is None or writer is None:
            reader, writer = await _stdio_streams()
        session = self.engine.sessions.create(transport="stdio", user=self.user)
        while True:
            line = await reader.readline()
            if not line:
                break
            line = line.strip()
            if not line:
                continue
            out = await self.engine.handle_rpc_bytes(line, user=self.user, session=session)
            if out is not None:
                writer.write(out + b"\n")
                await writer.drain()
        self.engine.sessions.remove(session.session_id)


class StdioSubprocess:
    """Manage a stdio MCP server subprocess (reference: translate.StdIOEndpoint :310)."""

    def __init__(self, command: List[str]):
        self.command = command
        self.proc: Optional[asyncio.subprocess.Process] = None
        self._lock = asyncio.Lock()
        self._next_id = 1

    async def start(self) -> None:
        self.proc = await asyncio.create_subprocess_exec(
            *self.command, stdin=asyncio.subprocess.PIPE, stdout=asyncio.subprocess.PIPE)

    async def request(self, method: str, params: Any = None, notification: bool = False,
                      timeout: float = 30.0) -> Optional[dict]:
        assert self.proc is not None
        async with self._lock:
            body: Dict[str, Any] = {"jsonrpc": "2.0", "method": method}
            if params is not None:
                body["params"] = params
            if not notification:
                body["id"] = self._next_id
                self._next_id += 1
            self.proc.stdin.write(json.dumps(body, separators=(",", ":")).encode() + b"\n")
            await self.proc.stdin.drain()
            if notification:
                return None
            while True:
                line = await asyncio.wait_for(self.proc.stdout.readline(), timeout=timeout)
                if not line:
                    raise RuntimeError("stdio subprocess closed")
                try:
                    obj = json.loads(line)
                except ValueError:
                    continue
                if obj.get("id") == body["id"]:
                    return obj

    async def send_raw(self, raw: bytes, want_id: Any, timeout: float = 30.0) -> Optional[bytes]:
        assert self.proc is not None
        async with self._lock:
            self.proc.stdin.write(raw.rstrip(b"\n") + b"\n")
            await self.proc.stdin.drain()
            if want_id is None:
                return None
            while True:
                line = await asyncio.wait_for(self.proc.stdout.readline(), timeout=timeout)
                if not line:
                    raise RuntimeError("stdio subprocess closed")
                try:
                    obj = json.loads(line)
                except ValueError:
                    continue
                if obj.get("id") == want_id:
                    return line.rstrip(b"\n")

    async def stop(self) -> None:
        if self.proc is not None:
            try:
                self.proc.stdin.close()
                await asyncio.wait_for(self.proc.wait(), timeout=5)
            except (asyncio.TimeoutError, ProcessLookupError):
                self.proc.kill()


class GatewayWrapper:
    """stdio client ↔ remote gateway bridge (reference: mcpgateway/wrapper.py).

    Reads JSON-RPC from stdin, forwards to the remote gateway's /rpc (or /mcp)
    over streamable HTTP, writes responses to stdout.
    """

    def __init__(self, base_url: str, token: Optional[str] = None, use_mcp: bool = True):
        import httpx

        self.base_url = base_url.rstrip("/")
        headers = {"content-type": "application/json"}
        if token:
            headers["authorization"] = token if token.lower().startswith(("bearer ", "basic ")) else f"Bearer {token}"
        self.client = httpx.AsyncClient(timeout=60.0, headers=headers)
        self.path = "/mcp" if use_mcp else "/rpc"
        self.session_id: Optional[str] = None

    async def forward(self, raw: bytes) -> Optional[bytes]:
        headers = {}
        if self.session_id:
            headers["mcp-session-id"] = self.session_id
        resp = await self.client.post(self.base_url + self.path, content=raw, headers=headers)
        sid = resp.headers.get("mcp-session-id")
        if sid:
            self.session_id = sid
        if resp.status_code == 202:
            return None
        return resp.content

    async def serve(self, reader=None, writer=None) -> None:
        if reader is None or writer is None:
            reader, writer = await _stdio_streams()
        while True:
            line = await reader.readline()
            if not line:
                break
            line = line.strip()
            if not line:
                continue
            try:
                out = await self.forward(line)
            except Exception as exc:
                try:
                    rid = json.loads(line).get("id")
                except Exception:
                    rid = None
                out = json.dumps({"jsonrpc": "2.0", "id": rid,
                                  "error": {"code": -32002, "message": f"gateway unreachable: {exc}"}}).encode()
            if out is not None:
                writer.write(out + b"\n")
                await writer.drain()
        await self.client.aclose()


def build_translate_app(subproc: StdioSubprocess):
    """Expose a stdio MCP server over streamable HTTP (+SSE) — the
    `translate` bridge (reference: mcpgateway/translate.py:647,1814)."""
    # note: `from __future__ import annotations` stringifies hints, so the
    # fastapi.Request annotation must resolve from module globals
    from fastapi import FastAPI, Response

    app = FastAPI(title="mcp-translate-bridge")

    @app.post("/mcp")
    async def mcp(request: fastapi.Request):
        raw = await request.body()
        try:
            want_id = json.loads(raw).get("id")
        except ValueError:
            return Response(content=json.dumps({"jsonrpc": "2.0", "id": None,
                                                "error": {"code": -32700, "message": "Parse error"}}),
                            media_type="application/json")
        out = await subproc.send_raw(raw, want_id)
        if out is None:
            return Response(status_code=202)
        return Response(content=out, media_type="application/json")

    @app.get("/healthz")
    async def healthz():
        return {"status": "ok"}

    return app
