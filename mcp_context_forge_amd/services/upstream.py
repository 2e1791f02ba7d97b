"""Upstream MCP clients — how the gateway talks to federated servers.

Reference analogs: gateway_service.connect_to_streamablehttp_server (:7070)
and connect_to_sse_server (:6900); tool_service MCP dispatch (:5849);
upstream_session_registry (session reuse). Implemented directly over httpx —
no MCP SDK dependency — because the wire format is plain JSON-RPC.

`InProcUpstream` is the fake-upstream test/bench harness (reference analog:
the containerized fast_time_server auto-registered in docker-compose.yml:1485):
it round-trips the request through real JSON bytes so the serialization cost
is honest, without a socket.
"""

from __future__ import annotations

import asyncio
import itertools
import json
import time
from typing import Any, Awaitable, Callable, Dict, List, Optional

import httpx

from ..protocol import jsonrpc
from ..protocol.mcp import PROTOCOL_VERSION


class UpstreamError(Exception):
    def __init__(self, message: str, code: int = jsonrpc.SERVER_UNAVAILABLE):
        self.code = code
        super().__init__(message)


class UpstreamClient:
    """Interface for one upstream MCP server."""

    async def initialize(self) -> Dict[str, Any]:
        raise NotImplementedError

    async def list_tools(self) -> List[Dict[str, Any]]:
        raise NotImplementedError

    async def list_resources(self) -> List[Dict[str, Any]]:
        return []

    async def list_prompts(self) -> List[Dict[str, Any]]:
        return []

    async def call_tool(self, name: str, arguments: Dict[str, Any], headers: Optional[Dict[str, str]] = None) -> Dict[str, Any]:
        raise NotImplementedError

    async def ping(self) -> bool:
        raise NotImplementedError

    async def aclose(self) -> None:
        pass


class HttpUpstreamClient(UpstreamClient):
    """Streamable-HTTP JSON-RPC client with session reuse.

    One persistent httpx.AsyncClient per upstream (reference:
    services/http_client_service.py:57 SharedHttpClient pooling).
    """

    _ids = itertools.count(1)

    def __init__(self, url: str, headers: Optional[Dict[str, str]] = None, timeout: float = 30.0,
                 client: Optional[httpx.AsyncClient] = None, token_provider=None):
        self.url = url
        self.base_headers = dict(headers or {})
        self.timeout = timeout
        self._client = client or httpx.AsyncClient(timeout=timeout)
        self._owned = client is None
        self.session_id: Optional[str] = None
        self._init_lock = asyncio.Lock()
        self.initialized = False
        # OAuth client-credentials provider (reference: oauth_manager flow)
        self.token_provider = token_provider

    async def _rpc(self, method: str, params: Any = None, notification: bool = False,
                   extra_headers: Optional[Dict[str, str]] = None) -> Any:
        body: Dict[str, Any] = {"jsonrpc": "2.0", "method": method}
        if params is not None:
            body["params"] = params
        if not notification:
            body["id"] = next(self._ids)
        headers = {"content-type": "application/json", "accept": "application/json, text/event-stream"}
        headers.update(self.base_headers)
        if extra_headers:
            headers.update(extra_headers)
        if self.session_id:
            headers["mcp-session-id"] = self.session_id
        if self.token_provider is not None:
            headers["authorization"] = f"Bearer {await self.token_provider.get_token(self._client)}"
        try:
            resp = await self._client.post(self.url, content=json.dumps(body).encode(), headers=headers)
            if resp.status_code == 401 and self.token_provider is not None:
                # token re-exchange retry (reference: tool_service :5742)
                self.token_provider.invalidate()
                headers["authorization"] = f"Bearer {await self.token_provider.get_token(self._client)}"
                resp = await self._client.post(self.url, content=json.dumps(body).encode(), headers=headers)
        except httpx.HTTPError as exc:
            raise UpstreamError(f"upstream {self.url} unreachable: {exc}") from exc
        sid = resp.headers.get("mcp-session-id")
        if sid:
            self.session_id = sid
        if notification:
            return None
        if resp.status_code >= 400:
            raise UpstreamError(f"upstream {self.url} HTTP {resp.status_code}")
        ctype = resp.headers.get("content-type", "")
        data: Optional[dict] = None
        if ctype.startswith("text/event-stream"):
            # single JSON-RPC response delivered over SSE framing
            for line in resp.text.splitlines():
                if line.startswith("data:"):
                    data = json.loads(line[5:].strip())
        else:
            data = resp.json()
        if data is None:
            raise UpstreamError(f"upstream {self.url}: empty response")
        if "error" in data:
            err = data["error"]
            raise UpstreamError(f"upstream error {err.get('code')}: {err.get('message')}", code=err.get("code", jsonrpc.SERVER_ERROR))
        return data.get("result")

    async def initialize(self) -> Dict[str, Any]:
        async with self._init_lock:
            if self.initialized:
                return {}
            result = await self._rpc(
                "initialize",
                {
                    "protocolVersion": PROTOCOL_VERSION,
                    "capabilities": {},
                    "clientInfo": {"name": "mcp-context-forge-amd", "version": "0.1.0"},
                },
            )
            await self._rpc("notifications/initialized", notification=True)
            self.initialized = True
            return result or {}

    async def list_tools(self) -> List[Dict[str, Any]]:
        await self.initialize()
        result = await self._rpc("tools/list", {})
        return (result or {}).get("tools", [])

    async def list_resources(self) -> List[Dict[str, Any]]:
        await self.initialize()
        try:
            result = await self._rpc("resources/list", {})
            return (result or {}).get("resources", [])
        except UpstreamError:
            return []

    async def list_prompts(self) -> List[Dict[str, Any]]:
        await self.initialize()
        try:
            result = await self._rpc("prompts/list", {})
            return (result or {}).get("prompts", [])
        except UpstreamError:
            return []

    async def call_tool(self, name: str, arguments: Dict[str, Any], headers: Optional[Dict[str, str]] = None) -> Dict[str, Any]:
        await self.initialize()
        result = await self._rpc("tools/call", {"name": name, "arguments": arguments}, extra_headers=headers)
        return result or {}

    async def ping(self) -> bool:
        try:
            await self._rpc("ping")
            return True
        except UpstreamError:
            return False

    async def aclose(self) -> None:
        if self._owned:
            await self._client.aclose()


class SseUpstreamClient(HttpUpstreamClient):
    """Legacy SSE transport client (reference: connect_to_sse_server :6900).

    Protocol: GET {url} opens a long-lived text/event-stream; the server's
    first frame is `event: endpoint` whose data is the message-POST URL;
    JSON-RPC requests are POSTed there (the POST returns 202) and responses
    arrive as `event: message` frames on the GET stream, correlated by id.
    Matches this gateway's own SSE pair (/servers/{id}/sse + /message), so
    two forges federate over SSE end-to-end.
    """

    def __init__(self, url: str, headers: Optional[Dict[str, str]] = None, timeout: float = 30.0,
                 client: Optional[httpx.AsyncClient] = None, token_provider=None):
        super().__init__(url, headers=headers, timeout=timeout, client=client,
                         token_provider=token_provider)
        self._endpoint: Optional[str] = None
        self._endpoint_ready: Optional[asyncio.Future] = None
        self._pending: Dict[Any, asyncio.Future] = {}
        self._reader_task: Optional[asyncio.Task] = None
        self._connect_lock = asyncio.Lock()

    async def _connect(self) -> None:
        async with self._connect_lock:
            if self._reader_task is not None and not self._reader_task.done():
                return
            loop = asyncio.get_running_loop()
            self._endpoint_ready = loop.create_future()
            self._reader_task = asyncio.create_task(self._read_stream())
            try:
                await asyncio.wait_for(asyncio.shield(self._endpoint_ready), timeout=self.timeout)
            except asyncio.TimeoutError as exc:
                raise UpstreamError(f"upstream {self.url}: no SSE endpoint event") from exc

    async def _read_stream(self) -> None:
        headers = {"accept": "text/event-stream"}
        headers.update(self.base_headers)
        if self.token_provider is not None:
            headers["authorization"] = f"Bearer {await self.token_provider.get_token(self._client)}"
        try:
            async with self._client.stream("GET", self.url, headers=headers, timeout=None) as resp:
                if resp.status_code >= 400:
                    raise UpstreamError(f"upstream {self.url} HTTP {resp.status_code}")
                event, data_lines = "", []
                async for line in resp.aiter_lines():
                    if line.startswith("event:"):
                        event = line[6:].strip()
                    elif line.startswith("data:"):
                        data_lines.append(line[5:].strip())
                    elif line == "":
                        if data_lines:
                            self._dispatch_frame(event, "\n".join(data_lines))
                        event, data_lines = "", []
        except Exception as exc:
            if self._endpoint_ready is not None and not self._endpoint_ready.done():
                self._endpoint_ready.set_exception(
                    UpstreamError(f"upstream {self.url} SSE stream failed: {exc}"))
            for fut in self._pending.values():
                if not fut.done():
                    fut.set_exception(UpstreamError(f"upstream {self.url} SSE stream lost: {exc}"))
            self._pending.clear()

    def _dispatch_frame(self, event: str, data: str) -> None:
        if event == "endpoint":
            # relative or absolute message-POST URL
            self._endpoint = data if "://" in data else str(httpx.URL(self.url).join(data))
            if self._endpoint_ready is not None and not self._endpoint_ready.done():
                self._endpoint_ready.set_result(True)
            return
        if event in ("message", ""):
            try:
                obj = json.loads(data)
            except Exception:
                return
            fut = self._pending.pop(obj.get("id"), None)
            if fut is not None and not fut.done():
                fut.set_result(obj)

    async def _rpc(self, method: str, params: Any = None, notification: bool = False,
                   extra_headers: Optional[Dict[str, str]] = None) -> Any:
        await self._connect()
        body: Dict[str, Any] = {"jsonrpc": "2.0", "method": method}
        if params is not None:
            body["params"] = params
        rid = None
        fut: Optional[asyncio.Future] = None
        if not notification:
            rid = next(self._ids)
            body["id"] = rid
            fut = asyncio.get_running_loop().create_future()
            self._pending[rid] = fut
        headers = {"content-type": "application/json"}
        headers.update(self.base_headers)
        if extra_headers:
            headers.update(extra_headers)
        if self.token_provider is not None:
            headers["authorization"] = f"Bearer {await self.token_provider.get_token(self._client)}"
        try:
            resp = await self._client.post(self._endpoint, content=json.dumps(body).encode(),
                                           headers=headers)
        except httpx.HTTPError as exc:
            self._pending.pop(rid, None)
            raise UpstreamError(f"upstream {self.url} unreachable: {exc}") from exc
        if resp.status_code >= 400:
            self._pending.pop(rid, None)
            raise UpstreamError(f"upstream {self.url} HTTP {resp.status_code}")
        if notification:
            return None
        try:
            data = await asyncio.wait_for(fut, timeout=self.timeout)
        except asyncio.TimeoutError as exc:
            self._pending.pop(rid, None)
            raise UpstreamError(f"upstream {self.url}: SSE response timeout") from exc
        if "error" in data:
            err = data["error"]
            raise UpstreamError(f"upstream error {err.get('code')}: {err.get('message')}",
                                code=err.get("code", jsonrpc.SERVER_ERROR))
        return data.get("result")

    async def ping(self) -> bool:
        try:
            await self._rpc("ping")
            return True
        except UpstreamError:
            return False

    async def aclose(self) -> None:
        if self._reader_task is not None:
            self._reader_task.cancel()
            try:
                await self._reader_task
            except (asyncio.CancelledError, Exception):
                pass
            self._reader_task = None
        await super().aclose()


ToolHandler = Callable[[Dict[str, Any]], Awaitable[Any]]


class InProcUpstream(UpstreamClient):
    """In-process fake MCP upstream with honest bytes-level round-trip.

    Used by unit tests and by bench.py's 64-upstream federation config
    (BASELINE.json config 2) — the serialization boundary is real (request
    and response pass through JSON bytes), only the socket is elided.
    """

    def __init__(self, name: str = "fake", latency_s: float = 0.0):
        self.name = name
        self.latency_s = latency_s
        self._tools: Dict[str, tuple] = {}
        self.calls = 0

    def add_tool(self, name: str, handler: ToolHandler, description: str = "",
                 input_schema: Optional[dict] = None, output_schema: Optional[dict] = None) -> None:
        self._tools[name] = (handler, description, input_schema or {"type": "object"}, output_schema)

    async def initialize(self) -> Dict[str, Any]:
        return {"protocolVersion": PROTOCOL_VERSION, "serverInfo": {"name": self.name, "version": "0"}, "capabilities": {"tools": {}}}

    async def list_tools(self) -> List[Dict[str, Any]]:
        out = []
        for name, (_h, desc, ischema, oschema) in self._tools.items():
            td = {"name": name, "description": desc, "inputSchema": ischema}
            if oschema:
                td["outputSchema"] = oschema
            out.append(td)
        return out

    async def call_tool(self, name: str, arguments: Dict[str, Any], headers: Optional[Dict[str, str]] = None) -> Dict[str, Any]:
        # bytes boundary in
        wire = json.dumps({"name": name, "arguments": arguments}).encode()
        req = json.loads(wire)
        self.calls += 1
        if self.latency_s:
            await asyncio.sleep(self.latency_s)
        ent = self._tools.get(req["name"])
        if ent is None:
            raise UpstreamError(f"tool {name} not found on upstream {self.name}", code=jsonrpc.METHOD_NOT_FOUND)
        handler = ent[0]
        value = await handler(req["arguments"])
        if isinstance(value, dict) and "content" in value:
            result = value
        else:
            result = {
                "content": [{"type": "text", "text": value if isinstance(value, str) else json.dumps(value)}],
                "structuredContent": value if isinstance(value, (dict, list)) else None,
                "isError": False,
            }
        # bytes boundary out
        return json.loads(json.dumps(result))

    async def ping(self) -> bool:
        return True


def make_fake_time_upstream(name: str = "fast_time", latency_s: float = 0.0) -> InProcUpstream:
    """fast_time_server analog (reference benchmark upstream; payload2.json
    targets its `convert_time` tool)."""
    up = InProcUpstream(name, latency_s)

    async def get_system_time(args: Dict[str, Any]) -> Any:
        return {"time": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()), "timezone": args.get("timezone", "UTC")}

    async def convert_time(args: Dict[str, Any]) -> Any:
        return {
            "time": args.get("time", "2026-01-01T00:00:00Z"),
            "source_timezone": args.get("source_timezone", "UTC"),
            "target_timezone": args.get("target_timezone", "UTC"),
            "converted": True,
        }

    async def echo(args: Dict[str, Any]) -> Any:
        return args

    up.add_tool("get_system_time", get_system_time, "Get current system time",
                {"type": "object", "properties": {"timezone": {"type": "string"}}})
    up.add_tool(
        "convert_time", convert_time, "Convert time between timezones",
        {
            "type": "object",
            "properties": {
                "time": {"type": "string"},
                "source_timezone": {"type": "string"},
                "target_timezone": {"type": "string"},
            },
            "required": ["time", "source_timezone", "target_timezone"],
        },
    )
    up.add_tool("echo", echo, "Echo arguments back")
    return up


class NativeInProcUpstream(UpstreamClient):
    """C++ in-process MCP upstream (ops/csrc/upstream.cpp) — the native
    analog of the reference's Go fast_time_server benchmark upstream
    (docker-compose.yml:1485). Exposes the same three tools; each call does
    a real parse of the raw argument bytes and serializes a full MCP tool
    result, in C++. The GPU pipeline batches calls (`native_kind` per tool);
    this per-call API keeps the CPU path and tests working identically.
    """

    TOOL_KINDS = {"convert_time": 0, "get_system_time": 1, "echo": 2}

    def __init__(self, name: str = "native_time"):
        self.name = name
        self.calls = 0

    async def initialize(self) -> Dict[str, Any]:
        return {"protocolVersion": PROTOCOL_VERSION, "serverInfo": {"name": self.name, "version": "0"},
                "capabilities": {"tools": {}}}

    async def list_tools(self) -> List[Dict[str, Any]]:
        schema_ct = {
            "type": "object",
            "properties": {"time": {"type": "string"}, "source_timezone": {"type": "string"},
                           "target_timezone": {"type": "string"}},
            "required": ["time", "source_timezone", "target_timezone"],
        }
        return [
            {"name": "convert_time", "description": "Convert time between timezones (native)", "inputSchema": schema_ct},
            {"name": "get_system_time", "description": "Current system time (native)",
             "inputSchema": {"type": "object", "properties": {"timezone": {"type": "string"}}}},
            {"name": "echo", "description": "Echo arguments (native)", "inputSchema": {"type": "object"}},
        ]

    def call_tool_raw(self, kind: int, args_raw: bytes, now_iso: str) -> bytes:
        import numpy as np

        from ..ops import hip

        blob = np.frombuffer(args_raw or b"{}", dtype=np.uint8)
        ab = np.array([0], dtype=np.int32)
        ae = np.array([blob.shape[0]], dtype=np.int32)
        kinds = np.array([kind], dtype=np.int32)
        out, rb, re_ = hip.upstream_call_batch(blob, ab, ae, kinds, now_iso)
        self.calls += 1
        return out[int(rb[0]):int(re_[0])].tobytes()

    async def call_tool(self, name: str, arguments: Dict[str, Any], headers: Optional[Dict[str, str]] = None) -> Dict[str, Any]:
        kind = self.TOOL_KINDS.get(name)
        if kind is None:
            raise UpstreamError(f"tool {name} not found on upstream {self.name}", code=jsonrpc.METHOD_NOT_FOUND)
        now = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
        raw = json.dumps(arguments or {}, separators=(",", ":")).encode()
        out = self.call_tool_raw(kind, raw, now)
        if not out:
            # non-canonical span (escapes/floats/non-ASCII): the C++ fast
            # path punts — compute the identical result in Python
            return self._python_result(kind, arguments or {}, now)
        return json.loads(out)

    def _python_result(self, kind: int, args: Dict[str, Any], now_iso: str) -> Dict[str, Any]:
        """Exact Python mirror of the C++ handlers for punted spans."""
        if kind == 0:
            sc = {"time": args.get("time", "1970-01-01T00:00:00Z"),
                  "source_timezone": args.get("source_timezone", "UTC"),
                  "target_timezone": args.get("target_timezone", "UTC"),
                  "converted": True}
            return {"content": [{"type": "text", "text": "converted"}],
                    "structuredContent": sc, "isError": False}
        if kind == 1:
            return {"content": [{"type": "text", "text": now_iso}],
                    "structuredContent": {"time": now_iso, "timezone": args.get("timezone", "UTC")},
                    "isError": False}
        return {"content": [{"type": "text", "text": json.dumps(args, separators=(",", ":"))}],
                "structuredContent": args, "isError": False}

    async def ping(self) -> bool:
        return True
