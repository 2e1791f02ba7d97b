"""Reverse proxy: expose firewalled local MCP servers through the gateway.

Reference analog: mcpgateway/reverse_proxy.py (:5-12) — an outbound tunnel
from a NAT'd machine registers a local stdio server with a remote gateway,
which then proxies tool calls back down the tunnel. The reference uses
WS/SSE; this build uses SSE (gateway→client request stream) + POST
(client→gateway responses) since the image has no websocket-client wheel.

Server side: `TunnelRegistry` + `TunnelUpstream` (an UpstreamClient whose
"network" is the tunnel queue). Client side: `ReverseProxyClient` pairs the
SSE stream with a local forward function (usually a stdio subprocess).
"""

from __future__ import annotations

import asyncio
import itertools
import json
import uuid
from typing import Any, Awaitable, Callable, Dict, List, Optional

import httpx

from ..protocol import jsonrpc
from ..protocol.mcp import PROTOCOL_VERSION
from ..services.upstream import UpstreamClient, UpstreamError


class TunnelState:
    def __init__(self, tunnel_id: str, name: str):
        self.tunnel_id = tunnel_id
        self.name = name
        self.requests: "asyncio.Queue[dict]" = asyncio.Queue()
        self.futures: Dict[int, asyncio.Future] = {}
        self.connected = False
        self._ids = itertools.count(1)

    async def roundtrip(self, body: dict, timeout: float = 30.0) -> dict:
        rid = next(self._ids)
        body = dict(body, id=rid)
        loop = asyncio.get_running_loop()
        fut: asyncio.Future = loop.create_future()
        self.futures[rid] = fut
        await self.requests.put(body)
        try:
            return await asyncio.wait_for(fut, timeout=timeout)
        except asyncio.TimeoutError as exc:
            self.futures.pop(rid, None)
            raise UpstreamError(f"reverse-proxy tunnel {self.name} timed out") from exc

    def resolve(self, rid: int, response: dict) -> bool:
        fut = self.futures.pop(rid, None)
        if fut is not None and not fut.done():
            fut.set_result(response)
            return True
        return False


class TunnelRegistry:
    def __init__(self):
        self.tunnels: Dict[str, TunnelState] = {}

    def create(self, name: str) -> TunnelState:
        t = TunnelState(uuid.uuid4().hex, name)
        self.tunnels[t.tunnel_id] = t
        return t

    def get(self, tunnel_id: str) -> Optional[TunnelState]:
        return self.tunnels.get(tunnel_id)

    def remove(self, tunnel_id: str) -> None:
        self.tunnels.pop(tunnel_id, None)


class TunnelUpstream(UpstreamClient):
    """Upstream client whose transport is a reverse tunnel."""

    def __init__(self, tunnel: TunnelState):
        self.tunnel = tunnel

    async def _rpc(self, method: str, params: Any = None) -> Any:
        body: dict = {"jsonrpc": "2.0", "method": method}
        if params is not None:
            body["params"] = params
        resp = await self.tunnel.roundtrip(body)
        if "error" in resp:
            err = resp["error"]
            raise UpstreamError(f"tunnel upstream error {err.get('code')}: {err.get('message')}",
                                code=err.get("code", jsonrpc.SERVER_ERROR))
        return resp.get("result")

    async def initialize(self) -> Dict[str, Any]:
        return await self._rpc("initialize", {"protocolVersion": PROTOCOL_VERSION,
                                              "capabilities": {},
                                              "clientInfo": {"name": "reverse-proxy", "version": "0"}}) or {}

    async def list_tools(self) -> List[Dict[str, Any]]:
        return ((await self._rpc("tools/list", {})) or {}).get("tools", [])

    async def list_resources(self) -> List[Dict[str, Any]]:
        try:
            return ((await self._rpc("resources/list", {})) or {}).get("resources", [])
        except UpstreamError:
            return []

    async def list_prompts(self) -> List[Dict[str, Any]]:
        try:
            return ((await self._rpc("prompts/list", {})) or {}).get("prompts", [])
        except UpstreamError:
            return []

    async def call_tool(self, name: str, arguments: Dict[str, Any],
                        headers: Optional[Dict[str, str]] = None) -> Dict[str, Any]:
        return await self._rpc("tools/call", {"name": name, "arguments": arguments}) or {}

    async def ping(self) -> bool:
        try:
            await self._rpc("ping")
            return True
        except UpstreamError:
            return False


class ReverseProxyClient:
    """Runs on the firewalled side: registers with the remote gateway and
    pumps tunneled requests into a local forward function."""

    def __init__(self, gateway_url: str, name: str,
                 forward: Callable[[bytes], Awaitable[Optional[bytes]]],
                 token: Optional[str] = None):
        self.base = gateway_url.rstrip("/")
        self.name = name
        self.forward = forward
        headers = {}
        if token:
            headers["authorization"] = token if token.lower().startswith(("bearer ", "basic ")) else f"Bearer {token}"
        self.client = httpx.AsyncClient(timeout=None, headers=headers)
        self.tunnel_id: Optional[str] = None
        self.handled = 0

    async def register(self) -> str:
        r = await self.client.post(f"{self.base}/reverse-proxy/register", json={"name": self.name})
        r.raise_for_status()
        self.tunnel_id = r.json()["tunnel_id"]
        return self.tunnel_id

    async def serve(self, stop: Optional[asyncio.Event] = None) -> None:
        assert self.tunnel_id, "call register() first"
        async with self.client.stream("GET", f"{self.base}/reverse-proxy/stream",
                                      params={"tunnel_id": self.tunnel_id}) as stream:
            async for line in stream.aiter_lines():
                if stop is not None and stop.is_set():
                    return
                if not line.startswith("data:"):
                    continue
                body = json.loads(line[5:])
                asyncio.ensure_future(self._handle(body))

    async def _handle(self, body: dict) -> None:
        rid = body.get("id")
        raw = json.dumps(body, separators=(",", ":")).encode()
        try:
            out = await self.forward(raw)
            resp = json.loads(out) if out else {"jsonrpc": "2.0", "id": rid, "result": None}
        except Exception as exc:
            resp = {"jsonrpc": "2.0", "id": rid,
                    "error": {"code": jsonrpc.SERVER_ERROR, "message": str(exc)}}
        self.handled += 1
        await self.client.post(f"{self.base}/reverse-proxy/respond",
                               params={"tunnel_id": self.tunnel_id}, json=resp)

    async def aclose(self) -> None:
        if self.tunnel_id:
            try:
                await self.client.delete(f"{self.base}/reverse-proxy/{self.tunnel_id}")
            except httpx.HTTPError:
                pass
        await self.client.aclose()
