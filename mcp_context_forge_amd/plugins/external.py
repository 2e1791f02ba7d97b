"""External-process plugins: hooks served by a separate HTTP microservice.

Reference analog: plugins/external/{opa,cedar,llmguard,clamav_server} —
policy/guard engines running as their own processes, reached over HTTP,
with the gateway enforcing timeouts and mode semantics. Round 1 collapsed
these into in-proc engines (plugins/integrations.py); this module restores
the real process boundary.

Wire protocol (documented contract; the reference's external plugins speak
cpex-over-MCP — this build's native protocol is plain JSON over HTTP):

    POST {url}/hook
      -> {"hook": "tool_pre_invoke", "plugin": "...", "name": "<tool>",
          "args": <payload>, "user": "...", "server_id": "...",
          "metadata": {...}}
      <- {"action": "allow" | "block" | "transform",
          "reason": "...",            (block)
          "code": "...",              (block, optional)
          "payload": <new payload>,   (transform)
          "metadata": {...}}          (optional, merged into result metadata)

    GET {url}/health -> 200 {"status": "ok"}

Failure semantics follow the plugin MODE exactly like in-proc plugins:
enforce -> a dead/erroring service BLOCKS the request (fail-closed);
enforce_ignore_error / permissive -> failures pass through (fail-open).

`build_external_service_app()` returns a ready-to-serve ASGI app skeleton
for WRITING such a service (the reference ships server scaffolds per
plugin); tests run one in a separate uvicorn server to prove the process
boundary.
"""

from __future__ import annotations

import asyncio
import logging
from typing import Any, Callable, Dict, Optional

import httpx

from .framework import HookType, Plugin, PluginContext, PluginMode, PluginResult

logger = logging.getLogger(__name__)

DEFAULT_HOOKS = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE)


class ExternalServicePlugin(Plugin):
    """Hook proxy to an external HTTP plugin service."""

    name = "external_service"
    hooks = DEFAULT_HOOKS
    priority = 60

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.url = (self.config.get("url") or "").rstrip("/")
        if not self.url:
            raise ValueError("external plugin requires config.url")
        self.plugin_name = self.config.get("service_name") or self.name
        self.timeout = float(self.config.get("timeout", 5.0))
        self.headers = dict(self.config.get("headers") or {})
        hooks = self.config.get("hooks")
        if hooks:
            self.hooks = tuple(HookType(h) for h in hooks)
        self._client: Optional[httpx.AsyncClient] = None
        self.calls = 0
        self.failures = 0

    def _ensure_client(self) -> httpx.AsyncClient:
        if self._client is None:
            self._client = httpx.AsyncClient(timeout=self.timeout, headers=self.headers)
        return self._client

    async def health(self) -> bool:
        try:
            r = await self._ensure_client().get(f"{self.url}/health")
            return r.status_code == 200
        except httpx.HTTPError:
            return False

    async def _call(self, ctx: PluginContext) -> PluginResult:
        body = {
            "hook": ctx.hook.value,
            "plugin": self.plugin_name,
            "name": ctx.name,
            "args": ctx.args,
            "user": ctx.user,
            "server_id": ctx.server_id,
            "metadata": {},
        }
        self.calls += 1
        try:
            resp = await self._ensure_client().post(f"{self.url}/hook", json=body)
            if resp.status_code >= 400:
                raise httpx.HTTPStatusError(f"HTTP {resp.status_code}", request=resp.request,
                                            response=resp)
            out = resp.json()
        except (httpx.HTTPError, ValueError) as exc:
            self.failures += 1
            # mode decides fail-open vs fail-closed — same contract the
            # framework applies to in-proc plugin exceptions
            if self.mode == PluginMode.ENFORCE:
                return PluginResult.block(
                    f"external plugin {self.plugin_name} unavailable: {exc}",
                    code="external_plugin_error")
            logger.warning("external plugin %s failed open: %s", self.plugin_name, exc)
            return PluginResult.ok()
        action = out.get("action", "allow")
        if action == "block":
            return PluginResult.block(out.get("reason", "blocked by external plugin"),
                                      code=out.get("code", "external_policy"),
                                      **(out.get("metadata") or {}))
        if action == "transform":
            return PluginResult.ok(out.get("payload"), **(out.get("metadata") or {}))
        return PluginResult.ok(**(out.get("metadata") or {}))

    # every hook routes through the same wire call
    tool_pre_invoke = _call
    tool_post_invoke = _call
    prompt_pre_fetch = _call
    prompt_post_fetch = _call
    resource_pre_fetch = _call
    resource_post_fetch = _call
    agent_pre_invoke = _call
    agent_post_invoke = _call
    http_pre_request = _call
    http_post_request = _call

    async def shutdown(self) -> None:
        if self._client is not None:
            await self._client.aclose()
            self._client = None


HookFn = Callable[[Dict[str, Any]], Any]


def build_external_service_app(handler: HookFn, name: str = "external-plugin"):
    """ASGI skeleton for an external plugin SERVICE (the other side of the
    wire). `handler(request_body) -> response_body` implements the policy;
    it may be sync or async. Reference analog: the server scaffolds under
    plugins/external/*/server.py."""

    async def app(scope, receive, send):
        if scope["type"] == "lifespan":
            while True:
                msg = await receive()
                if msg["type"] == "lifespan.startup":
                    await send({"type": "lifespan.startup.complete"})
                elif msg["type"] == "lifespan.shutdown":
                    await send({"type": "lifespan.shutdown.complete"})
                    return
        if scope["type"] != "http":
            return

        async def respond(status: int, payload: dict) -> None:
            import json as _json

            body = _json.dumps(payload).encode()
            await send({"type": "http.response.start", "status": status,
                        "headers": [(b"content-type", b"application/json"),
                                    (b"content-length", str(len(body)).encode())]})
            await send({"type": "http.response.body", "body": body})

        if scope["path"] == "/health":
            await respond(200, {"status": "ok", "plugin": name})
            return
        if scope["path"] == "/hook" and scope["method"] == "POST":
            import json as _json

            raw = b""
            while True:
                msg = await receive()
                raw += msg.get("body", b"")
                if not msg.get("more_body", False):
                    break
            try:
                req = _json.loads(raw or b"{}")
            except ValueError:
                await respond(400, {"error": "bad json"})
                return
            try:
                out = handler(req)
                if asyncio.iscoroutine(out):
                    out = await out
            except Exception as exc:  # the service's own bug → 500
                await respond(500, {"error": str(exc)})
                return
            await respond(200, out or {"action": "allow"})
            return
        await respond(404, {"error": "not found"})

    return app
