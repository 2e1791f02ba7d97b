"""A2A agent registry + invocation.

Reference analogs: services/a2a_service.py (invoke_agent :1997 with UAID
hop-count federation-loop guard, SSRF-validated routing :2602) and
services/a2a_protocol.py (legacy `message/send` vs v1 `SendMessage` JSON-RPC
wire formats :33-36,141; payload synthesis :316,340).

Agent invocations route through the same plugin chain hooks
(agent_pre_invoke / agent_post_invoke) and, per BASELINE.json config 4, the
moderation/PII HIP classifiers run on the batched agent path when the GPU
pipeline is attached.
"""

from __future__ import annotations

import asyncio
import ipaddress
import itertools
import json
import time
import urllib.parse
import uuid
from typing import Any, Dict, List, Optional

import httpx

from ..plugins.framework import HookType, PluginContext, PluginManager, PluginViolationError
from ..protocol import jsonrpc
from ..registry.registry import NotFoundError, Registry

MAX_HOPS = 3
UAID_HEADER = "x-a2a-uaid"
HOP_HEADER = "x-a2a-hop-count"


class A2AError(Exception):
    def __init__(self, message: str, code: int = jsonrpc.SERVER_ERROR):
        self.code = code
        super().__init__(message)


def validate_outbound_url(url: str, allow_private: bool = False) -> None:
    """SSRF guard (reference: a2a_service SSRF-validated routing :2602)."""
    parsed = urllib.parse.urlparse(url)
    if parsed.scheme not in ("http", "https"):
        raise A2AError(f"unsupported scheme {parsed.scheme!r}", code=jsonrpc.INVALID_PARAMS)
    host = parsed.hostname or ""
    if allow_private:
        return
    try:
        addr = ipaddress.ip_address(host)
        if addr.is_private or addr.is_loopback or addr.is_link_local or addr.is_reserved:
            raise A2AError(f"private address {host} rejected", code=jsonrpc.POLICY_DENIED)
    except ValueError:
        if host in ("localhost", "metadata.google.internal"):
            raise A2AError(f"host {host} rejected", code=jsonrpc.POLICY_DENIED)


def build_payload(protocol_version: str, message: str, context: Optional[dict] = None) -> dict:
    """Wire-format synthesis (reference: a2a_protocol.py:316,340)."""
    if protocol_version.startswith("1"):
        # v1 JSON-RPC SendMessage
        return {
            "jsonrpc": "2.0",
            "id": uuid.uuid4().hex,
            "method": "SendMessage",
            "params": {
                "message": {
                    "messageId": uuid.uuid4().hex,
                    "role": "user",
                    "parts": [{"kind": "text", "text": message}],
                    **({"contextId": context.get("context_id")} if context and context.get("context_id") else {}),
                }
            },
        }
    # legacy message/send
    return {
        "jsonrpc": "2.0",
        "id": uuid.uuid4().hex,
        "method": "message/send",
        "params": {"message": {"role": "user", "parts": [{"type": "text", "text": message}]}},
    }


def extract_text(response: dict) -> str:
    """Pull the reply text out of either wire format."""
    result = response.get("result", response)
    if isinstance(result, dict):
        msg = result.get("message") or result
        parts = msg.get("parts") if isinstance(msg, dict) else None
        if parts:
            return " ".join(p.get("text", "") for p in parts if isinstance(p, dict))
        for key in ("text", "content", "output"):
            if isinstance(result.get(key), str):
                return result[key]
    return json.dumps(result, default=str)


class A2AService:
    def __init__(self, registry: Registry, plugins: Optional[PluginManager] = None,
                 allow_private_urls: bool = True):
        # allow_private defaults True for test/bench loopback agents; production
        # config flips it (reference: SSRF strictness is config-driven too)
        self.registry = registry
        self.plugins = plugins or PluginManager([])
        self.allow_private = allow_private_urls
        self._crypto = None   # lazy (PBKDF2 key stretch is ~50 ms)
        self._client: Optional[httpx.AsyncClient] = None
        self._tasks: Dict[str, dict] = {}  # task store (reference: upsert_task :3248)
        self._local_handlers: Dict[str, Any] = {}

    @property
    def crypto(self):
        """Agent credentials sealed at rest like gateway auth material."""
        if self._crypto is None:
            from ..auth.crypto import EncryptionService
            from ..config import get_settings

            self._crypto = EncryptionService(get_settings().auth_encryption_secret)
        return self._crypto

    def register_local_agent(self, name: str, handler, description: str = "", **fields) -> dict:
        """In-proc agent for tests/bench (endpoint_url inproc://)."""
        from ..utils import slugify

        ent = self.registry.create("a2a_agent", name=name, slug=slugify(name),
                                   endpoint_url=f"inproc://{name}", description=description, **fields)
        self._local_handlers[name] = handler
        return ent

    async def register_agent(self, name: str, endpoint_url: str, agent_type: str = "generic",
                             protocol_version: str = "1.0", description: str = "",
                             auth_type: Optional[str] = None, auth_value: Optional[str] = None,
                             tags: Optional[List[str]] = None, config: Optional[dict] = None) -> dict:
        from ..utils import slugify

        if not endpoint_url.startswith("inproc://"):
            validate_outbound_url(endpoint_url, self.allow_private)
        return self.registry.create(
            "a2a_agent", name=name, slug=slugify(name), endpoint_url=endpoint_url,
            agent_type=agent_type, protocol_version=protocol_version, description=description,
            auth_type=auth_type, auth_value=self.crypto.seal(auth_value),
            tags=tags or [], config=config or {})

    async def invoke_agent(self, name: str, message: str, user: Optional[str] = None,
                           context: Optional[dict] = None, hop_count: int = 0,
                           uaid: Optional[str] = None) -> Dict[str, Any]:
        """Invoke an agent (reference: invoke_agent :1997)."""
        agent = self.registry.find("a2a_agent", name)
        if agent is None or not agent.get("enabled", True):
            raise A2AError(f"Agent not found: {name}", code=jsonrpc.INVALID_PARAMS)
        if hop_count >= MAX_HOPS:
            raise A2AError("A2A federation loop detected (hop count exceeded)", code=jsonrpc.POLICY_DENIED)
        uaid = uaid or uuid.uuid4().hex

        ctx = PluginContext(hook=HookType.AGENT_PRE_INVOKE, name=name, args={"message": message}, user=user)
        try:
            ctx = await self.plugins.invoke_hook(HookType.AGENT_PRE_INVOKE, ctx)
        except PluginViolationError as exc:
            raise A2AError(str(exc), code=jsonrpc.POLICY_DENIED) from exc
        if isinstance(ctx.args, dict):
            message = ctx.args.get("message", message)

        t0 = time.monotonic()
        if agent["endpoint_url"].startswith("inproc://"):
            handler = self._local_handlers.get(name)
            if handler is None:
                raise A2AError(f"no local handler for agent {name}", code=jsonrpc.SERVER_UNAVAILABLE)
            reply_text = await handler(message, context or {})
            raw_response: dict = {"result": {"message": {"parts": [{"kind": "text", "text": reply_text}]}}}
        else:
            raw_response = await self._invoke_remote(agent, message, context, hop_count, uaid)
            reply_text = extract_text(raw_response)

        result = {
            "agent": name,
            "response": reply_text,
            "raw": raw_response,
            "latency_ms": round((time.monotonic() - t0) * 1000, 2),
            "uaid": uaid,
        }
        ctx.hook = HookType.AGENT_POST_INVOKE
        ctx.args = result
        try:
            ctx = await self.plugins.invoke_hook(HookType.AGENT_POST_INVOKE, ctx)
        except PluginViolationError as exc:
            raise A2AError(str(exc), code=jsonrpc.POLICY_DENIED) from exc
        return ctx.args if isinstance(ctx.args, dict) else result

    async def _invoke_remote(self, agent: dict, message: str, context: Optional[dict],
                             hop_count: int, uaid: str) -> dict:
        """Cross-gateway HTTP leg (reference: _invoke_remote_agent :2602)."""
        validate_outbound_url(agent["endpoint_url"], self.allow_private)
        if self._client is None:
            self._client = httpx.AsyncClient(timeout=30.0)
        payload = build_payload(agent.get("protocol_version", "1.0"), message, context)
        headers = {"content-type": "application/json", HOP_HEADER: str(hop_count + 1), UAID_HEADER: uaid}
        if agent.get("auth_type") == "bearer" and agent.get("auth_value"):
            headers["authorization"] = f"Bearer {self.crypto.open_(agent['auth_value'])}"
        try:
            resp = await self._client.post(agent["endpoint_url"], json=payload, headers=headers)
        except httpx.HTTPError as exc:
            raise A2AError(f"agent unreachable: {exc}", code=jsonrpc.SERVER_UNAVAILABLE) from exc
        if resp.status_code >= 400:
            raise A2AError(f"agent HTTP {resp.status_code}", code=jsonrpc.SERVER_ERROR)
        return resp.json()

    # -- task store (reference: upsert_task :3248) -------------------------------
    def upsert_task(self, task_id: str, status: str, detail: Optional[dict] = None) -> dict:
        task = self._tasks.setdefault(task_id, {"id": task_id, "created_at": time.time()})
        task["status"] = status
        task["updated_at"] = time.time()
        if detail:
            task["detail"] = detail
        return task

    def get_task(self, task_id: str) -> Optional[dict]:
        return self._tasks.get(task_id)

    async def aclose(self) -> None:
        if self._client is not None:
            await self._client.aclose()
