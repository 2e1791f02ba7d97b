"""Tool invocation — THE hot path.

Reference analog: services/tool_service.py `invoke_tool` (:5067) with the
phase structure documented there: resolve (PHASE 1, lookup cache → DB),
payload extraction (PHASE 2), plugin tool_pre_invoke hooks (:5530), dispatch
by integration_type — REST (:5476), MCP (:5849), A2A (:6526), gRPC (:6642) —
then tool_post_invoke hooks, jsonpath filter, output-schema check and
metrics buffering (:6898).

This class is the per-request (CPU) path and the semantics oracle; the GPU
pipeline (gpu/pipeline.py) executes the same phases over a staged batch and
calls back into `dispatch()` for the upstream leg.
"""

from __future__ import annotations

import asyncio
import json
import time
from typing import Any, Awaitable, Callable, Dict, List, Optional

from ..plugins.framework import HookType, PluginContext, PluginManager, PluginViolationError
from ..protocol import jsonrpc
from ..registry.registry import Registry
from ..utils import RetryManager, jsonpath_filter
from ..utils.jsonschema import validate as schema_validate
from .metrics import MetricsBuffer
from .upstream import InProcUpstream, UpstreamClient, UpstreamError

ToolHandler = Callable[[Dict[str, Any]], Awaitable[Any]]


class ToolNotFoundError(Exception):
    pass


class ToolInvocationError(Exception):
    def __init__(self, message: str, code: int = jsonrpc.SERVER_ERROR):
        self.code = code
        super().__init__(message)


class ToolService:
    def __init__(
        self,
        registry: Registry,
        plugin_manager: Optional[PluginManager] = None,
        metrics: Optional[MetricsBuffer] = None,
        max_retries: int = 1,
    ):
        self.registry = registry
        self.plugins = plugin_manager or PluginManager([])
        self.metrics = metrics or MetricsBuffer()
        self.retry = RetryManager(max_retries=max_retries)
        self._local_handlers: Dict[str, ToolHandler] = {}
        self._upstreams: Dict[str, UpstreamClient] = {}  # gateway_id -> client
        self._rest_client = None  # lazy httpx.AsyncClient
        self.a2a_service = None  # wired by the engine (A2A tool dispatch)

    # -- wiring ---------------------------------------------------------------
    def register_local_tool(self, name: str, handler: ToolHandler, description: str = "",
                            input_schema: Optional[dict] = None, output_schema: Optional[dict] = None,
                            **fields: Any) -> Dict[str, Any]:
        """Register an in-process tool (integration_type=LOCAL)."""
        ent = self.registry.create(
            "tool",
            name=name,
            original_name=name,
            description=description,
            integration_type="LOCAL",
            input_schema=input_schema or {"type": "object"},
            output_schema=output_schema,
            **fields,
        )
        self._local_handlers[name] = handler
        return ent

    def attach_upstream(self, gateway_id: str, client: UpstreamClient) -> None:
        self._upstreams[gateway_id] = client

    def upstream_for(self, gateway_id: Optional[str]) -> Optional[UpstreamClient]:
        return self._upstreams.get(gateway_id or "")

    # -- resolve (PHASE 1) ------------------------------------------------------
    def resolve(self, name: str) -> Dict[str, Any]:
        tool = self.registry.lookup_tool(name)
        if tool is None:
            raise ToolNotFoundError(f"Tool not found: {name}")
        if not tool.get("reachable", True):
            raise ToolInvocationError(f"Tool {name} currently unreachable", code=jsonrpc.SERVER_UNAVAILABLE)
        return tool

    # -- dispatch (the upstream leg; shared with the GPU pipeline) --------------
    def _apply_jsonpath(self, tool: Dict[str, Any], result: Dict[str, Any]) -> Dict[str, Any]:
        """Tool-level result filter (reference: main.py:1281 jsonpath_modifier
        applied via the tool row's jsonpath_filter)."""
        jp = tool.get("jsonpath_filter")
        if not jp or not isinstance(result, dict) or result.get("isError"):
            return result
        payload = result.get("structuredContent")
        if payload is None:
            return result
        filtered = jsonpath_filter(payload, jp)
        return {**result,
                "structuredContent": filtered,
                "content": [{"type": "text",
                             "text": filtered if isinstance(filtered, str)
                             else json.dumps(filtered, default=str)}]}

    async def dispatch(self, tool: Dict[str, Any], arguments: Dict[str, Any],
                       headers: Optional[Dict[str, str]] = None) -> Dict[str, Any]:
        itype = tool.get("integration_type", "MCP")
        if itype == "LOCAL":
            handler = self._local_handlers.get(tool["name"])
            if handler is None:
                raise ToolInvocationError(f"no local handler for {tool['name']}")
            value = await handler(arguments or {})
            if isinstance(value, dict) and "content" in value:
                return self._apply_jsonpath(tool, value)
            return self._apply_jsonpath(tool, {
                "content": [{"type": "text", "text": value if isinstance(value, str) else json.dumps(value, default=str)}],
                "structuredContent": value if isinstance(value, (dict, list)) else None,
                "isError": False,
            })
        if itype == "MCP":
            client = self._upstreams.get(tool.get("gateway_id") or "")
            if client is None:
                raise ToolInvocationError(f"no upstream for tool {tool['name']}", code=jsonrpc.SERVER_UNAVAILABLE)
            out = await self.retry.run(
                lambda: client.call_tool(tool["original_name"], arguments or {}, headers),
                retry_on=(UpstreamError,),
            )
            return self._apply_jsonpath(tool, out)
        if itype == "REST":
            return await self._dispatch_rest(tool, arguments, headers)
        if itype == "A2A":
            # A2A-integrated tool: arguments carry the message (reference: tool_service :6526)
            if self.a2a_service is None:
                raise ToolInvocationError("A2A service not wired", code=jsonrpc.SERVER_UNAVAILABLE)
            message = arguments.get("message") if isinstance(arguments, dict) else None
            if not isinstance(message, str):
                message = json.dumps(arguments or {}, default=str)
            agent_name = tool.get("original_name") or tool["name"]
            try:
                res = await self.a2a_service.invoke_agent(agent_name, message)
            except Exception as exc:
                code = getattr(exc, "code", jsonrpc.SERVER_ERROR)
                raise ToolInvocationError(str(exc), code=code) from exc
            return {
                "content": [{"type": "text", "text": res.get("response", "")}],
                "structuredContent": {k: v for k, v in res.items() if k != "raw"},
                "isError": False,
            }
        raise ToolInvocationError(f"unsupported integration_type {itype}")

    async def _dispatch_rest(self, tool: Dict[str, Any], arguments: Dict[str, Any],
                             headers: Optional[Dict[str, str]]) -> Dict[str, Any]:
        """REST adapter (reference: tool_service.py:5476)."""
        import httpx

        if self._rest_client is None:
            self._rest_client = httpx.AsyncClient(timeout=30.0)
        req_headers = dict(tool.get("headers") or {})
        if headers:
            req_headers.update(headers)
        method = (tool.get("request_type") or "POST").upper()
        url = tool.get("url")
        if not url:
            raise ToolInvocationError(f"REST tool {tool['name']} has no url")
        try:
            if method == "GET":
                resp = await self._rest_client.get(url, params=arguments or {}, headers=req_headers)
            else:
                resp = await self._rest_client.request(method, url, json=arguments or {}, headers=req_headers)
        except httpx.HTTPError as exc:
            raise ToolInvocationError(f"REST upstream error: {exc}", code=jsonrpc.SERVER_UNAVAILABLE) from exc
        if resp.status_code >= 400:
            return {"content": [{"type": "text", "text": f"HTTP {resp.status_code}: {resp.text[:500]}"}], "isError": True}
        try:
            value = resp.json()
        except ValueError:
            value = resp.text
        if tool.get("jsonpath_filter"):
            value = jsonpath_filter(value, tool["jsonpath_filter"])
        return {
            "content": [{"type": "text", "text": value if isinstance(value, str) else json.dumps(value, default=str)}],
            "structuredContent": value if isinstance(value, (dict, list)) else None,
            "isError": False,
        }

    # -- the full per-request path ----------------------------------------------
    async def invoke_tool(self, name: str, arguments: Optional[Dict[str, Any]] = None,
                          user: Optional[str] = None, server_id: Optional[str] = None,
                          headers: Optional[Dict[str, str]] = None) -> Dict[str, Any]:
        t0 = time.monotonic()
        tool = self.resolve(name)
        arguments = arguments or {}
        ctx = PluginContext(
            hook=HookType.TOOL_PRE_INVOKE,
            name=name,
            args=arguments,
            user=user,
            server_id=server_id,
            headers=dict(headers or {}),
            state={
                "input_schema": tool.get("input_schema"),
                "output_schema": tool.get("output_schema"),
                "request_text": json.dumps(arguments, separators=(",", ":"), sort_keys=True, default=str),
            },
        )
        success = False
        try:
            # pre hooks (may mutate args, may block, may set cache_hit)
            try:
                ctx = await self.plugins.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx)
            except PluginViolationError as exc:
                raise ToolInvocationError(str(exc), code=jsonrpc.POLICY_DENIED) from exc
            arguments = ctx.args if isinstance(ctx.args, dict) else arguments

            if "cache_hit" in ctx.state:
                result = ctx.state["cache_hit"]
                success = True
                return result

            result = await self.dispatch(tool, arguments, ctx.headers or headers)

            # output schema check (reference: tool_service output_schema handling)
            oschema = tool.get("output_schema")
            if oschema and isinstance(result, dict):
                payload = result.get("structuredContent", result)
                errs = schema_validate(payload, oschema)
                if errs:
                    result = {
                        "content": [{"type": "text", "text": "output schema violation: " + "; ".join(errs[:3])}],
                        "isError": True,
                    }

            # post hooks (may rewrite the result)
            ctx.hook = HookType.TOOL_POST_INVOKE
            ctx.args = result
            try:
                ctx = await self.plugins.invoke_hook(HookType.TOOL_POST_INVOKE, ctx)
            except PluginViolationError as exc:
                raise ToolInvocationError(str(exc), code=jsonrpc.POLICY_DENIED) from exc
            result = ctx.args if ctx.args is not None else result
            success = not (isinstance(result, dict) and result.get("isError"))
            return result
        finally:
            ms = (time.monotonic() - t0) * 1000.0
            self.metrics.record_tool_metric(tool.get("id", name), ms, success)

    async def list_tools(self, server_id: Optional[str] = None, include_disabled: bool = False) -> List[Dict[str, Any]]:
        tools = self.registry.list("tool", include_disabled=include_disabled)
        if server_id:
            server = self.registry.find("server", server_id) or (
                self.registry.get("server", server_id) if server_id in self.registry._by_id["server"] else None
            )
            if server:
                allowed = set(server.get("associated_tools") or [])
                tools = [t for t in tools if t["id"] in allowed]
        return tools

    async def aclose(self) -> None:
        for c in self._upstreams.values():
            try:
                await c.aclose()
            except Exception:
                pass
        if self._rest_client is not None:
            await self._rest_client.aclose()
