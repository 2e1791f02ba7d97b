"""Prompt / resource / completion / roots services.

Reference analogs: services/prompt_service.py (Jinja rendering),
services/resource_service.py (read path, templates, subscriptions),
services/completion_service.py, services/root_service.py.
"""

from __future__ import annotations

import re
from typing import Any, Dict, List, Optional

from jinja2 import Environment, StrictUndefined
from jinja2.sandbox import SandboxedEnvironment

from ..plugins.framework import HookType, PluginContext, PluginManager, PluginViolationError
from ..protocol import jsonrpc
from ..registry.registry import NotFoundError, Registry


class PromptService:
    def __init__(self, registry: Registry, plugins: Optional[PluginManager] = None):
        self.registry = registry
        self.plugins = plugins or PluginManager([])
        self._env = SandboxedEnvironment(undefined=StrictUndefined, autoescape=False)

    async def get_prompt(self, name: str, arguments: Optional[Dict[str, Any]] = None,
                         user: Optional[str] = None) -> Dict[str, Any]:
        prompt = self.registry.find("prompt", name)
        if prompt is None or not prompt.get("enabled", True):
            raise jsonrpc.JSONRPCError(jsonrpc.INVALID_PARAMS, f"Prompt not found: {name}")
        args = arguments or {}
        ctx = PluginContext(hook=HookType.PROMPT_PRE_FETCH, name=name, args=args, user=user)
        try:
            ctx = await self.plugins.invoke_hook(HookType.PROMPT_PRE_FETCH, ctx)
        except PluginViolationError as exc:
            raise jsonrpc.JSONRPCError(jsonrpc.POLICY_DENIED, str(exc)) from exc
        args = ctx.args if isinstance(ctx.args, dict) else args
        template = prompt.get("template") or ""
        try:
            text = self._env.from_string(template).render(**args)
        except Exception as exc:
            raise jsonrpc.JSONRPCError(jsonrpc.INVALID_PARAMS, f"template render failed: {exc}") from exc
        result = {
            "description": prompt.get("description", ""),
            "messages": [{"role": "user", "content": {"type": "text", "text": text}}],
        }
        ctx.hook = HookType.PROMPT_POST_FETCH
        ctx.args = result
        try:
            ctx = await self.plugins.invoke_hook(HookType.PROMPT_POST_FETCH, ctx)
        except PluginViolationError as exc:
            raise jsonrpc.JSONRPCError(jsonrpc.POLICY_DENIED, str(exc)) from exc
        return ctx.args if isinstance(ctx.args, dict) else result

    def list_prompts(self) -> List[Dict[str, Any]]:
        out = []
        for p in self.registry.list("prompt", include_disabled=False):
            args = (p.get("argument_schema") or {}).get("arguments", [])
            out.append({"name": p["name"], "description": p.get("description", ""), "arguments": args})
        return out


_TEMPLATE_RE = re.compile(r"\{(\w+)\}")


class ResourceService:
    def __init__(self, registry: Registry, plugins: Optional[PluginManager] = None):
        self.registry = registry
        self.plugins = plugins or PluginManager([])
        self._subscriptions: Dict[str, set] = {}

    async def read_resource(self, uri: str, user: Optional[str] = None) -> Dict[str, Any]:
        ctx = PluginContext(hook=HookType.RESOURCE_PRE_FETCH, name=uri, args={"uri": uri}, user=user)
        try:
            ctx = await self.plugins.invoke_hook(HookType.RESOURCE_PRE_FETCH, ctx)
        except PluginViolationError as exc:
            raise jsonrpc.JSONRPCError(jsonrpc.POLICY_DENIED, str(exc)) from exc
        res = self.registry.find("resource", uri) or self._match_template(uri)
        if res is None or not res.get("enabled", True):
            raise jsonrpc.JSONRPCError(jsonrpc.INVALID_PARAMS, f"Resource not found: {uri}")
        content = res.get("content")
        if content is None and res.get("template"):
            content = res["template"]
        result = {
            "contents": [{"uri": uri, "mimeType": res.get("mime_type", "text/plain"), "text": content or ""}]
        }
        ctx.hook = HookType.RESOURCE_POST_FETCH
        ctx.args = result
        try:
            ctx = await self.plugins.invoke_hook(HookType.RESOURCE_POST_FETCH, ctx)
        except PluginViolationError as exc:
            raise jsonrpc.JSONRPCError(jsonrpc.POLICY_DENIED, str(exc)) from exc
        return ctx.args if isinstance(ctx.args, dict) else result

    def _match_template(self, uri: str) -> Optional[Dict[str, Any]]:
        for r in self.registry.list("resource"):
            tpl_uri = r.get("uri", "")
            if "{" not in tpl_uri:
                continue
            pattern = "^" + _TEMPLATE_RE.sub(r"(?P<\1>[^/]+)", re.escape(tpl_uri).replace(r"\{", "{").replace(r"\}", "}")) + "$"
            if re.match(pattern, uri):
                return r
        return None

    def list_resources(self) -> List[Dict[str, Any]]:
        return [
            {"uri": r["uri"], "name": r.get("name", ""), "description": r.get("description", ""),
             "mimeType": r.get("mime_type", "text/plain")}
            for r in self.registry.list("resource", include_disabled=False)
            if "{" not in (r.get("uri") or "")
        ]

    def list_templates(self) -> List[Dict[str, Any]]:
        return [
            {"uriTemplate": r["uri"], "name": r.get("name", ""), "description": r.get("description", "")}
            for r in self.registry.list("resource", include_disabled=False)
            if "{" in (r.get("uri") or "")
        ]

    def subscribe(self, session_id: str, uri: str) -> None:
        self._subscriptions.setdefault(uri, set()).add(session_id)

    def unsubscribe(self, session_id: str, uri: str) -> None:
        self._subscriptions.get(uri, set()).discard(session_id)


class CompletionService:
    """completion/complete (reference: services/completion_service.py)."""

    def __init__(self, registry: Registry):
        self.registry = registry

    async def complete(self, ref: Dict[str, Any], argument: Dict[str, Any]) -> Dict[str, Any]:
        name = argument.get("name", "")
        prefix = argument.get("value", "")
        values: List[str] = []
        if ref.get("type") == "ref/prompt":
            prompt = self.registry.find("prompt", ref.get("name", ""))
            if prompt:
                args = (prompt.get("argument_schema") or {}).get("arguments", [])
                for a in args:
                    if a.get("name") == name:
                        values = [v for v in (a.get("suggestions") or []) if str(v).startswith(prefix)]
        elif ref.get("type") == "ref/resource":
            values = [r["uri"] for r in self.registry.list("resource") if r["uri"].startswith(prefix)][:100]
        return {"completion": {"values": values[:100], "total": len(values), "hasMore": len(values) > 100}}


class RootService:
    def __init__(self):
        self._roots: List[Dict[str, str]] = []

    def list_roots(self) -> List[Dict[str, str]]:
        return list(self._roots)

    def add_root(self, uri: str, name: str = "") -> Dict[str, str]:
        root = {"uri": uri, "name": name}
        self._roots.append(root)
        return root

    def remove_root(self, uri: str) -> bool:
        before = len(self._roots)
        self._roots = [r for r in self._roots if r["uri"] != uri]
        return len(self._roots) < before
