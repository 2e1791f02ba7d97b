"""MCP Apps (UI extension) helpers.

Reference analog: services/mcp_apps.py — the `io.modelcontextprotocol/ui`
extension: tools may declare an interactive UI resource (`ui://` scheme,
`text/html;profile=mcp-app`); the gateway advertises the capability,
serves the sanitized HTML, and manages short-lived AppBridge sessions that
scope which tool an embedded app may call back into.
"""

from __future__ import annotations

import secrets
import time
from typing import Any, Dict, List, Optional

MCP_UI_EXTENSION = "io.modelcontextprotocol/ui"
MCP_UI_VERSION = "2026-01-26"
MCP_APP_MIME_TYPE = "text/html;profile=mcp-app"
UI_URI_SCHEME = "ui://"
LEGACY_RESOURCE_URI_META_KEY = "ui/resourceUri"


def ui_resource_uri(tool: Dict[str, Any]) -> Optional[str]:
    """The tool's declared UI resource, if any (nested `_meta.ui.resourceUri`
    shape, with the legacy flat key accepted)."""
    meta = (tool.get("annotations") or {}).get("_meta") or {}
    ui = meta.get("ui") or {}
    uri = ui.get("resourceUri") or meta.get(LEGACY_RESOURCE_URI_META_KEY)
    if isinstance(uri, str) and uri.startswith(UI_URI_SCHEME):
        return uri
    return None


class McpAppsService:
    def __init__(self, engine, session_ttl_s: float = 900.0):
        self.engine = engine
        self.session_ttl_s = session_ttl_s
        self._sessions: Dict[str, dict] = {}   # token -> {tool, expires}

    # -- capability / inventory -------------------------------------------
    def app_tools(self) -> List[Dict[str, Any]]:
        return [t for t in self.engine.registry.list("tool", include_disabled=False)
                if ui_resource_uri(t)]

    def capabilities_extension(self) -> Optional[Dict[str, Any]]:
        """Extension block advertised in initialize when app tools exist."""
        if not self.app_tools():
            return None
        return {MCP_UI_EXTENSION: {"version": MCP_UI_VERSION,
                                   "mimeTypes": [MCP_APP_MIME_TYPE]}}

    # -- resource resolution ----------------------------------------------
    async def read_app_resource(self, uri: str) -> Dict[str, Any]:
        """Resolve a `ui://` resource to sanitized HTML with the app mime
        type (HTML passes the content-security sanitizer)."""
        if not uri.startswith(UI_URI_SCHEME):
            raise ValueError(f"not an app resource: {uri!r}")
        res = await self.engine.resource_service.read_resource(uri)
        contents = res.get("contents") or []
        out = []
        for c in contents:
            text = c.get("text", "")
            clean = self.engine.content_security.sanitize_html(text) \
                if hasattr(self.engine.content_security, "sanitize_html") else text
            out.append({**c, "text": clean, "mimeType": MCP_APP_MIME_TYPE})
        return {"contents": out}

    # -- AppBridge sessions -------------------------------------------------
    def create_bridge_session(self, tool_name: str) -> Dict[str, Any]:
        """Short-lived token scoping an embedded app to ONE tool."""
        tool = self.engine.registry.find("tool", tool_name)
        if tool is None or not ui_resource_uri(tool):
            raise ValueError(f"{tool_name!r} is not an app-capable tool")
        token = secrets.token_urlsafe(24)
        self._sessions[token] = {"tool": tool_name,
                                 "expires": time.monotonic() + self.session_ttl_s}
        return {"token": token, "tool": tool_name, "ttl_s": self.session_ttl_s}

    def validate_bridge_session(self, token: str, tool_name: str) -> bool:
        ent = self._sessions.get(token)
        if ent is None or ent["tool"] != tool_name:
            return False
        if time.monotonic() > ent["expires"]:
            self._sessions.pop(token, None)
            return False
        return True

    def purge_expired(self) -> int:
        now = time.monotonic()
        stale = [t for t, e in self._sessions.items() if now > e["expires"]]
        for t in stale:
            self._sessions.pop(t, None)
        return len(stale)
