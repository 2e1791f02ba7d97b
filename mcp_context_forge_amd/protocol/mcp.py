"""MCP protocol types and constants.

Mirrors the protocol surface of the reference (mcpgateway/config.py:255
PROTOCOL_VERSION '2025-11-25'; mcpgateway/main.py:3962 /protocol/initialize,
:3994 ping, :4057 completion, :4082 sampling) without importing any SDK —
the wire format is plain JSON-RPC over our transports.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

PROTOCOL_VERSION = "2025-11-25"
SUPPORTED_PROTOCOL_VERSIONS = ("2024-11-05", "2025-03-26", "2025-06-18", "2025-11-25")

SERVER_NAME = "mcp-context-forge-amd"
SERVER_VERSION = "0.1.0"


def server_capabilities() -> Dict[str, Any]:
    """Capabilities advertised on initialize (reference: main.py:3962-3993)."""
    return {
        "tools": {"listChanged": True},
        "resources": {"subscribe": True, "listChanged": True},
        "prompts": {"listChanged": True},
        "logging": {},
        "completions": {},
    }


def initialize_result(requested_version: Optional[str] = None) -> Dict[str, Any]:
    version = requested_version if requested_version in SUPPORTED_PROTOCOL_VERSIONS else PROTOCOL_VERSION
    return {
        "protocolVersion": version,
        "capabilities": server_capabilities(),
        "serverInfo": {"name": SERVER_NAME, "version": SERVER_VERSION},
    }


@dataclass
class ToolDef:
    """Wire-level MCP tool definition (tools/list entry)."""

    name: str
    description: str = ""
    inputSchema: Dict[str, Any] = field(default_factory=lambda: {"type": "object"})
    outputSchema: Optional[Dict[str, Any]] = None
    annotations: Optional[Dict[str, Any]] = None

    def to_dict(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {
            "name": self.name,
            "description": self.description,
            "inputSchema": self.inputSchema,
        }
        if self.outputSchema:
            out["outputSchema"] = self.outputSchema
        if self.annotations:
            out["annotations"] = self.annotations
        return out


@dataclass
class ResourceDef:
    uri: str
    name: str = ""
    description: str = ""
    mimeType: str = "text/plain"

    def to_dict(self) -> Dict[str, Any]:
        return {"uri": self.uri, "name": self.name, "description": self.description, "mimeType": self.mimeType}


@dataclass
class PromptDef:
    name: str
    description: str = ""
    arguments: List[Dict[str, Any]] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {"name": self.name, "description": self.description, "arguments": self.arguments}


def text_content(text: str) -> Dict[str, Any]:
    return {"type": "text", "text": text}


def tool_result(content: List[Dict[str, Any]], is_error: bool = False, structured: Optional[Any] = None) -> Dict[str, Any]:
    out: Dict[str, Any] = {"content": content, "isError": is_error}
    if structured is not None:
        out["structuredContent"] = structured
    return out
