"""API request schemas (reference: mcpgateway/schemas.py, 9,189 LoC of
pydantic models). Scoped to the mutating surface: every create/update body
validates against a typed model before it reaches the registry, so field
typos and type errors answer 422 with field-level detail instead of
leaking into storage. Extra fields are allowed (the registry's column
filter drops unknowns) except where the reference forbids them."""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from pydantic import BaseModel, Field, field_validator

_NAME_MAX = 255


class _Entity(BaseModel):
    model_config = {"extra": "allow"}


class ToolCreate(_Entity):
    name: str = Field(min_length=1, max_length=_NAME_MAX)
    description: str = ""
    url: Optional[str] = None
    integration_type: Optional[str] = None
    request_type: Optional[str] = None
    input_schema: Optional[Dict[str, Any]] = None
    output_schema: Optional[Dict[str, Any]] = None
    headers: Optional[Dict[str, str]] = None
    annotations: Optional[Dict[str, Any]] = None
    tags: List[str] = Field(default_factory=list)
    enabled: bool = True

    @field_validator("integration_type")
    @classmethod
    def _itype(cls, v):
        if v is not None and v not in ("LOCAL", "MCP", "REST", "A2A", "GRPC"):
            raise ValueError(f"integration_type must be LOCAL|MCP|REST|A2A|GRPC, got {v!r}")
        return v

    @field_validator("request_type")
    @classmethod
    def _rtype(cls, v):
        if v is not None and v.upper() not in ("GET", "POST", "PUT", "PATCH", "DELETE"):
            raise ValueError(f"invalid request_type {v!r}")
        return v


class GatewayCreate(_Entity):
    name: str = Field(min_length=1, max_length=_NAME_MAX)
    url: str = ""
    transport: str = "streamablehttp"
    description: str = ""
    auth_type: Optional[str] = None
    auth_value: Optional[Any] = None
    tags: List[str] = Field(default_factory=list)
    defer: bool = False

    @field_validator("transport")
    @classmethod
    def _transport(cls, v):
        if v not in ("streamablehttp", "sse"):
            raise ValueError(f"transport must be streamablehttp|sse, got {v!r}")
        return v

    @field_validator("auth_type")
    @classmethod
    def _auth(cls, v):
        if v is not None and v not in ("basic", "bearer", "headers", "oauth"):
            raise ValueError(f"auth_type must be basic|bearer|headers|oauth, got {v!r}")
        return v


class ServerCreate(_Entity):
    name: str = Field(min_length=1, max_length=_NAME_MAX)
    description: str = ""
    associated_tools: List[str] = Field(default_factory=list)
    tags: List[str] = Field(default_factory=list)
    enabled: bool = True


class ResourceCreate(_Entity):
    uri: str = Field(min_length=1, max_length=767)
    name: Optional[str] = None
    description: str = ""
    mime_type: str = "text/plain"
    content: Optional[str] = None
    template: Optional[str] = None
    tags: List[str] = Field(default_factory=list)


class PromptCreate(_Entity):
    name: str = Field(min_length=1, max_length=_NAME_MAX)
    description: str = ""
    template: Optional[str] = None
    argument_schema: Optional[Dict[str, Any]] = None
    tags: List[str] = Field(default_factory=list)


class A2AAgentCreate(_Entity):
    name: str = Field(min_length=1, max_length=_NAME_MAX)
    endpoint_url: str = ""
    agent_type: str = "generic"
    protocol_version: str = "1.0"
    description: str = ""
    auth_type: Optional[str] = None
    auth_value: Optional[Any] = None
    config: Optional[Dict[str, Any]] = None
    tags: List[str] = Field(default_factory=list)


CREATE_SCHEMAS: Dict[str, type] = {
    "tool": ToolCreate,
    "gateway": GatewayCreate,
    "server": ServerCreate,
    "resource": ResourceCreate,
    "prompt": PromptCreate,
    "a2a_agent": A2AAgentCreate,
}


def validate_create(kind: str, body: Dict[str, Any]) -> Dict[str, Any]:
    """Validate a create body; returns the normalized dict.
    Raises pydantic.ValidationError (callers map to 422)."""
    model = CREATE_SCHEMAS.get(kind)
    if model is None:
        return body
    return model(**body).model_dump(exclude_none=True)
