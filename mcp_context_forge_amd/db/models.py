"""Registry ORM models.

Keeps the reference's registry schema shape (reference: mcpgateway/db.py —
Tool :3254, Resource :3667, Prompt :4058, Server :4394, Gateway :4694,
A2AAgent :4901, metrics :2564-2851, SessionRecord :5311, EmailUser :1465,
Role :1154, tokens :5338-5697, StructuredLogEntry :6141, audit :6624) on
SQLAlchemy 2.0, trimmed to the columns the runtime actually consults.
Durability lives here (SQLite/Postgres); the *hot* lookup structures are
the in-memory registry caches plus the GPU pipeline's native toolmap and
per-tool flag tables (gpu/pipeline.py:_rebuild_tool_meta); HBM holds the
scan banks, classifier weights and semantic-cache key matrix.
"""

from __future__ import annotations

import datetime
import uuid
from typing import Any, Optional

from sqlalchemy import JSON, Boolean, DateTime, Float, ForeignKey, Integer, String, Text
from sqlalchemy.orm import DeclarativeBase, Mapped, mapped_column


def _uuid() -> str:
    return uuid.uuid4().hex


def utcnow() -> datetime.datetime:
    return datetime.datetime.now(datetime.timezone.utc).replace(tzinfo=None)


class Base(DeclarativeBase):
    pass


class DbGateway(Base):
    """Federated peer gateway / upstream MCP server (reference: db.py:4694)."""

    __tablename__ = "gateways"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    name: Mapped[str] = mapped_column(String(255), unique=True, index=True)
    url: Mapped[str] = mapped_column(String(767))
    description: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    transport: Mapped[str] = mapped_column(String(32), default="streamablehttp")  # sse | streamablehttp
    enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    reachable: Mapped[bool] = mapped_column(Boolean, default=True)
    status: Mapped[str] = mapped_column(String(32), default="active")  # pending|active|unreachable|deleting
    auth_type: Mapped[Optional[str]] = mapped_column(String(32), nullable=True)  # basic|bearer|headers
    auth_value: Mapped[Optional[str]] = mapped_column(Text, nullable=True)  # encrypted blob
    capabilities: Mapped[dict] = mapped_column(JSON, default=dict)
    tags: Mapped[list] = mapped_column(JSON, default=list)
    passthrough_headers: Mapped[list] = mapped_column(JSON, default=list)
    consecutive_failures: Mapped[int] = mapped_column(Integer, default=0)
    last_seen: Mapped[Optional[datetime.datetime]] = mapped_column(DateTime, nullable=True)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    updated_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, onupdate=utcnow)
    owner_rank: Mapped[int] = mapped_column(Integer, default=0)  # MI355X: GPU rank that owns this upstream shard
    # async lifecycle state machine (reference: _process_gateway_lifecycle_*
    # db rows claimed by the loop, gateway_service.py:4077-4362)
    retry_count: Mapped[int] = mapped_column(Integer, default=0)
    next_retry_at: Mapped[Optional[float]] = mapped_column(Float, nullable=True)  # epoch seconds
    last_error: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    failure_class: Mapped[Optional[str]] = mapped_column(String(32), nullable=True)


class DbTool(Base):
    """Registered tool (reference: db.py:3254)."""

    __tablename__ = "tools"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    original_name: Mapped[str] = mapped_column(String(255))
    custom_name: Mapped[Optional[str]] = mapped_column(String(255), nullable=True)
    # qualified name exposed to clients: "<gateway-slug>-<original_name>" for federated tools
    name: Mapped[str] = mapped_column(String(255), unique=True, index=True)
    url: Mapped[Optional[str]] = mapped_column(String(767), nullable=True)
    description: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    integration_type: Mapped[str] = mapped_column(String(16), default="MCP")  # MCP | REST | A2A | GRPC | LOCAL
    request_type: Mapped[str] = mapped_column(String(16), default="POST")
    input_schema: Mapped[dict] = mapped_column(JSON, default=lambda: {"type": "object"})
    output_schema: Mapped[Optional[dict]] = mapped_column(JSON, nullable=True)
    annotations: Mapped[Optional[dict]] = mapped_column(JSON, nullable=True)
    headers: Mapped[dict] = mapped_column(JSON, default=dict)
    auth_type: Mapped[Optional[str]] = mapped_column(String(32), nullable=True)
    auth_value: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    jsonpath_filter: Mapped[Optional[str]] = mapped_column(String(512), nullable=True)
    enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    reachable: Mapped[bool] = mapped_column(Boolean, default=True)
    gateway_id: Mapped[Optional[str]] = mapped_column(String(36), ForeignKey("gateways.id"), nullable=True, index=True)
    tags: Mapped[list] = mapped_column(JSON, default=list)
    visibility: Mapped[str] = mapped_column(String(16), default="public")
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    updated_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, onupdate=utcnow)


class DbResource(Base):
    """Registered resource (reference: db.py:3667)."""

    __tablename__ = "resources"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    uri: Mapped[str] = mapped_column(String(767), unique=True, index=True)
    name: Mapped[str] = mapped_column(String(255))
    description: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    mime_type: Mapped[str] = mapped_column(String(128), default="text/plain")
    template: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    content: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    gateway_id: Mapped[Optional[str]] = mapped_column(String(36), ForeignKey("gateways.id"), nullable=True, index=True)
    tags: Mapped[list] = mapped_column(JSON, default=list)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    updated_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, onupdate=utcnow)


class DbPrompt(Base):
    """Registered prompt template (reference: db.py:4058)."""

    __tablename__ = "prompts"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    name: Mapped[str] = mapped_column(String(255), unique=True, index=True)
    description: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    template: Mapped[str] = mapped_column(Text, default="")
    argument_schema: Mapped[dict] = mapped_column(JSON, default=dict)
    enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    gateway_id: Mapped[Optional[str]] = mapped_column(String(36), ForeignKey("gateways.id"), nullable=True, index=True)
    tags: Mapped[list] = mapped_column(JSON, default=list)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    updated_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, onupdate=utcnow)


class DbServer(Base):
    """Virtual server composing tools/resources/prompts (reference: db.py:4394)."""

    __tablename__ = "servers"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    name: Mapped[str] = mapped_column(String(255), unique=True, index=True)
    description: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    icon: Mapped[Optional[str]] = mapped_column(String(767), nullable=True)
    enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    associated_tools: Mapped[list] = mapped_column(JSON, default=list)       # tool ids
    associated_resources: Mapped[list] = mapped_column(JSON, default=list)
    associated_prompts: Mapped[list] = mapped_column(JSON, default=list)
    tags: Mapped[list] = mapped_column(JSON, default=list)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    updated_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, onupdate=utcnow)


class DbA2AAgent(Base):
    """A2A agent registration (reference: db.py:4901)."""

    __tablename__ = "a2a_agents"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    name: Mapped[str] = mapped_column(String(255), unique=True, index=True)
    slug: Mapped[str] = mapped_column(String(255), index=True)
    description: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    endpoint_url: Mapped[str] = mapped_column(String(767))
    agent_type: Mapped[str] = mapped_column(String(32), default="generic")
    protocol_version: Mapped[str] = mapped_column(String(16), default="1.0")
    capabilities: Mapped[dict] = mapped_column(JSON, default=dict)
    config: Mapped[dict] = mapped_column(JSON, default=dict)
    auth_type: Mapped[Optional[str]] = mapped_column(String(32), nullable=True)
    auth_value: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    reachable: Mapped[bool] = mapped_column(Boolean, default=True)
    tags: Mapped[list] = mapped_column(JSON, default=list)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    updated_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, onupdate=utcnow)


class DbUser(Base):
    """Email-auth user (reference: db.py:1465 EmailUser)."""

    __tablename__ = "email_users"

    email: Mapped[str] = mapped_column(String(255), primary_key=True)
    password_hash: Mapped[str] = mapped_column(Text)
    full_name: Mapped[Optional[str]] = mapped_column(String(255), nullable=True)
    is_admin: Mapped[bool] = mapped_column(Boolean, default=False)
    is_active: Mapped[bool] = mapped_column(Boolean, default=True)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)


class DbTeam(Base):
    """Multi-tenancy team (reference: db.py:1931 EmailTeam)."""

    __tablename__ = "email_teams"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    name: Mapped[str] = mapped_column(String(255), unique=True)
    slug: Mapped[str] = mapped_column(String(255), unique=True)
    description: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    is_personal: Mapped[bool] = mapped_column(Boolean, default=False)
    created_by: Mapped[Optional[str]] = mapped_column(String(255), nullable=True)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)


class DbTeamMember(Base):
    __tablename__ = "email_team_members"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    team_id: Mapped[str] = mapped_column(String(36), ForeignKey("email_teams.id"), index=True)
    user_email: Mapped[str] = mapped_column(String(255), index=True)
    role: Mapped[str] = mapped_column(String(32), default="member")  # owner|member


class DbRole(Base):
    """RBAC role (reference: db.py:1154)."""

    __tablename__ = "roles"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    name: Mapped[str] = mapped_column(String(255), unique=True)
    description: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    permissions: Mapped[list] = mapped_column(JSON, default=list)  # ["tools.invoke", "*", ...]
    scope: Mapped[str] = mapped_column(String(32), default="global")


class DbUserRole(Base):
    __tablename__ = "user_roles"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    user_email: Mapped[str] = mapped_column(String(255), index=True)
    role_id: Mapped[str] = mapped_column(String(36), ForeignKey("roles.id"))
    scope_id: Mapped[Optional[str]] = mapped_column(String(36), nullable=True)


class DbApiToken(Base):
    """API token (reference: db.py:5338-5697 token catalog, hashed+revocable)."""

    __tablename__ = "email_api_tokens"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    user_email: Mapped[str] = mapped_column(String(255), index=True)
    name: Mapped[str] = mapped_column(String(255))
    token_hash: Mapped[str] = mapped_column(String(128), unique=True, index=True)
    scopes: Mapped[list] = mapped_column(JSON, default=list)
    server_id: Mapped[Optional[str]] = mapped_column(String(36), nullable=True)
    expires_at: Mapped[Optional[datetime.datetime]] = mapped_column(DateTime, nullable=True)
    revoked: Mapped[bool] = mapped_column(Boolean, default=False)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    last_used_at: Mapped[Optional[datetime.datetime]] = mapped_column(DateTime, nullable=True)


class DbOAuthToken(Base):
    """Stored upstream OAuth tokens (reference: token_storage_service.py +
    db.py:5338 OAuth tables). Token material is sealed with the
    auth_encryption_secret before it reaches this row."""

    __tablename__ = "oauth_tokens"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    storage_key: Mapped[str] = mapped_column(String(255), unique=True, index=True)
    access_token: Mapped[str] = mapped_column(Text)           # sealed
    refresh_token: Mapped[Optional[str]] = mapped_column(Text, nullable=True)  # sealed
    token_type: Mapped[str] = mapped_column(String(32), default="Bearer")
    scopes: Mapped[list] = mapped_column(JSON, default=list)
    expires_at: Mapped[Optional[float]] = mapped_column(Float, nullable=True)  # epoch seconds
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    updated_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, onupdate=utcnow)


class DbTokenUsage(Base):
    """Hourly-bucketed per-credential usage (reference: TokenUsageLog
    db.py:5584 + TokenUsageMiddleware — here aggregated in memory and
    flushed, same shape as the metrics buffer)."""

    __tablename__ = "token_usage"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    bucket: Mapped[datetime.datetime] = mapped_column(DateTime, index=True)
    credential: Mapped[str] = mapped_column(String(255), index=True)  # token id or auth method
    user_email: Mapped[Optional[str]] = mapped_column(String(255), nullable=True)
    requests: Mapped[int] = mapped_column(Integer, default=0)


class DbSessionRecord(Base):
    """Transport session (reference: db.py:5311) — DB backend of the session registry."""

    __tablename__ = "mcp_sessions"

    session_id: Mapped[str] = mapped_column(String(64), primary_key=True)
    transport: Mapped[str] = mapped_column(String(32), default="sse")
    owner_rank: Mapped[int] = mapped_column(Integer, default=0)
    server_id: Mapped[Optional[str]] = mapped_column(String(36), nullable=True)
    user_email: Mapped[Optional[str]] = mapped_column(String(255), nullable=True)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    last_accessed: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)


class DbSessionMessage(Base):
    """Cross-worker session message (reference: db.py:5324)."""

    __tablename__ = "mcp_session_messages"

    id: Mapped[int] = mapped_column(Integer, primary_key=True, autoincrement=True)
    session_id: Mapped[str] = mapped_column(String(64), index=True)
    message: Mapped[str] = mapped_column(Text)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)


class DbToolMetric(Base):
    """Raw per-invocation metric row (reference: db.py:2564)."""

    __tablename__ = "tool_metrics"

    id: Mapped[int] = mapped_column(Integer, primary_key=True, autoincrement=True)
    tool_id: Mapped[str] = mapped_column(String(36), index=True)
    timestamp: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, index=True)
    response_time_ms: Mapped[float] = mapped_column(Float)
    is_success: Mapped[bool] = mapped_column(Boolean, default=True)
    error_message: Mapped[Optional[str]] = mapped_column(Text, nullable=True)
    count: Mapped[int] = mapped_column(Integer, default=1)  # batched rows (revision 0002)


class DbMetricRollup(Base):
    """Hourly rollup (reference: db.py:2851 hourly metrics)."""

    __tablename__ = "metric_rollups_hourly"

    id: Mapped[int] = mapped_column(Integer, primary_key=True, autoincrement=True)
    entity_type: Mapped[str] = mapped_column(String(32), index=True)  # tool|gateway|a2a|prompt|resource
    entity_id: Mapped[str] = mapped_column(String(36), index=True)
    hour: Mapped[datetime.datetime] = mapped_column(DateTime, index=True)
    count: Mapped[int] = mapped_column(Integer, default=0)
    error_count: Mapped[int] = mapped_column(Integer, default=0)
    total_ms: Mapped[float] = mapped_column(Float, default=0.0)
    min_ms: Mapped[float] = mapped_column(Float, default=0.0)
    max_ms: Mapped[float] = mapped_column(Float, default=0.0)


class DbObservabilitySpan(Base):
    """Self-hosted trace span (reference: db.py:2857-3105 Observability*)."""

    __tablename__ = "observability_spans"

    id: Mapped[int] = mapped_column(Integer, primary_key=True, autoincrement=True)
    trace_id: Mapped[str] = mapped_column(String(64), index=True)
    span_id: Mapped[str] = mapped_column(String(32), index=True)
    parent_span_id: Mapped[Optional[str]] = mapped_column(String(32), nullable=True)
    name: Mapped[str] = mapped_column(String(255))
    start_ns: Mapped[int] = mapped_column(Integer)
    end_ns: Mapped[int] = mapped_column(Integer)
    attributes: Mapped[dict] = mapped_column(JSON, default=dict)
    status: Mapped[str] = mapped_column(String(16), default="OK")


class DbAuditLog(Base):
    """Audit trail (reference: db.py:6624)."""

    __tablename__ = "audit_logs"

    id: Mapped[int] = mapped_column(Integer, primary_key=True, autoincrement=True)
    timestamp: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, index=True)
    actor: Mapped[Optional[str]] = mapped_column(String(255), nullable=True)
    action: Mapped[str] = mapped_column(String(64))
    entity_type: Mapped[Optional[str]] = mapped_column(String(32), nullable=True)
    entity_id: Mapped[Optional[str]] = mapped_column(String(64), nullable=True)
    detail: Mapped[dict] = mapped_column(JSON, default=dict)


class DbStructuredLog(Base):
    """Structured log persistence (reference: db.py:6141)."""

    __tablename__ = "structured_logs"

    id: Mapped[int] = mapped_column(Integer, primary_key=True, autoincrement=True)
    timestamp: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, index=True)
    level: Mapped[str] = mapped_column(String(16), index=True)
    logger: Mapped[str] = mapped_column(String(128))
    message: Mapped[str] = mapped_column(Text)
    context: Mapped[dict] = mapped_column(JSON, default=dict)


class DbPluginBinding(Base):
    """Per-tool plugin binding: attach/override a plugin for one tool
    (reference: db.py:6875 tool_plugin_bindings + routers/tool_plugin_bindings.py).

    `name` is the composite unique key "<tool_name>::<plugin_name>"."""

    __tablename__ = "plugin_bindings"

    id: Mapped[str] = mapped_column(String(36), primary_key=True, default=_uuid)
    name: Mapped[str] = mapped_column(String(512), unique=True, index=True)
    tool_name: Mapped[str] = mapped_column(String(255), index=True)
    plugin_name: Mapped[str] = mapped_column(String(255))
    mode: Mapped[Optional[str]] = mapped_column(String(32), nullable=True)  # enforce|permissive|disabled|None=keep
    config: Mapped[Optional[dict]] = mapped_column(JSON, nullable=True)     # per-tool config override
    enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    created_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow)
    updated_at: Mapped[datetime.datetime] = mapped_column(DateTime, default=utcnow, onupdate=utcnow)


class DbGlobalConfig(Base):
    """Runtime-mutable global config row (reference: db.py:2550 GlobalConfig)."""

    __tablename__ = "global_config"

    id: Mapped[int] = mapped_column(Integer, primary_key=True, default=1)
    passthrough_headers: Mapped[list] = mapped_column(JSON, default=list)
    plugins_enabled: Mapped[bool] = mapped_column(Boolean, default=True)
    settings: Mapped[dict] = mapped_column(JSON, default=dict)
