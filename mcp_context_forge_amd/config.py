"""Gateway settings.

MI355X-native analog of the reference's pydantic-settings ``Settings``
(reference: mcpgateway/config.py:175, ~1,150 fields). We keep the same
env-var contract for the fields that matter operationally and add the GPU
pipeline knobs that replace the reference's Redis/cache configuration
(reference: mcpgateway/config.py:2891 cache_type — here the caches are
HBM-resident tensors per BASELINE.json).

pydantic-settings is not in the image, so env loading is implemented
directly: every field of :class:`Settings` can be overridden with the
``FORGE_`` prefix (e.g. ``FORGE_PORT=8080``), falling back to the
reference's unprefixed names where they exist (``HOST``, ``PORT``,
``DATABASE_URL``, ``JWT_SECRET_KEY`` ...).
"""

from __future__ import annotations

import json
import os
from typing import Any, List, Optional

from pydantic import BaseModel, Field


class Settings(BaseModel):
    # --- app / network ---
    app_name: str = "MCP Context Forge AMD"
    host: str = "0.0.0.0"
    port: int = 4444
    app_root_path: str = ""
    environment: str = "development"

    # --- auth (reference: mcpgateway/auth.py, config basic_auth_* / jwt_*) ---
    basic_auth_user: str = "admin"
    basic_auth_password: str = "changeme"
    auth_required: bool = True
    jwt_secret_key: str = "my-test-key"
    # accepted verification algorithms (allowlist; reference: auth.py RS*/HS*)
    jwt_accepted_algorithms: List[str] = Field(default_factory=lambda: ["HS256"])
    # RS256 key material: a JWKS from inline JSON, a file path, or a URL
    jwks_inline: str = ""
    jwks_file: str = ""
    jwks_url: str = ""
    # credential-at-rest sealing key, DISTINCT from the JWT signing key so a
    # leaked signing secret cannot decrypt stored upstream credentials
    # (reference: config auth_encryption_secret, separate from jwt_secret_key)
    auth_encryption_secret: str = "my-test-salt"
    jwt_algorithm: str = "HS256"
    jwt_audience: str = "mcpgateway-api"
    jwt_issuer: str = "mcpgateway"
    token_expiry: int = 10080  # minutes
    platform_admin_email: str = "admin@example.com"
    platform_admin_password: str = "changeme"

    # --- database (reference: db.py:122 build_engine) ---
    database_url: str = "sqlite:///./mcp.db"
    db_pool_size: int = 16
    db_max_overflow: int = 8

    # --- protocol ---
    protocol_version: str = "2025-11-25"

    # --- federation (reference: config federation_*, health_check_interval :2766) ---
    federation_enabled: bool = True
    federation_timeout: int = 30
    federation_sync_timeout: int = 60
    health_check_interval: int = 60
    health_check_timeout: int = 10
    unhealthy_threshold: int = 3
    # async registration lifecycle (reference: gateway_service.py:4077-4362)
    gateway_max_retries: int = 8
    gateway_retry_base_s: float = 2.0
    gateway_retry_cap_s: float = 300.0
    gateway_lifecycle_tick_s: float = 1.0
    # DB leader lease for shared-DB multi-process deployments (services/leader.py);
    # collective (torchrun) worlds use rank 0 by construction instead
    leader_election_enabled: bool = False
    leader_lease_ttl_s: float = 15.0
    max_tool_retries: int = 3
    retry_base_delay_ms: int = 100
    retry_max_delay_ms: int = 5000
    retry_jitter: float = 0.25

    # --- transports / sessions ---
    sse_keepalive_interval: int = 30
    session_persistence: bool = False  # DB-backed session continuity (reference: database backend)
    session_ttl: int = 3600
    message_ttl: int = 600
    event_store_max_events: int = 512
    websocket_ping_interval: int = 30

    # --- rate limiting (reference: middleware/rate_limit_middleware.py) ---
    rate_limit_enabled: bool = False
    rate_limit_requests_per_minute: int = 6000
    rate_limit_burst: int = 200

    # --- security / validation (reference: common/validators.py SecurityValidator) ---
    max_request_body_bytes: int = 4 * 1024 * 1024
    max_json_depth: int = 64
    max_string_length: int = 1 * 1024 * 1024
    max_header_bytes: int = 16 * 1024
    security_headers_enabled: bool = True
    cors_allow_origins: List[str] = Field(default_factory=lambda: ["*"])
    skip_ssl_verify: bool = False

    # --- passthrough headers (reference: utils/passthrough_headers.py) ---
    passthrough_headers: List[str] = Field(default_factory=lambda: ["x-tenant-id", "x-request-id"])

    # --- plugins (reference: plugins/config.yaml + PLUGINS_ENABLED) ---
    plugins_enabled: bool = True
    plugin_config_file: str = "plugins/config.yaml"

    # --- metrics / observability ---
    metrics_buffer_flush_interval: float = 60.0
    metrics_buffer_max_size: int = 1000
    otel_enable_observability: bool = False
    # OTLP/HTTP export (services/otel_export.py): collector base URL, e.g.
    # http://otel-collector:4318 — spans ship to {endpoint}/v1/traces
    otel_endpoint: str = ""
    otel_headers: str = ""            # JSON object of extra headers (auth etc.)
    otel_service_name: str = "mcp-context-forge-amd"
    log_level: str = "INFO"

    # --- admin UI / APIs ---
    admin_ui_enabled: bool = True
    admin_api_enabled: bool = True

    # --- GPU pipeline (MI355X-native; replaces cache_type=redis in the reference) ---
    gpu_enabled: bool = True          # auto-falls back to CPU reference path when no HIP device
    gpu_batch_max_requests: int = 8192
    # micro-batch linger: with two batches in flight the pipeline fills
    # itself; measured at the 1000-user knee: 100 µs beats 500 µs by ~12%
    # RPS and ~0.5 ms p50 (profiles/README_r02.md linger sweep)
    # 30 µs measured optimal at the 1000-conn knee after the round-2 host
    # optimizations (the cycle shortened ~2.3×, so less lingering pays;
    # sweeps: 100 µs was best pre-pool, 30 µs wins 248k→261k RPS after)
    gpu_batch_window_us: int = 30
    gpu_feature_dim: int = 4096       # hashed count-vector dim for classifiers/semantic cache
    gpu_classifier_hidden: int = 1024
    gpu_classifier_classes: int = 8
    gpu_semcache_capacity: int = 65536
    gpu_semcache_threshold: float = 0.92
    # two-stage lookup: random-projection sketch dim (0 = full sweep)
    gpu_semcache_sketch_dim: int = 256
    gpu_streams: int = 2              # compute + copy overlap
    gpu_dtype: str = "bf16"

    # --- native HTTP edge (C++ epoll owner loop, transports/native_edge.py) ---
    native_edge_port: int = 0         # 0 = disabled
    native_edge_threads: int = 0      # 0 = auto (min(8, cpus // 2))

    # --- multi-GPU scale-out (reference analog: session_affinity over Redis; here RCCL) ---
    world_size: int = 1
    rank: int = 0
    upstream_shards: int = 64         # upstreams per GPU rank

    # --- multi-worker HTTP edge (transports/edge.py) ---
    edge_socket: str = ""             # owner-side unix socket path ("" = disabled)
    edge_workers: int = 0             # worker processes to spawn from `serve --workers`

    # --- well-known / misc ---
    docs_enabled: bool = True
    version: str = "0.1.0"

    @classmethod
    def from_env(cls, env: Optional[dict] = None) -> "Settings":
        env = dict(os.environ if env is None else env)
        values: dict[str, Any] = {}
        for name, field_info in cls.model_fields.items():
            raw = env.get("FORGE_" + name.upper())
            if raw is None:
                raw = env.get(name.upper())
            if raw is None:
                continue
            ann = field_info.annotation
            try:
                if ann is bool:
                    values[name] = raw.strip().lower() in ("1", "true", "yes", "on")
                elif ann is int:
                    values[name] = int(raw)
                elif ann is float:
                    values[name] = float(raw)
                elif ann == List[str]:
                    values[name] = json.loads(raw) if raw.startswith("[") else [s.strip() for s in raw.split(",")]
                else:
                    values[name] = raw
            except (ValueError, json.JSONDecodeError):
                continue
        return cls(**values)


_settings: Optional[Settings] = None


def get_settings() -> Settings:
    global _settings
    if _settings is None:
        _settings = Settings.from_env()
    return _settings


def set_settings(s: Settings) -> None:
    global _settings
    _settings = s


def reset_settings() -> None:
    global _settings
    _settings = None
