"""GatewayEngine — composition root + JSON-RPC method dispatch.

Reference analogs: mcpgateway/main.py lifespan (:1443) wires ~60 service
singletons; `_handle_rpc_authenticated` (:11197) is the method dispatch
matrix (initialize / ping / tools/list / tools/call / resources/* /
prompts/* / completion/complete / logging/setLevel / notifications/*).

The engine is transport-agnostic: HTTP (/rpc, /mcp), SSE, WS, stdio and the
bench harness all feed `handle_rpc_bytes` / `handle_rpc`. When a GPU is
present the batched entry point `process_rpc_batch` stages the whole batch
through the HIP plugin pipeline (gpu/pipeline.py) before fan-out.
"""

from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import Any, Dict, List, Optional

from .config import Settings, get_settings
from .db.engine import Database
from .plugins.framework import PluginManager
from .plugins.loader import default_chain_specs, load_plugin_manager
from .protocol import jsonrpc
from .protocol.mcp import PROTOCOL_VERSION, initialize_result
from .registry.registry import NotFoundError, Registry
from .services.a2a_service import A2AService
from .services.content import CompletionService, PromptService, ResourceService, RootService
from .services.gateway_service import GatewayService
from .services.governance import CatalogService, ContentSecurity, PasswordPolicy, TokenBlocklist
from .services.llm_proxy import LLMProxyError, LLMProxyService
from .services.metrics import MetricsBuffer
from .services.observability import AuditTrail, ComplianceService, ObservabilityService, SiemExporter
from .services.sessions import CancellationService, ElicitationService, SessionRegistry
from .services.tool_service import ToolInvocationError, ToolNotFoundError, ToolService

logger = logging.getLogger(__name__)


class GatewayEngine:
    def __init__(self, settings: Optional[Settings] = None, plugin_manager: Optional[PluginManager] = None,
                 rank: int = 0, world_size: int = 1):
        self.settings = settings or get_settings()
        self.rank = rank
        self.world_size = world_size
        self.db = Database(self.settings.database_url, self.settings.db_pool_size, self.settings.db_max_overflow)
        self.db.migrate()
        self.registry = Registry(self.db)
        if plugin_manager is not None:
            self.plugins = plugin_manager
        elif self.settings.plugins_enabled:
            import os

            if os.path.exists(self.settings.plugin_config_file):
                self.plugins = load_plugin_manager(self.settings.plugin_config_file)
            else:
                self.plugins = load_plugin_manager(specs=default_chain_specs())
        else:
            self.plugins = PluginManager([], enabled=False)
        self.metrics = MetricsBuffer(self.db, self.settings.metrics_buffer_flush_interval,
                                     self.settings.metrics_buffer_max_size)
        from .services.metrics import TokenUsageTracker

        self.token_usage = TokenUsageTracker(self.db)
        self.tool_service = ToolService(self.registry, self.plugins, self.metrics,
                                        max_retries=self.settings.max_tool_retries)
        self.gateway_service = GatewayService(self.registry, self.tool_service, self.settings, rank, world_size)
        self.a2a_service = A2AService(self.registry, self.plugins)
        self.tool_service.a2a_service = self.a2a_service
        self.prompt_service = PromptService(self.registry, self.plugins)
        self.resource_service = ResourceService(self.registry, self.plugins)
        self.completion_service = CompletionService(self.registry)
        self.root_service = RootService()
        self.llm_proxy = LLMProxyService()
        from .services.chat_service import McpChatService

        self.chat = McpChatService(self)
        from .services.diagnostics import PerformanceService, SupportBundle, ToolOps

        self.support_bundle = SupportBundle(self)
        self.performance = PerformanceService(self)
        self.toolops = ToolOps(self)
        self.observability = ObservabilityService(self.db)
        if self.settings.otel_endpoint:
            # OTLP/HTTP export to a collector (reference: init_telemetry :970)
            from .services.otel_export import OtlpHttpExporter

            hdrs = {}
            if self.settings.otel_headers:
                try:
                    hdrs = json.loads(self.settings.otel_headers)
                except ValueError:
                    logger.warning("otel_headers is not valid JSON; ignoring")
            self.observability.exporter = OtlpHttpExporter(
                self.settings.otel_endpoint, headers=hdrs,
                service_name=self.settings.otel_service_name)
        self.catalog = CatalogService()
        from .services.governance import TagService

        self.tags = TagService(self.registry)
        from .services.mcp_apps import McpAppsService

        self.mcp_apps = McpAppsService(self)
        from .services.governance import ServerClassificationService

        self.classification = ServerClassificationService(self.registry)
        self.password_policy = PasswordPolicy()
        self.token_blocklist = TokenBlocklist()
        self.content_security = ContentSecurity()
        self.audit = AuditTrail(self.db)
        self.siem = SiemExporter(self.db)
        self.compliance = ComplianceService(self)
        self.sessions = SessionRegistry(self.settings.session_ttl, rank, world_size=world_size,
                                        db=self.db if self.settings.session_persistence else None)
        self.cancellations = CancellationService()
        self.elicitation = ElicitationService(self.sessions)
        self.sync_plugin_bindings()
        self.started_at = time.time()
        self.gpu_pipeline = None  # attached lazily by gpu.pipeline when available
        self.bus = None           # RcclBus when running multi-rank (parallel/runtime.py)
        self.forward_rpc = None   # set by DistributedGateway: (dest, raw, ...) -> bytes
        self.forward_rpc_batch = None  # (dest, raws, users) -> List[bytes]
        # cross-rank tool ownership (reference analog: every gunicorn worker
        # sees every tool via the shared DB; here each rank shards upstreams
        # and peers learn name -> owner_rank via sync_tool_ownership)
        self.foreign_tools: Dict[str, int] = {}
        self._log_level = "info"
        self._maintenance_task = None
        self._maintenance_stop = None
        self.leader_elector = None  # DbLeaderElector when leader_election_enabled

    # -- lifecycle ---------------------------------------------------------------
    async def startup(self) -> None:
        if self.settings.leader_election_enabled:
            # shared-DB multi-process deployment: background singletons
            # (health + lifecycle loops) run only on the lease holder
            # (reference: Redis leader election :1254; services/leader.py)
            from .services.leader import DbLeaderElector

            self.leader_elector = DbLeaderElector(self.db, ttl_s=self.settings.leader_lease_ttl_s)
            await self.leader_elector.start()
            self.gateway_service.leader_check = lambda: self.leader_elector.is_leader
        if self.settings.federation_enabled and self.settings.health_check_interval > 0:
            await self.gateway_service.start_health_loop()
        if self.settings.federation_enabled:
            # resume pending/deleting lifecycle rows left from a previous run
            self.gateway_service.ensure_lifecycle_loop()
        # periodic maintenance (reference: lifespan background tasks — metrics
        # buffer flush :60s, session reaper, span persistence)
        if self._maintenance_task is None:
            self._maintenance_stop = asyncio.Event()
            self._maintenance_task = asyncio.create_task(self._maintenance_loop())

    async def _maintenance_loop(self) -> None:
        interval = max(5.0, min(60.0, self.settings.metrics_buffer_flush_interval))
        while not self._maintenance_stop.is_set():
            try:
                await asyncio.wait_for(self._maintenance_stop.wait(), timeout=interval)
                return
            except asyncio.TimeoutError:
                pass
            try:
                self.metrics.maybe_flush()
                self.token_usage.flush()
                self.sessions.cleanup_expired()
                self.observability.flush()
                self.token_blocklist.purge_expired()
            except Exception:  # pragma: no cover - defensive
                logger.exception("maintenance loop error")

    async def shutdown(self) -> None:
        if self._maintenance_task is not None:
            self._maintenance_stop.set()
            self._maintenance_task.cancel()
            try:
                await self._maintenance_task
            except (asyncio.CancelledError, Exception):
                pass
            self._maintenance_task = None
        if self.leader_elector is not None:
            await self.leader_elector.stop()
            self.leader_elector = None
        await self.gateway_service.stop()
        await self.a2a_service.aclose()
        await self.llm_proxy.aclose()
        await self.tool_service.aclose()
        await self.plugins.shutdown()
        self.observability.flush()
        self.metrics.flush()
        self.db.close()

    def enable_gpu(self) -> bool:
        """Attach the HIP batch pipeline if a device is present. Returns success."""
        if self.gpu_pipeline is not None:
            return True
        if not self.settings.gpu_enabled:
            return False
        try:
            import torch

            if not torch.cuda.is_available():
                return False
            from .gpu.pipeline import GpuPluginPipeline

            self.gpu_pipeline = GpuPluginPipeline(self)
            return True
        except Exception as exc:  # pragma: no cover - GPU-only path
            logger.warning("GPU pipeline unavailable: %s", exc)
            return False

    # -- RPC dispatch (reference: main.py:11330-11537 method chain) ----------------
    async def handle_rpc(self, req: jsonrpc.JSONRPCRequest, user: Optional[str] = None,
                         server_id: Optional[str] = None, session: Optional[Any] = None,
                         headers: Optional[Dict[str, str]] = None) -> Optional[jsonrpc.JSONRPCResponse]:
        method = req.method
        params = req.params if isinstance(req.params, dict) else {}
        rid = req.id
        try:
            if method == "initialize":
                if session is not None:
                    session.initialized = True
                    session.protocol_version = params.get("protocolVersion")
                result = initialize_result(params.get("protocolVersion"))
                ext = self.mcp_apps.capabilities_extension()
                if ext:
                    result["capabilities"]["extensions"] = ext
            elif method == "notifications/cancelled":
                # reference: cancellation_router — cancel the in-flight task
                if session is not None and params.get("requestId") is not None:
                    self.cancellations.cancel(session.session_id, params["requestId"])
                return None
            elif method in ("ping", "notifications/initialized", "notifications/roots/list_changed"):
                if method != "ping":
                    return None  # notifications get no response
                result = {}
            elif method == "tools/list":
                tools = await self.tool_service.list_tools(server_id=server_id)
                result = {
                    "tools": [
                        {
                            "name": t["name"],
                            "description": t.get("description") or "",
                            "inputSchema": t.get("input_schema") or {"type": "object"},
                            **({"outputSchema": t["output_schema"]} if t.get("output_schema") else {}),
                            **({"annotations": t["annotations"]} if t.get("annotations") else {}),
                        }
                        for t in tools
                    ]
                }
            elif method == "tools/call":
                name = params.get("name")
                if not isinstance(name, str) or not name:
                    raise jsonrpc.JSONRPCError(jsonrpc.INVALID_PARAMS, "missing tool name")
                with self.observability.span("tools/call", tool=name, user=user or ""):
                    coro = self.tool_service.invoke_tool(
                        name, params.get("arguments") or {}, user=user, server_id=server_id, headers=headers
                    )
                    if session is not None and rid is not None:
                        # cancellable (MCP notifications/cancelled): run as a
                        # registered task; a cancelled request gets NO response
                        task = asyncio.ensure_future(coro)
                        self.cancellations.register(session.session_id, rid, task)
                        try:
                            result = await task
                        except asyncio.CancelledError:
                            return None
                        finally:
                            self.cancellations.unregister(session.session_id, rid)
                    else:
                        result = await coro
            elif method == "resources/list":
                result = {"resources": self.resource_service.list_resources()}
            elif method == "resources/templates/list":
                result = {"resourceTemplates": self.resource_service.list_templates()}
            elif method == "resources/read":
                uri = params.get("uri")
                if not uri:
                    raise jsonrpc.JSONRPCError(jsonrpc.INVALID_PARAMS, "missing uri")
                result = await self.resource_service.read_resource(uri, user=user)
            elif method == "resources/subscribe":
                if session is not None and params.get("uri"):
                    self.resource_service.subscribe(session.session_id, params["uri"])
                result = {}
            elif method == "resources/unsubscribe":
                if session is not None and params.get("uri"):
                    self.resource_service.unsubscribe(session.session_id, params["uri"])
                result = {}
            elif method == "prompts/list":
                result = {"prompts": self.prompt_service.list_prompts()}
            elif method == "prompts/get":
                name = params.get("name")
                if not name:
                    raise jsonrpc.JSONRPCError(jsonrpc.INVALID_PARAMS, "missing prompt name")
                result = await self.prompt_service.get_prompt(name, params.get("arguments") or {}, user=user)
            elif method == "completion/complete":
                result = await self.completion_service.complete(params.get("ref") or {}, params.get("argument") or {})
            elif method == "roots/list":
                result = {"roots": self.root_service.list_roots()}
            elif method == "logging/setLevel":
                self._log_level = params.get("level", "info")
                result = {}
            elif method == "sampling/createMessage":
                # reference: handlers/sampling.py — forwarded to the LLM proxy
                if not self.llm_proxy.registry.providers:
                    raise jsonrpc.JSONRPCError(jsonrpc.METHOD_NOT_FOUND, "sampling requires a configured LLM provider")
                messages = [
                    {"role": m.get("role", "user"),
                     "content": (m.get("content") or {}).get("text", "") if isinstance(m.get("content"), dict) else str(m.get("content", ""))}
                    for m in params.get("messages", [])
                ]
                if params.get("systemPrompt"):
                    messages.insert(0, {"role": "system", "content": params["systemPrompt"]})
                try:
                    out = await self.llm_proxy.chat_completions({
                        "messages": messages,
                        "max_tokens": params.get("maxTokens", 256),
                        "temperature": params.get("temperature", 1.0),
                    })
                except LLMProxyError as exc:
                    raise jsonrpc.JSONRPCError(jsonrpc.SERVER_ERROR, str(exc)) from exc
                choice = (out.get("choices") or [{}])[0]
                result = {
                    "role": "assistant",
                    "content": {"type": "text", "text": (choice.get("message") or {}).get("content", "")},
                    "model": out.get("model", ""),
                    "stopReason": choice.get("finish_reason", "endTurn"),
                }
            else:
                raise jsonrpc.JSONRPCError(jsonrpc.METHOD_NOT_FOUND, f"Method not found: {method}")
            if req.is_notification:
                return None
            return jsonrpc.result_response(rid, result)
        except jsonrpc.JSONRPCError as exc:
            if req.is_notification:
                return None
            return jsonrpc.JSONRPCResponse(id=rid, error=exc)
        except ToolNotFoundError as exc:
            return jsonrpc.error_response(rid, jsonrpc.INVALID_PARAMS, str(exc))
        except ToolInvocationError as exc:
            return jsonrpc.error_response(rid, exc.code, str(exc))
        except Exception as exc:  # pragma: no cover - defensive
            logger.exception("rpc internal error")
            return jsonrpc.error_response(rid, jsonrpc.INTERNAL_ERROR, str(exc))

    async def handle_rpc_bytes(self, raw: bytes, user: Optional[str] = None,
                               server_id: Optional[str] = None, session: Optional[Any] = None,
                               headers: Optional[Dict[str, str]] = None) -> Optional[bytes]:
        """Bytes-in/bytes-out single request (transport fast path)."""
        try:
            req = jsonrpc.parse_request_bytes(raw)
        except jsonrpc.JSONRPCError as exc:
            # a JSON-RPC *response* object (no method) is a client reply to a
            # server-initiated request — elicitation (reference: elicitation
            # responses ride the same POST channel)
            try:
                obj = json.loads(raw)
            except Exception:
                obj = None
            if isinstance(obj, dict) and "method" not in obj and obj.get("id") is not None \
                    and ("result" in obj or "error" in obj):
                self.elicitation.resolve(obj["id"], obj.get("result"), obj.get("error"))
                return None
            return jsonrpc.JSONRPCResponse(id=None, error=exc).to_bytes()
        # cross-rank routing: a tools/call for a tool owned by a peer rank
        # rides the RCCL bus to its owner (reference: session_affinity
        # forward_request_to_owner :747 over Redis)
        if self.foreign_tools and self.forward_rpc is not None and req.method == "tools/call" \
                and isinstance(req.params, dict):
            dest = self.foreign_tools.get(req.params.get("name") or "")
            if dest is not None and dest != self.rank:
                return await self.forward_rpc(dest, raw, user, server_id)
        resp = await self.handle_rpc(req, user=user, server_id=server_id, session=session, headers=headers)
        return resp.to_bytes() if resp is not None else None

    # -- batched entry point (GPU hot path; bench + micro-batching transports) ----
    async def process_rpc_batch(self, raws: List[bytes], user: Optional[str] = None,
                                server_id: Optional[str] = None,
                                users: Optional[List[Optional[str]]] = None) -> List[Optional[bytes]]:
        """Process a batch of raw JSON-RPC requests.

        With a GPU pipeline attached, tools/call requests take the staged
        HIP plugin chain (parse → scan/mask/validate/classify on device →
        fan-out → post chain); everything else falls through per-request.
        `users` (per-request identities from the micro-batch collector)
        overrides the scalar `user` and scopes cache tenancy per row.
        """
        if self.gpu_pipeline is not None:
            # measured: splitting into two concurrent half-batches LOSES ~9%
            # (host stages are GIL-serialized; to_thread hops cost more than
            # the GPU-sync overlap buys) — one batch at a time is fastest.
            # The C++ stores stay mutex-protected so concurrent callers of
            # process_batch (e.g. collector + direct) remain safe.
            return await self.gpu_pipeline.process_batch(raws, user=user, server_id=server_id, users=users)
        if users is not None:
            return list(await asyncio.gather(
                *(self.handle_rpc_bytes(r, user=u, server_id=server_id) for r, u in zip(raws, users))))
        return list(await asyncio.gather(*(self.handle_rpc_bytes(r, user=user, server_id=server_id) for r in raws)))

    def invalidate_peers(self, what: str = "registry") -> None:
        """Publish an invalidation to peer ranks over the RCCL bus (the
        reference's Redis pub/sub invalidation). No-op single-rank."""
        if self.bus is not None:
            try:
                self.bus.publish({"kind": "invalidate", "what": what})
            except Exception:  # pragma: no cover - bus teardown race
                logger.warning("peer invalidation publish failed", exc_info=True)

    # -- per-tool plugin bindings (reference: routers/tool_plugin_bindings.py) --
    def sync_plugin_bindings(self) -> None:
        """Load registry plugin_binding rows into the plugin manager
        (tool -> plugin -> {mode, config}); bumps plugins.version so the
        GPU pipeline recompiles its per-tool flag/host-chain tables."""
        bmap: Dict[str, Dict[str, Dict[str, Any]]] = {}
        for row in self.registry.list("plugin_binding"):
            if not row.get("enabled", True):
                continue
            bmap.setdefault(row["tool_name"], {})[row["plugin_name"]] = {
                "mode": row.get("mode"), "config": row.get("config")}
        self.plugins.set_bindings(bmap)

    def set_plugin_binding(self, tool_name: str, plugin_name: str,
                           mode: Optional[str] = None, config: Optional[dict] = None) -> Dict[str, Any]:
        key = f"{tool_name}::{plugin_name}"
        existing = self.registry.find("plugin_binding", key)
        if existing is not None:
            ent = self.registry.update("plugin_binding", existing["id"], mode=mode, config=config, enabled=True)
        else:
            ent = self.registry.create("plugin_binding", name=key, tool_name=tool_name,
                                       plugin_name=plugin_name, mode=mode, config=config)
        self.sync_plugin_bindings()
        self.invalidate_peers("plugin_binding")
        return ent

    def delete_plugin_binding(self, tool_name: str, plugin_name: str) -> None:
        key = f"{tool_name}::{plugin_name}"
        existing = self.registry.find("plugin_binding", key)
        if existing is None:
            from .registry.registry import NotFoundError

            raise NotFoundError(f"binding {key!r} not found")
        self.registry.delete("plugin_binding", existing["id"])
        self.sync_plugin_bindings()
        self.invalidate_peers("plugin_binding")

    async def notify_list_changed(self, kind: str) -> None:
        """Fan out listChanged notifications to live sessions (reference:
        services/notification_service.py - upstream listChanged to listeners)."""
        method = {"tool": "notifications/tools/list_changed",
                  "resource": "notifications/resources/list_changed",
                  "prompt": "notifications/prompts/list_changed"}.get(kind)
        if method is None:
            return
        await self.sessions.broadcast_all({"jsonrpc": "2.0", "method": method})

    # -- health/version ------------------------------------------------------------
    def health(self) -> Dict[str, Any]:
        return {"status": "healthy", "uptime_s": round(time.time() - self.started_at, 1)}

    def version_info(self) -> Dict[str, Any]:
        import torch

        return {
            "name": "mcp-context-forge-amd",
            "version": self.settings.version,
            "protocol_version": PROTOCOL_VERSION,
            "rank": self.rank,
            "world_size": self.world_size,
            "gpu": bool(self.gpu_pipeline),
            "torch": torch.__version__,
        }
