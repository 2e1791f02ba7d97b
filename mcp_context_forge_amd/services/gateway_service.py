"""Federation: peer gateway / upstream MCP server lifecycle.

Reference analog: services/gateway_service.py — register_gateway (:1636),
_initialize_gateway (:5008) → connect + initialize + tools/resources/prompts
sync (:5648/:5776/:5889), health loop check_health_of_gateways (:4412) with
bounded concurrency (:4477 limited_check), failure handling (:4362) and
reactivation (:4529). The reference elects a health-check leader via Redis
SET NX (:1254); here rank 0 of the RCCL world is the leader by construction
(parallel/ world), and single-rank deployments just run the loop.
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Any, Dict, List, Optional

from ..config import Settings
from ..registry.registry import ConflictError, NotFoundError, Registry
from ..utils import slugify
from .tool_service import ToolService
from .upstream import HttpUpstreamClient, InProcUpstream, SseUpstreamClient, UpstreamClient, UpstreamError

logger = logging.getLogger(__name__)


class GatewayConnectionError(Exception):
    pass


class GatewayService:
    def __init__(self, registry: Registry, tool_service: ToolService, settings: Optional[Settings] = None,
                 rank: int = 0, world_size: int = 1):
        self.registry = registry
        self.tools = tool_service
        self.settings = settings or Settings()
        self.rank = rank
        self.world_size = world_size
        self._health_task: Optional[asyncio.Task] = None
        self._lifecycle_task: Optional[asyncio.Task] = None
        # shared-DB multi-process mode: background singletons gate on the
        # DB lease (services/leader.py); None = this process is the leader
        # (single process, or rank 0 of a collective world by construction)
        self.leader_check: Optional[Any] = None
        self._injected_clients: Dict[str, UpstreamClient] = {}  # deferred test/bench clients
        self._stop = asyncio.Event()
        from ..auth.crypto import EncryptionService
        from ..auth.oauth import TokenStorage

        # credential material is sealed at rest (reference: EncryptedText db.py:277)
        # keyed by auth_encryption_secret, NOT the JWT signing key
        self.crypto = EncryptionService(self.settings.auth_encryption_secret)
        # upstream OAuth tokens (auth-code flow) persist sealed in the DB
        self.token_storage = TokenStorage(self.registry.db, self.crypto)
        self._oauth_providers: Dict[str, Any] = {}  # gateway_id -> provider

    # -- client construction ----------------------------------------------------
    def _make_client(self, gateway: Dict[str, Any]) -> UpstreamClient:
        import json as _json

        headers: Dict[str, str] = {}
        token_provider = None
        auth_value = self.crypto.open_(gateway.get("auth_value"))
        if gateway.get("auth_type") == "bearer" and auth_value:
            headers["authorization"] = f"Bearer {auth_value}"
        elif gateway.get("auth_type") == "headers" and auth_value:
            try:
                headers.update(_json.loads(auth_value))
            except Exception:
                pass
        elif gateway.get("auth_type") == "oauth" and auth_value:
            # client-credentials OR authorization-code upstream auth
            # (reference: oauth_manager.py both flows)
            from ..auth.oauth import provider_from_auth_value

            cfg = auth_value
            gid = gateway.get("id", gateway.get("name", ""))
            token_provider = self._oauth_providers.get(gid)
            if token_provider is None:
                token_provider = provider_from_auth_value(
                    _json.loads(cfg) if isinstance(cfg, str) else cfg,
                    storage=self.token_storage,
                    storage_key=f"gateway:{gid}",
                    state_secret=self.settings.jwt_secret_key)
                self._oauth_providers[gid] = token_provider
        cls = SseUpstreamClient if gateway.get("transport") == "sse" else HttpUpstreamClient
        return cls(gateway["url"], headers=headers,
                   timeout=self.settings.federation_timeout,
                   token_provider=token_provider)

    # -- failure classification (reference: handshake classifier :7469) --------
    @staticmethod
    def classify_failure(exc: BaseException) -> str:
        if isinstance(exc, asyncio.TimeoutError):
            return "timeout"
        msg = str(exc).lower()
        if isinstance(exc, UpstreamError) or "upstream" in msg:
            if "http 401" in msg or "http 403" in msg:
                return "auth_error"
            if "http 4" in msg or "http 5" in msg:
                return "http_error"
            if "unreachable" in msg or "connect" in msg:
                return "connect_error"
            return "protocol_error"
        if "connect" in msg or "refused" in msg or "resolve" in msg:
            return "connect_error"
        return "internal_error"

    def _retry_backoff(self, retry_count: int) -> float:
        """Exponential backoff with jitter for pending-gateway retries
        (reference: _calculate_gateway_retry_backoff :4332)."""
        import random

        base = self.settings.gateway_retry_base_s
        cap = self.settings.gateway_retry_cap_s
        d = min(cap, base * (2 ** max(0, retry_count - 1)))
        return d * (1.0 + self.settings.retry_jitter * random.random())

    # -- registration (reference: register_gateway :1636) -----------------------
    async def register_gateway(self, name: str, url: str, transport: str = "streamablehttp",
                               description: str = "", auth_type: Optional[str] = None,
                               auth_value: Optional[str] = None, tags: Optional[List[str]] = None,
                               client: Optional[UpstreamClient] = None,
                               owner_rank: Optional[int] = None,
                               defer: bool = False) -> Dict[str, Any]:
        """Gateway registration — sync or async lifecycle.

        Sync (default): connect, initialize, sync capabilities before
        returning; a failure raises and marks the row unreachable.

        `defer=True` (reference: _register_gateway_pending :1564): the row
        is created in status `pending` and returned immediately; the
        lifecycle loop claims it, attempts initialization with exponential
        backoff + failure classification, and flips it to `active` when the
        peer becomes reachable (or `failed` after gateway_max_retries).

        `client` lets tests/bench inject an InProcUpstream (fake upstream
        harness) — the sync + health machinery is identical either way.
        """
        gateway = self.registry.create(
            "gateway",
            name=name,
            url=url,
            transport=transport,
            description=description,
            auth_type=auth_type,
            auth_value=self.crypto.seal(auth_value),
            tags=tags or [],
            status="pending",
            retry_count=0,
            next_retry_at=time.time(),
            owner_rank=owner_rank if owner_rank is not None else (hash(name) % self.world_size),
        )
        if defer:
            if client is not None:
                self._injected_clients[gateway["id"]] = client
            self.ensure_lifecycle_loop()
            return gateway
        try:
            await self._initialize_gateway(gateway, client)
        except Exception as exc:
            self.registry.update("gateway", gateway["id"], status="unreachable", reachable=False,
                                 last_error=str(exc)[:500], failure_class=self.classify_failure(exc))
            raise GatewayConnectionError(f"failed to initialize gateway {name}: {exc}") from exc
        return self.registry.get("gateway", gateway["id"])

    async def _initialize_gateway(self, gateway: Dict[str, Any], client: Optional[UpstreamClient] = None) -> None:
        """Connect + initialize + sync registry (reference: _initialize_gateway :5008)."""
        client = client or self._make_client(gateway)
        init = await asyncio.wait_for(client.initialize(), timeout=self.settings.federation_sync_timeout)
        tools = await client.list_tools()
        resources = await client.list_resources()
        prompts = await client.list_prompts()
        self.tools.attach_upstream(gateway["id"], client)
        self.registry.sync_gateway_tools(gateway, tools)
        self._sync_resources(gateway, resources)
        self._sync_prompts(gateway, prompts)
        self.registry.update(
            "gateway",
            gateway["id"],
            status="active",
            reachable=True,
            consecutive_failures=0,
            capabilities=(init or {}).get("capabilities", {}),
        )

    def _sync_resources(self, gateway: Dict[str, Any], resources: List[Dict[str, Any]]) -> None:
        for r in resources:
            uri = r.get("uri")
            if not uri:
                continue
            existing = self.registry.find("resource", uri)
            fields = dict(uri=uri, name=r.get("name", uri), description=r.get("description", ""),
                          mime_type=r.get("mimeType", "text/plain"), gateway_id=gateway["id"])
            if existing:
                self.registry.update("resource", existing["id"], **fields)
            else:
                self.registry.create("resource", **fields)

    def _sync_prompts(self, gateway: Dict[str, Any], prompts: List[Dict[str, Any]]) -> None:
        for p in prompts:
            name = p.get("name")
            if not name:
                continue
            qname = f"{slugify(gateway['name'])}-{name}"
            existing = self.registry.find("prompt", qname)
            fields = dict(name=qname, description=p.get("description", ""),
                          argument_schema={"arguments": p.get("arguments", [])}, gateway_id=gateway["id"])
            if existing:
                self.registry.update("prompt", existing["id"], **fields)
            else:
                self.registry.create("prompt", **fields)

    # -- health loop (reference: check_health_of_gateways :4412) ----------------
    async def check_health_once(self, concurrency: int = 16) -> Dict[str, bool]:
        if self.leader_check is not None and not self.leader_check():
            return {}  # follower: the lease holder runs the checks
        sem = asyncio.Semaphore(concurrency)
        results: Dict[str, bool] = {}

        async def limited_check(gw: Dict[str, Any]) -> None:
            async with sem:
                ok = await self._check_gateway(gw)
                results[gw["id"]] = ok

        gws = [g for g in self.registry.list("gateway") if g.get("enabled", True)
               and g.get("owner_rank", 0) % self.world_size == self.rank]
        await asyncio.gather(*(limited_check(g) for g in gws))
        return results

    async def _check_gateway(self, gw: Dict[str, Any]) -> bool:
        client = self.tools.upstream_for(gw["id"])
        try:
            if client is None:
                await self._initialize_gateway(gw)
                return True
            ok = await asyncio.wait_for(client.ping(), timeout=self.settings.health_check_timeout)
        except Exception:
            ok = False
        if ok:
            if not gw.get("reachable", True):
                # reactivation (reference: _mark_gateway_reachable :4529)
                try:
                    await self._initialize_gateway(gw, client)
                except Exception:
                    return False
            self.registry.update("gateway", gw["id"], reachable=True, status="active", consecutive_failures=0)
            return True
        fails = int(gw.get("consecutive_failures", 0)) + 1
        fields: Dict[str, Any] = {"consecutive_failures": fails}
        if fails >= self.settings.unhealthy_threshold:
            # failure handling (reference: _handle_gateway_failure :4362)
            fields.update(reachable=False, status="unreachable")
            for t in self.registry.tools_for_gateway(gw["id"]):
                self.registry.update("tool", t["id"], reachable=False)
        self.registry.update("gateway", gw["id"], **fields)
        return False

    # -- async lifecycle loop (reference: _run_gateway_lifecycle_loop :4154) --
    def ensure_lifecycle_loop(self) -> None:
        if self._lifecycle_task is None or self._lifecycle_task.done():
            self._stop.clear()

            async def loop() -> None:
                tick = max(0.05, self.settings.gateway_lifecycle_tick_s)
                while not self._stop.is_set():
                    try:
                        await self.lifecycle_tick()
                    except Exception:  # pragma: no cover - defensive
                        logger.exception("gateway lifecycle tick error")
                    try:
                        await asyncio.wait_for(self._stop.wait(), timeout=tick)
                        return
                    except asyncio.TimeoutError:
                        pass

            self._lifecycle_task = asyncio.create_task(loop())

    async def lifecycle_tick(self) -> Dict[str, int]:
        """One pass over claimable rows (reference: _claim_due_gateway_
        lifecycle_ids :4111 + _process_pending_gateway :4250). Rows are
        sharded by owner_rank, so claims cannot race across ranks; within a
        rank the pending→initializing flip guards re-entry."""
        now = time.time()
        counts = {"activated": 0, "retried": 0, "failed": 0, "deleted": 0}
        if self.leader_check is not None and not self.leader_check():
            return counts  # follower: lifecycle rows belong to the leader
        for gw in list(self.registry.list("gateway", include_disabled=True)):
            if gw.get("owner_rank", 0) % self.world_size != self.rank:
                continue
            status = gw.get("status")
            if status == "pending" and (gw.get("next_retry_at") or 0) <= now:
                self.registry.update("gateway", gw["id"], status="initializing")
                client = self._injected_clients.get(gw["id"])
                try:
                    await self._initialize_gateway(gw, client)
                    self.registry.update("gateway", gw["id"], retry_count=0,
                                         last_error=None, failure_class=None, next_retry_at=None)
                    counts["activated"] += 1
                except Exception as exc:
                    rc = int(gw.get("retry_count") or 0) + 1
                    cls = self.classify_failure(exc)
                    fields = dict(retry_count=rc, last_error=str(exc)[:500], failure_class=cls)
                    if rc >= self.settings.gateway_max_retries:
                        # terminal until a manual refresh (reference: failed
                        # lifecycle rows need operator action)
                        fields.update(status="failed", reachable=False, next_retry_at=None)
                        counts["failed"] += 1
                    else:
                        fields.update(status="pending", reachable=False,
                                      next_retry_at=now + self._retry_backoff(rc))
                        counts["retried"] += 1
                    self.registry.update("gateway", gw["id"], **fields)
            elif status == "deleting":
                await self._finish_delete(gw["id"])
                counts["deleted"] += 1
        return counts

    # -- upstream OAuth authorization-code flow (reference: oauth_router +
    # oauth_manager.py token exchange for upstreams) ------------------------
    def oauth_provider_for(self, gateway_id: str):
        from ..auth.oauth import AuthorizationCodeProvider, OAuthError, provider_from_auth_value

        prov = self._oauth_providers.get(gateway_id)
        if prov is None:
            import json as _json

            gw = self.registry.get("gateway", gateway_id)
            if gw.get("auth_type") != "oauth":
                raise OAuthError(f"gateway {gateway_id} does not use OAuth auth")
            cfg = self.crypto.open_(gw.get("auth_value"))
            prov = provider_from_auth_value(
                _json.loads(cfg) if isinstance(cfg, str) else (cfg or {}),
                storage=self.token_storage, storage_key=f"gateway:{gateway_id}",
                state_secret=self.settings.jwt_secret_key)
            self._oauth_providers[gateway_id] = prov
        if not isinstance(prov, AuthorizationCodeProvider):
            raise OAuthError("gateway uses client-credentials (no browser flow needed)")
        return prov

    def begin_upstream_authorization(self, gateway_id: str,
                                     redirect_uri: Optional[str] = None) -> Dict[str, str]:
        return self.oauth_provider_for(gateway_id).begin_authorization(redirect_uri)

    async def complete_upstream_authorization(self, code: str, state: str,
                                              redirect_uri: Optional[str] = None) -> str:
        """Callback half: state carries the storage key `gateway:{id}`."""
        from ..auth.oauth import verify_state

        payload = verify_state(state, self.settings.jwt_secret_key)
        key = payload.get("k", "")
        gateway_id = key.split(":", 1)[1] if key.startswith("gateway:") else key
        prov = self.oauth_provider_for(gateway_id)
        await prov.complete_authorization(code, state, redirect_uri=redirect_uri)
        return gateway_id

    async def retry_failed_gateway(self, gateway_id: str) -> Dict[str, Any]:
        """Operator action: put a failed row back into the pending lifecycle."""
        self.registry.update("gateway", gateway_id, status="pending", retry_count=0,
                             next_retry_at=time.time(), last_error=None, failure_class=None)
        self.ensure_lifecycle_loop()
        return self.registry.get("gateway", gateway_id)

    async def start_health_loop(self) -> None:
        if self._health_task is not None:
            return
        self._stop.clear()

        async def loop() -> None:
            while not self._stop.is_set():
                try:
                    await self.check_health_once()
                except Exception as exc:  # pragma: no cover
                    logger.warning("health loop error: %s", exc)
                try:
                    await asyncio.wait_for(self._stop.wait(), timeout=self.settings.health_check_interval)
                except asyncio.TimeoutError:
                    pass

        self._health_task = asyncio.create_task(loop())

    async def stop(self) -> None:
        self._stop.set()
        for attr in ("_health_task", "_lifecycle_task"):
            task = getattr(self, attr)
            if task:
                task.cancel()
                try:
                    await task
                except (asyncio.CancelledError, Exception):
                    pass
                setattr(self, attr, None)

    async def refresh_gateway(self, gateway_id: str) -> Dict[str, Any]:
        """Manual refresh (reference: refresh_gateway_manually :6548)."""
        gw = self.registry.get("gateway", gateway_id)
        await self._initialize_gateway(gw, self.tools.upstream_for(gateway_id))
        return self.registry.get("gateway", gateway_id)

    async def delete_gateway(self, gateway_id: str, defer: bool = False) -> None:
        """Delete a gateway. `defer=True` (reference: deleting lifecycle
        rows :4077) marks the row `deleting` and lets the lifecycle loop
        finish the teardown; callers observing the registry see the state."""
        self.registry.get("gateway", gateway_id)  # raises NotFoundError
        self.registry.update("gateway", gateway_id, status="deleting")
        if defer:
            self.ensure_lifecycle_loop()
            return
        await self._finish_delete(gateway_id)

    async def _finish_delete(self, gateway_id: str) -> None:
        for t in self.registry.tools_for_gateway(gateway_id):
            self.registry.delete("tool", t["id"])
        client = self.tools.upstream_for(gateway_id)
        if client:
            await client.aclose()
            self.tools._upstreams.pop(gateway_id, None)
        self._injected_clients.pop(gateway_id, None)
        try:
            self.registry.delete("gateway", gateway_id)
        except NotFoundError:  # pragma: no cover - already gone
            pass
