"""Unified entity registry with in-memory hot lookup.

Reference analogs: mcpgateway/services/{tool,resource,prompt,server}_service
CRUD + mcpgateway/cache/{tool_lookup_cache,registry_cache}. The reference
caches lookups in Redis; here the hot name→tool mapping is an in-process
dict plus, when the GPU pipeline is attached, a native C++ toolmap and
flat per-tool flag tables rebuilt on every registry generation bump
(gpu/pipeline.py:_rebuild_tool_meta) — name resolution for a whole
micro-batch is one C call, never a DB or network hop. The DB is
durability only. (What lives in HBM proper: the DFA scan banks, the
classifier weights and the semantic-cache key matrix — the decision
tables stay host-side C++ because the per-batch decision pass is
latency-, not bandwidth-, bound.)

All methods are synchronous and thread-safe; the asyncio layer calls them
directly (they only touch memory + short sqlite transactions).
"""

from __future__ import annotations

import threading
from typing import Any, Dict, List, Optional, Type

from sqlalchemy import select

from ..db.engine import Database
from ..db.models import (
    DbA2AAgent,
    DbGateway,
    DbPluginBinding,
    DbPrompt,
    DbResource,
    DbServer,
    DbTool,
)
from ..utils import qualified_tool_name


def _row_to_dict(row: Any) -> Dict[str, Any]:
    return {c.key: getattr(row, c.key) for c in row.__table__.columns}


class RegistryError(Exception):
    pass


class NotFoundError(RegistryError):
    pass


class ConflictError(RegistryError):
    pass


_MODEL: Dict[str, Type] = {
    "tool": DbTool,
    "gateway": DbGateway,
    "resource": DbResource,
    "prompt": DbPrompt,
    "server": DbServer,
    "a2a_agent": DbA2AAgent,
    "plugin_binding": DbPluginBinding,
}

_UNIQUE_FIELD = {
    "tool": "name",
    "gateway": "name",
    "resource": "uri",
    "prompt": "name",
    "server": "name",
    "a2a_agent": "name",
    "plugin_binding": "name",
}


class Registry:
    """CRUD + hot lookup for every registry entity kind."""

    def __init__(self, db: Database):
        self.db = db
        self._lock = threading.RLock()
        # hot caches: kind -> unique_key -> entity dict (reference: tool_lookup_cache L1)
        self._cache: Dict[str, Dict[str, Dict[str, Any]]] = {k: {} for k in _MODEL}
        self._by_id: Dict[str, Dict[str, Dict[str, Any]]] = {k: {} for k in _MODEL}
        self._generation = 0  # bumped on any mutation; GPU mirror re-syncs on change
        self.load_all()

    # -- cache maintenance ---------------------------------------------------
    def load_all(self) -> None:
        with self._lock, self.db.session() as s:
            for kind, model in _MODEL.items():
                rows = s.execute(select(model)).scalars().all()
                key = _UNIQUE_FIELD[kind]
                self._cache[kind] = {getattr(r, key): _row_to_dict(r) for r in rows}
                self._by_id[kind] = {r.id: self._cache[kind][getattr(r, key)] for r in rows}
            self._generation += 1

    @property
    def generation(self) -> int:
        return self._generation

    def _put_cache(self, kind: str, ent: Dict[str, Any]) -> None:
        self._cache[kind][ent[_UNIQUE_FIELD[kind]]] = ent
        self._by_id[kind][ent["id"]] = ent
        self._generation += 1

    def _drop_cache(self, kind: str, ent: Dict[str, Any]) -> None:
        self._cache[kind].pop(ent[_UNIQUE_FIELD[kind]], None)
        self._by_id[kind].pop(ent["id"], None)
        self._generation += 1

    # -- generic CRUD ----------------------------------------------------------
    def create(self, kind: str, **fields: Any) -> Dict[str, Any]:
        model = _MODEL[kind]
        ukey = _UNIQUE_FIELD[kind]
        with self._lock:
            if fields.get(ukey) in self._cache[kind]:
                raise ConflictError(f"{kind} {fields.get(ukey)!r} already exists")
            with self.db.session() as s:
                row = model(**fields)
                s.add(row)
                s.flush()
                ent = _row_to_dict(row)
            self._put_cache(kind, ent)
            return ent

    def get(self, kind: str, entity_id: str) -> Dict[str, Any]:
        ent = self._by_id[kind].get(entity_id)
        if ent is None:
            raise NotFoundError(f"{kind} {entity_id!r} not found")
        return ent

    def find(self, kind: str, key: str) -> Optional[Dict[str, Any]]:
        """Hot-path lookup by unique key (name/uri). O(1) dict hit, no DB."""
        return self._cache[kind].get(key)

    def list(self, kind: str, include_disabled: bool = True, **filters: Any) -> List[Dict[str, Any]]:
        out = list(self._cache[kind].values())
        if not include_disabled:
            out = [e for e in out if e.get("enabled", True)]
        for k, v in filters.items():
            out = [e for e in out if e.get(k) == v]
        return sorted(out, key=lambda e: e.get("created_at") or 0 if e.get("created_at") else 0)

    def update(self, kind: str, entity_id: str, **fields: Any) -> Dict[str, Any]:
        model = _MODEL[kind]
        with self._lock:
            old = self.get(kind, entity_id)
            with self.db.session() as s:
                row = s.get(model, entity_id)
                if row is None:
                    raise NotFoundError(f"{kind} {entity_id!r} not found")
                for k, v in fields.items():
                    setattr(row, k, v)
                s.flush()
                ent = _row_to_dict(row)
            self._drop_cache(kind, old)
            self._put_cache(kind, ent)
            return ent

    def delete(self, kind: str, entity_id: str) -> None:
        model = _MODEL[kind]
        with self._lock:
            ent = self.get(kind, entity_id)
            with self.db.session() as s:
                row = s.get(model, entity_id)
                if row is not None:
                    s.delete(row)
            self._drop_cache(kind, ent)

    def set_enabled(self, kind: str, entity_id: str, enabled: bool) -> Dict[str, Any]:
        return self.update(kind, entity_id, enabled=enabled)

    # -- tool-specific hot path ------------------------------------------------
    def lookup_tool(self, name: str) -> Optional[Dict[str, Any]]:
        """Hot-path tool resolve (reference: tool_service._resolve_tool_for_invocation :5143)."""
        t = self._cache["tool"].get(name)
        if t is not None and t.get("enabled", True):
            return t
        return None

    def tools_for_gateway(self, gateway_id: str) -> List[Dict[str, Any]]:
        return [t for t in self._cache["tool"].values() if t.get("gateway_id") == gateway_id]

    def sync_gateway_tools(self, gateway: Dict[str, Any], tool_defs: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        """Upsert the tools reported by a federated gateway
        (reference: gateway_service._update_or_create_tools :5648)."""
        seen = set()
        out = []
        for td in tool_defs:
            qname = qualified_tool_name(gateway["name"], td["name"])
            seen.add(qname)
            existing = self._cache["tool"].get(qname)
            fields = dict(
                original_name=td["name"],
                name=qname,
                description=td.get("description", ""),
                input_schema=td.get("inputSchema", {"type": "object"}),
                output_schema=td.get("outputSchema"),
                annotations=td.get("annotations"),
                integration_type="MCP",
                gateway_id=gateway["id"],
                url=gateway["url"],
            )
            if existing:
                out.append(self.update("tool", existing["id"], **fields))
            else:
                out.append(self.create("tool", **fields))
        # prune tools the upstream no longer reports
        for t in self.tools_for_gateway(gateway["id"]):
            if t["name"] not in seen:
                self.delete("tool", t["id"])
        return out

    # -- export/import (reference: services/export_service.py:268) -------------
    def export_configuration(self) -> Dict[str, Any]:
        return {
            "version": "1.0",
            "entities": {kind: self.list(kind) for kind in _MODEL},
        }

    def import_configuration(self, payload: Dict[str, Any], conflict_strategy: str = "update") -> Dict[str, int]:
        counts = {"created": 0, "updated": 0, "skipped": 0}
        for kind, ents in payload.get("entities", {}).items():
            if kind not in _MODEL:
                continue
            ukey = _UNIQUE_FIELD[kind]
            for ent in ents:
                ent = {k: v for k, v in ent.items() if k not in ("created_at", "updated_at")}
                existing = self._cache[kind].get(ent.get(ukey))
                if existing is None:
                    self.create(kind, **ent)
                    counts["created"] += 1
                elif conflict_strategy == "update":
                    eid = ent.pop("id", existing["id"])
                    self.update(kind, existing["id"], **{k: v for k, v in ent.items() if k != "id"})
                    counts["updated"] += 1
                else:
                    counts["skipped"] += 1
        return counts
