"""Governance services: catalog, password policy, token blocklist,
content-security limits.

Reference analogs: services/catalog_service.py (curated mcp-catalog.yml),
services/password_policy.py, services/token_blocklist.py,
services/content_security.py.
"""

from __future__ import annotations

import re
import time
from pathlib import Path
from typing import Any, Dict, List, Optional, Set

import yaml


class CatalogService:
    """Curated MCP-server catalog (reference: catalog_service.py + mcp-catalog.yml)."""

    def __init__(self, catalog_file: Optional[str] = None):
        self.servers: List[Dict[str, Any]] = []
        if catalog_file and Path(catalog_file).exists():
            self.load(catalog_file)
        else:
            self.servers = self.default_catalog()

    def load(self, path: str) -> None:
        raw = yaml.safe_load(Path(path).read_text()) or {}
        self.servers = raw.get("catalog_servers", raw.get("servers", []))

    @staticmethod
    def default_catalog() -> List[Dict[str, Any]]:
        return [
            {"id": "fast-time", "name": "Fast Time Server", "url": "http://localhost:8888/mcp",
             "category": "utilities", "auth_type": "none", "tags": ["time"],
             "description": "Benchmark time-conversion upstream"},
            {"id": "github", "name": "GitHub MCP", "url": "https://api.githubcopilot.com/mcp/",
             "category": "development", "auth_type": "oauth", "tags": ["git", "code"],
             "description": "GitHub tools over MCP"},
        ]

    def list(self, category: Optional[str] = None) -> List[Dict[str, Any]]:
        out = self.servers
        if category:
            out = [s for s in out if s.get("category") == category]
        return out

    def get(self, catalog_id: str) -> Optional[Dict[str, Any]]:
        for s in self.servers:
            if s.get("id") == catalog_id:
                return s
        return None

    async def register_from_catalog(self, catalog_id: str, gateway_service) -> Dict[str, Any]:
        entry = self.get(catalog_id)
        if entry is None:
            raise KeyError(f"catalog entry {catalog_id!r} not found")
        return await gateway_service.register_gateway(
            name=entry.get("name", catalog_id), url=entry["url"],
            description=entry.get("description", ""), tags=entry.get("tags"))


class PasswordPolicy:
    """Password strength rules (reference: services/password_policy.py)."""

    def __init__(self, min_length: int = 10, require_upper: bool = True, require_lower: bool = True,
                 require_digit: bool = True, require_special: bool = False,
                 forbidden: Optional[List[str]] = None):
        self.min_length = min_length
        self.require_upper = require_upper
        self.require_lower = require_lower
        self.require_digit = require_digit
        self.require_special = require_special
        self.forbidden = set(forbidden or ["password", "changeme", "12345678", "qwerty"])

    def validate(self, password: str) -> List[str]:
        errs = []
        if len(password) < self.min_length:
            errs.append(f"must be at least {self.min_length} characters")
        if self.require_upper and not re.search(r"[A-Z]", password):
            errs.append("must contain an uppercase letter")
        if self.require_lower and not re.search(r"[a-z]", password):
            errs.append("must contain a lowercase letter")
        if self.require_digit and not re.search(r"\d", password):
            errs.append("must contain a digit")
        if self.require_special and not re.search(r"[^\w\s]", password):
            errs.append("must contain a special character")
        if password.lower() in self.forbidden:
            errs.append("password is too common")
        return errs


class TokenBlocklist:
    """Revoked-JTI blocklist (reference: services/token_blocklist.py).

    Multi-rank note: revocations propagate with the registry-invalidation
    broadcast (parallel.collectives.broadcast_object) instead of Redis.
    """

    def __init__(self):
        self._blocked: Dict[str, float] = {}  # jti -> expiry epoch (0 = forever)
        self.version = 0  # bumped on every revocation; auth caches key on it

    def block(self, jti: str, expires_at: float = 0.0) -> None:
        self._blocked[jti] = expires_at
        self.version += 1

    def is_blocked(self, jti: Optional[str]) -> bool:
        if not jti:
            return False
        exp = self._blocked.get(jti)
        if exp is None:
            return False
        if exp and exp < time.time():
            del self._blocked[jti]
            return False
        return True

    def purge_expired(self) -> int:
        now = time.time()
        stale = [j for j, e in self._blocked.items() if e and e < now]
        for j in stale:
            del self._blocked[j]
        return len(stale)

    def snapshot(self) -> Dict[str, float]:
        return dict(self._blocked)

    def merge(self, other: Dict[str, float]) -> None:
        self._blocked.update(other)


class ContentSecurity:
    """Payload size/type limits (reference: services/content_security.py)."""

    def __init__(self, max_result_bytes: int = 8 << 20, max_content_items: int = 256,
                 allowed_content_types: Optional[Set[str]] = None):
        self.max_result_bytes = max_result_bytes
        self.max_content_items = max_content_items
        self.allowed_content_types = allowed_content_types or {"text", "image", "audio", "resource"}

    def check_result(self, result: Any) -> List[str]:
        errs = []
        if isinstance(result, dict):
            content = result.get("content", [])
            if len(content) > self.max_content_items:
                errs.append(f"too many content items ({len(content)})")
            total = 0
            for c in content:
                if isinstance(c, dict):
                    if c.get("type") not in self.allowed_content_types:
                        errs.append(f"content type {c.get('type')!r} not allowed")
                    total += len(c.get("text", "") or "") + len(c.get("data", "") or "")
            if total > self.max_result_bytes:
                errs.append(f"result too large ({total} bytes)")
        return errs


class TagService:
    """Cross-entity tag aggregation (reference: services/tag_service.py —
    GET /tags with per-kind counts and optional entity listings)."""

    KINDS = ("tool", "gateway", "server", "resource", "prompt", "a2a_agent")

    def __init__(self, registry):
        self.registry = registry

    def list_tags(self, kinds: Optional[List[str]] = None,
                  include_entities: bool = False) -> List[Dict[str, Any]]:
        kinds = [k for k in (kinds or self.KINDS) if k in self.KINDS]
        agg: Dict[str, Dict[str, Any]] = {}
        for kind in kinds:
            for ent in self.registry.list(kind):
                for tag in ent.get("tags") or []:
                    rec = agg.setdefault(tag, {"name": tag, "count": 0,
                                               "by_kind": {}, "entities": []})
                    rec["count"] += 1
                    rec["by_kind"][kind] = rec["by_kind"].get(kind, 0) + 1
                    if include_entities:
                        key = ent.get("name") or ent.get("uri") or ent.get("id")
                        rec["entities"].append({"kind": kind, "id": ent.get("id"), "name": key})
        out = sorted(agg.values(), key=lambda r: (-r["count"], r["name"]))
        if not include_entities:
            for r in out:
                r.pop("entities", None)
        return out

    def entities_for_tag(self, tag: str, kinds: Optional[List[str]] = None) -> List[Dict[str, Any]]:
        for rec in self.list_tags(kinds, include_entities=True):
            if rec["name"] == tag:
                return rec["entities"]
        return []


class ServerClassificationService:
    """Heuristic server/tool classification (reference:
    services/server_classification.py): bucket registered entities into
    capability categories from names, descriptions and tags — feeds the
    admin catalog view and policy defaults."""

    CATEGORIES = {
        "time": ("time", "clock", "timezone", "date"),
        "data": ("sql", "database", "query", "table", "csv", "dataset"),
        "devops": ("deploy", "kubernetes", "docker", "ci", "build", "git"),
        "communication": ("mail", "slack", "message", "notify", "chat"),
        "search": ("search", "lookup", "find", "index"),
        "ai": ("llm", "embed", "summar", "classif", "generate"),
        "files": ("file", "read", "write", "fs", "storage"),
    }

    def __init__(self, registry):
        self.registry = registry

    def classify_text(self, text: str) -> str:
        t = text.lower()
        best, hits = "other", 0
        for cat, kws in self.CATEGORIES.items():
            n = sum(1 for k in kws if k in t)
            if n > hits:
                best, hits = cat, n
        return best

    def classify_all(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {"servers": [], "gateways": [], "by_category": {}}
        for kind, dest in (("server", "servers"), ("gateway", "gateways")):
            for ent in self.registry.list(kind):
                tools = [t for t in self.registry.list("tool")
                         if t.get("gateway_id") == ent.get("id")] if kind == "gateway" else []
                text = " ".join([ent.get("name", ""), ent.get("description") or "",
                                 " ".join(ent.get("tags") or [])]
                                + [t.get("name", "") + " " + (t.get("description") or "")
                                   for t in tools])
                cat = self.classify_text(text)
                out[dest].append({"id": ent.get("id"), "name": ent.get("name"),
                                  "category": cat, "tools": len(tools)})
                out["by_category"].setdefault(cat, 0)
                out["by_category"][cat] += 1
        return out
