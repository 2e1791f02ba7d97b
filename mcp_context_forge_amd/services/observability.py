"""Self-hosted observability: traces/spans + structured logs + audit trail.

Reference analogs: mcpgateway/observability.py (create_span :1264,
trace_operation :1198; OTel exporters are optional there too),
services/observability_service.py (:231 ingestion into the Observability*
tables), services/log_storage_service.py + structured_logger.py,
services/audit_trail_service.py. No OTel SDK in this image, so spans are
recorded natively with the same shape and persisted to the same-style
tables; an OTLP exporter can bolt on later without changing call sites.
"""

from __future__ import annotations

import contextvars
import logging
import os
import threading
import time
import uuid
from contextlib import contextmanager
from typing import Any, Dict, List, Optional

from ..db.engine import Database
from ..db.models import DbAuditLog, DbObservabilitySpan, DbStructuredLog

_current_span: contextvars.ContextVar = contextvars.ContextVar("forge_span", default=None)


class Span:
    __slots__ = ("trace_id", "span_id", "parent_span_id", "name", "start_ns", "end_ns", "attributes", "status")

    def __init__(self, name: str, trace_id: Optional[str] = None, parent: Optional["Span"] = None):
        self.trace_id = trace_id or (parent.trace_id if parent else uuid.uuid4().hex)
        self.span_id = uuid.uuid4().hex[:16]
        self.parent_span_id = parent.span_id if parent else None
        self.name = name
        self.start_ns = time.time_ns()
        self.end_ns = 0
        self.attributes: Dict[str, Any] = {}
        self.status = "OK"

    def set_attribute(self, key: str, value: Any) -> None:
        self.attributes[key] = value

    def end(self, status: str = "OK") -> None:
        self.end_ns = time.time_ns()
        self.status = status

    @property
    def duration_ms(self) -> float:
        return (self.end_ns - self.start_ns) / 1e6 if self.end_ns else 0.0


class ObservabilityService:
    """Span buffer + periodic persistence (reference: ObservabilityService :231)."""

    def __init__(self, db: Optional[Database] = None, max_buffer: int = 2048, enabled: bool = True):
        self.db = db
        self.enabled = enabled
        self.max_buffer = max_buffer
        self._spans: List[Span] = []
        self._lock = threading.Lock()
        # optional OTLP/HTTP shipper (services/otel_export.py); set by the
        # engine when Settings.otel_endpoint is configured
        self.exporter = None

    @contextmanager
    def span(self, name: str, **attrs: Any):
        """create_span analog (observability.py:1264) — contextvar-nested."""
        if not self.enabled:
            yield None
            return
        parent = _current_span.get()
        sp = Span(name, parent=parent)
        sp.attributes.update(attrs)
        token = _current_span.set(sp)
        try:
            yield sp
            sp.end("OK")
        except Exception:
            sp.end("ERROR")
            raise
        finally:
            _current_span.reset(token)
            self._record(sp)

    def _record(self, sp: Span) -> None:
        with self._lock:
            self._spans.append(sp)
            need = len(self._spans) >= self.max_buffer
        if need:
            self.flush()

    def current_trace_id(self) -> Optional[str]:
        sp = _current_span.get()
        return sp.trace_id if sp else None

    def flush(self) -> int:
        with self._lock:
            spans, self._spans = self._spans, []
        if spans and self.exporter is not None:
            self.exporter.export(spans)  # fail-open, background thread
        if not spans or self.db is None:
            return len(spans)
        with self.db.session() as s:
            for sp in spans:
                s.add(DbObservabilitySpan(
                    trace_id=sp.trace_id, span_id=sp.span_id, parent_span_id=sp.parent_span_id,
                    name=sp.name, start_ns=sp.start_ns, end_ns=sp.end_ns,
                    attributes=sp.attributes, status=sp.status))
        return len(spans)

    def query_traces(self, limit: int = 100) -> List[Dict[str, Any]]:
        if self.db is None:
            return []
        from sqlalchemy import select

        with self.db.session() as s:
            rows = s.execute(select(DbObservabilitySpan)
                             .order_by(DbObservabilitySpan.start_ns.desc()).limit(limit)).scalars().all()
            return [{"trace_id": r.trace_id, "span_id": r.span_id, "parent": r.parent_span_id,
                     "name": r.name, "duration_ms": (r.end_ns - r.start_ns) / 1e6,
                     "status": r.status, "attributes": r.attributes} for r in rows]


class DbLogHandler(logging.Handler):
    """Structured-log persistence (reference: log_storage_service + db.py:6141)."""

    def __init__(self, db: Database, level=logging.WARNING, max_buffer: int = 500):
        super().__init__(level)
        self.db = db
        self.buffer: List[dict] = []
        self.max_buffer = max_buffer
        self._lock2 = threading.Lock()

    def emit(self, record: logging.LogRecord) -> None:
        try:
            row = {"level": record.levelname, "logger": record.name,
                   "message": self.format(record), "context": {"module": record.module, "line": record.lineno}}
            with self._lock2:
                self.buffer.append(row)
                need = len(self.buffer) >= self.max_buffer
            if need:
                self.flush_to_db()
        except Exception:  # pragma: no cover
            pass

    def flush_to_db(self) -> int:
        with self._lock2:
            rows, self.buffer = self.buffer, []
        if rows:
            with self.db.session() as s:
                s.bulk_insert_mappings(DbStructuredLog, rows)
        return len(rows)


class AuditTrail:
    """Audit log (reference: audit_trail_service + db.py:6624)."""

    def __init__(self, db: Database):
        self.db = db

    def record(self, actor: Optional[str], action: str, entity_type: Optional[str] = None,
               entity_id: Optional[str] = None, **detail: Any) -> None:
        with self.db.session() as s:
            s.add(DbAuditLog(actor=actor, action=action, entity_type=entity_type,
                             entity_id=entity_id, detail=detail))

    def query(self, limit: int = 100) -> List[dict]:
        from sqlalchemy import select

        with self.db.session() as s:
            rows = s.execute(select(DbAuditLog).order_by(DbAuditLog.id.desc()).limit(limit)).scalars().all()
            return [{"actor": r.actor, "action": r.action, "entity_type": r.entity_type,
                     "entity_id": r.entity_id, "detail": r.detail, "timestamp": str(r.timestamp)} for r in rows]


class SiemExporter:
    """Security-event export (reference: services/siem_export_service.py +
    security_logger.py — JSON events shipped to OpenSearch/SIEM). Here:
    audit + structured security events rendered as JSONL for pull-based
    export (/admin/siem/export) or pushed to a webhook sink."""

    def __init__(self, db: Database):
        self.db = db
        self.exported = 0

    def export_jsonl(self, since: Optional[str] = None, limit: int = 1000) -> str:
        import json as _json

        from sqlalchemy import select

        from ..db.models import DbAuditLog, DbStructuredLog

        lines: List[str] = []
        with self.db.session() as s:
            q = select(DbAuditLog).order_by(DbAuditLog.id.desc()).limit(limit)
            for r in s.execute(q).scalars():
                lines.append(_json.dumps({
                    "type": "audit", "timestamp": r.timestamp.isoformat(),
                    "actor": r.actor, "action": r.action,
                    "entity_type": r.entity_type, "entity_id": r.entity_id,
                    "detail": r.detail}))
            q = select(DbStructuredLog).where(DbStructuredLog.level.in_(("WARNING", "ERROR", "CRITICAL"))) \
                .order_by(DbStructuredLog.id.desc()).limit(limit)
            for r in s.execute(q).scalars():
                lines.append(_json.dumps({
                    "type": "log", "timestamp": r.timestamp.isoformat(), "level": r.level,
                    "logger": r.logger, "message": r.message, "context": r.context}))
        self.exported += len(lines)
        return "\n".join(lines)

    async def push_webhook(self, url: str, since: Optional[str] = None, limit: int = 1000) -> int:
        import httpx

        payload = self.export_jsonl(since, limit)
        n = payload.count("\n") + 1 if payload else 0
        async with httpx.AsyncClient(timeout=15.0) as c:
            r = await c.post(url, content=payload.encode(),
                             headers={"content-type": "application/x-ndjson"})
            r.raise_for_status()
        return n


class ComplianceService:
    """Compliance posture report (reference: services/compliance.py):
    aggregate security-relevant configuration + usage into one snapshot."""

    def __init__(self, engine):
        self.engine = engine

    def report(self) -> Dict[str, Any]:
        e = self.engine
        s = e.settings
        plugins = [{"name": p.name, "mode": p.mode.value, "priority": p.priority}
                   for p in e.plugins.plugins]
        users = e.registry.db  # db handle; user counts via ORM
        from sqlalchemy import func, select

        from ..db.models import DbApiToken, DbAuditLog, DbUser

        with e.db.session() as sess:
            n_users = sess.execute(select(func.count()).select_from(DbUser)).scalar() or 0
            n_tokens = sess.execute(select(func.count()).select_from(DbApiToken)).scalar() or 0
            n_audit = sess.execute(select(func.count()).select_from(DbAuditLog)).scalar() or 0
        return {
            "auth": {"required": s.auth_required, "sso_enabled": bool(getattr(s, "sso_providers", None)),
                     "users": n_users, "api_tokens": n_tokens},
            "plugins": {"enabled": e.plugins.enabled, "count": len(plugins), "chain": plugins,
                        "bindings": sum(len(v) for v in e.plugins.bindings.values())},
            "transport": {"federation_enabled": s.federation_enabled,
                          "rate_limit_rpm": getattr(s, "rate_limit_rpm", None),
                          "max_body_bytes": getattr(s, "max_request_body_bytes", None)},
            "audit": {"events": n_audit},
            "gpu": {"pipeline_attached": e.gpu_pipeline is not None},
            "uptime_s": round(__import__("time").time() - e.started_at, 1),
        }
