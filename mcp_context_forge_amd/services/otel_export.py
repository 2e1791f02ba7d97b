"""OTLP/HTTP trace exporter — the OTel export path, dependency-free.

Reference analog: mcpgateway/observability.py:970 init_telemetry with
OTLP/Jaeger/Zipkin/console exporters via the opentelemetry SDK. This image
has no opentelemetry wheels, so the exporter speaks the OTLP/HTTP **JSON**
encoding (opentelemetry-proto ExportTraceServiceRequest, stable since
OTLP 1.0) directly: any OTLP collector endpoint (otel-collector, Jaeger
>=1.35 `/v1/traces`, Grafana Tempo, Langfuse OTLP ingest) accepts it.

Wired into ObservabilityService: when Settings.otel_endpoint is set, every
flush also ships the span batch to `{endpoint}/v1/traces` on a background
thread with a bounded retry queue — export failures never touch the
request path (fail-open telemetry, same as the reference's
BatchSpanProcessor)."""

from __future__ import annotations

import json
import logging
import queue
import threading
from typing import Any, Dict, List, Optional

logger = logging.getLogger(__name__)


def _attr_value(v: Any) -> Dict[str, Any]:
    """AnyValue encoding (opentelemetry-proto common/v1)."""
    if isinstance(v, bool):
        return {"boolValue": v}
    if isinstance(v, int):
        return {"intValue": str(v)}
    if isinstance(v, float):
        return {"doubleValue": v}
    if isinstance(v, (list, tuple)):
        return {"arrayValue": {"values": [_attr_value(x) for x in v]}}
    return {"stringValue": str(v)}


def _attrs(d: Dict[str, Any]) -> List[Dict[str, Any]]:
    return [{"key": str(k), "value": _attr_value(v)} for k, v in (d or {}).items()]


def spans_to_otlp(spans, service_name: str = "mcp-context-forge-amd",
                  resource_attrs: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
    """Our Span records → ExportTraceServiceRequest (JSON encoding).

    trace_id must be 32 hex chars, span_id 16 — our ids already are
    (uuid4().hex / [:16]). status maps OK→1 (STATUS_CODE_OK), ERROR→2."""
    otlp_spans = []
    for sp in spans:
        otlp_spans.append({
            "traceId": sp.trace_id,
            "spanId": sp.span_id,
            **({"parentSpanId": sp.parent_span_id} if sp.parent_span_id else {}),
            "name": sp.name,
            "kind": 2,  # SPAN_KIND_SERVER
            "startTimeUnixNano": str(sp.start_ns),
            "endTimeUnixNano": str(sp.end_ns or sp.start_ns),
            "attributes": _attrs(sp.attributes),
            "status": {"code": 1 if sp.status == "OK" else 2,
                       **({} if sp.status == "OK" else {"message": sp.status})},
        })
    resource = {"attributes": _attrs({"service.name": service_name,
                                      **(resource_attrs or {})})}
    return {"resourceSpans": [{
        "resource": resource,
        "scopeSpans": [{"scope": {"name": "mcp_context_forge_amd.observability"},
                        "spans": otlp_spans}],
    }]}


class OtlpHttpExporter:
    """Background OTLP/HTTP shipper with a bounded queue.

    export(spans) enqueues; a daemon thread POSTs batches to
    {endpoint}/v1/traces. Overflow drops oldest (telemetry must never
    apply backpressure to the data plane). Synchronous `export_now` is
    for tests/shutdown flushes."""

    def __init__(self, endpoint: str, headers: Optional[Dict[str, str]] = None,
                 service_name: str = "mcp-context-forge-amd", timeout: float = 5.0,
                 max_queue: int = 64):
        self.endpoint = endpoint.rstrip("/")
        self.headers = {"content-type": "application/json", **(headers or {})}
        self.service_name = service_name
        self.timeout = timeout
        self._q: "queue.Queue[list]" = queue.Queue(maxsize=max_queue)
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.exported = 0
        self.dropped = 0
        self.errors = 0

    def start(self) -> None:
        if self._thread is None:
            self._thread = threading.Thread(target=self._run, name="otlp-export", daemon=True)
            self._thread.start()

    def export(self, spans: list) -> None:
        if not spans:
            return
        self.start()
        try:
            self._q.put_nowait(list(spans))
        except queue.Full:
            try:
                self._q.get_nowait()  # drop oldest batch
                self.dropped += 1
                self._q.put_nowait(list(spans))
            except (queue.Empty, queue.Full):
                self.dropped += 1

    def export_now(self, spans: list) -> bool:
        return self._post(list(spans))

    def _post(self, spans: list) -> bool:
        import httpx

        body = json.dumps(spans_to_otlp(spans, self.service_name)).encode()
        try:
            with httpx.Client(timeout=self.timeout) as c:
                r = c.post(f"{self.endpoint}/v1/traces", content=body, headers=self.headers)
            if r.status_code >= 400:
                self.errors += 1
                return False
            self.exported += len(spans)
            return True
        except Exception as exc:
            self.errors += 1
            logger.debug("OTLP export failed: %s", exc)
            return False

    def _run(self) -> None:
        while not self._stop.is_set():
            try:
                spans = self._q.get(timeout=0.25)
            except queue.Empty:
                continue
            self._post(spans)

    def stop(self, drain: bool = True) -> None:
        if drain:
            while True:
                try:
                    self._post(self._q.get_nowait())
                except queue.Empty:
                    break
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None
