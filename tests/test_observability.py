"""Self-hosted tracing / audit / structured logs / metric rollups
(reference analogs: observability_service, audit_trail, metrics_rollup)."""

import json
import logging

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine


def test_spans_nested_and_persisted(run, bare_engine):
    obs = bare_engine.observability
    with obs.span("outer", kind="test") as outer:
        with obs.span("inner") as inner:
            assert inner.trace_id == outer.trace_id
            assert inner.parent_span_id == outer.span_id
    obs.flush()
    traces = obs.query_traces()
    names = {t["name"] for t in traces}
    assert {"outer", "inner"} <= names
    inner_row = next(t for t in traces if t["name"] == "inner")
    assert inner_row["parent"] is not None and inner_row["duration_ms"] >= 0


def test_tool_call_creates_span(run, bare_engine):
    async def echo(args):
        return args

    bare_engine.tool_service.register_local_tool("echo", echo)

    async def go():
        await bare_engine.handle_rpc_bytes(
            b'{"jsonrpc":"2.0","id":1,"method":"tools/call","params":{"name":"echo","arguments":{}}}')

    run(go())
    bare_engine.observability.flush()
    traces = bare_engine.observability.query_traces()
    assert any(t["name"] == "tools/call" and t["attributes"].get("tool") == "echo" for t in traces)


def test_span_error_status(bare_engine):
    try:
        with bare_engine.observability.span("boom"):
            raise ValueError("x")
    except ValueError:
        pass
    bare_engine.observability.flush()
    assert any(t["status"] == "ERROR" for t in bare_engine.observability.query_traces())


def test_audit_trail(bare_engine):
    bare_engine.audit.record("admin", "create", "tool", "t1", name="x")
    rows = bare_engine.audit.query()
    assert rows[0]["action"] == "create" and rows[0]["detail"] == {"name": "x"}


def test_db_log_handler(bare_engine):
    from mcp_context_forge_amd.services.observability import DbLogHandler

    h = DbLogHandler(bare_engine.db, level=logging.WARNING)
    lg = logging.getLogger("forge.test")
    lg.addHandler(h)
    lg.warning("something odd %d", 7)
    lg.removeHandler(h)
    assert h.flush_to_db() == 1
    from sqlalchemy import select

    from mcp_context_forge_amd.db.models import DbStructuredLog

    with bare_engine.db.session() as s:
        rows = s.execute(select(DbStructuredLog)).scalars().all()
        assert rows and "something odd 7" in rows[0].message


def test_metric_rollup(run, bare_engine):
    from mcp_context_forge_amd.services.metrics import rollup_hourly

    for i in range(5):
        bare_engine.metrics.record_tool_metric("tool-a", 10.0 + i, i % 2 == 0)
    bare_engine.metrics.record_aggregate("tool-b", count=100, errors=3, response_time_ms=42.0)
    bare_engine.metrics.flush()
    created = rollup_hourly(bare_engine.db)
    assert created >= 2
    from sqlalchemy import select

    from mcp_context_forge_amd.db.models import DbMetricRollup, DbToolMetric

    with bare_engine.db.session() as s:
        assert s.execute(select(DbToolMetric)).scalars().all() == []  # raw pruned
        rolls = {r.entity_id: r for r in s.execute(select(DbMetricRollup)).scalars().all()}
        assert rolls["tool-a"].count == 5


def test_siem_export_and_compliance(run):
    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine

    async def go():
        e = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                   auth_required=False))
        e.audit.record("admin", "create", "tool", "t1")
        e.audit.record("admin", "delete", "tool", "t1")
        out = e.siem.export_jsonl()
        lines = [json.loads(l) for l in out.splitlines()]
        assert len(lines) >= 2
        assert {l["action"] for l in lines if l["type"] == "audit"} == {"create", "delete"}
        rep = e.compliance.report()
        assert rep["plugins"]["count"] > 0
        assert rep["audit"]["events"] >= 2
        assert rep["gpu"]["pipeline_attached"] is False
        await e.shutdown()

    run(go())


def test_admin_log_search_endpoint(run):
    """Log search (reference: routers/log_search.py)."""
    import base64

    import httpx

    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine
    from mcp_context_forge_amd.services.observability import DbLogHandler
    from mcp_context_forge_amd.transports.http_app import build_app

    async def go():
        e = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                   auth_required=True))
        h = DbLogHandler(e.db)
        import logging as _l

        rec = _l.LogRecord("forge.test", _l.WARNING, __file__, 1, "upstream flaked badly", (), None)
        h.emit(rec)
        rec2 = _l.LogRecord("other.mod", _l.ERROR, __file__, 2, "boom", (), None)
        h.emit(rec2)
        h.flush_to_db()
        app = build_app(e)
        admin = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}
        async with app.router.lifespan_context(app):
            async with httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                         base_url="http://gw") as c:
                r = await c.get("/admin/logs", headers=admin)
                assert r.status_code == 200 and len(r.json()) == 2
                r = await c.get("/admin/logs", headers=admin, params={"level": "error"})
                assert [x["message"] for x in r.json()] == ["boom"]
                r = await c.get("/admin/logs", headers=admin, params={"q": "flaked"})
                assert len(r.json()) == 1
                r = await c.get("/admin/logs", headers=admin, params={"logger_name": "forge"})
                assert len(r.json()) == 1
        await e.shutdown()

    run(go())
