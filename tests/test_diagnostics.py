"""Diagnostics tier (services/diagnostics.py — reference analogs:
support_bundle.py, wellknown/perf endpoints, mcpgateway/toolops/)."""

import asyncio

import pytest

from mcp_context_forge_amd.config import Settings
from mcp_context_forge_amd.engine import GatewayEngine
from mcp_context_forge_amd.services.diagnostics import (PerformanceService, SupportBundle,
                                                        ToolOps)
from mcp_context_forge_amd.services.upstream import make_fake_time_upstream


async def _engine():
    e = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                               auth_required=False,
                               jwt_secret_key="sekrit-value"))
    await e.gateway_service.register_gateway(name="t", url="inproc://t",
                                             client=make_fake_time_upstream())
    return e


def test_support_bundle_redacts_secrets(run):
    async def go():
        e = await _engine()
        b = SupportBundle(e).collect()
        assert b["settings"]["jwt_secret_key"] == "***"
        assert "sekrit-value" not in str(b)
        assert b["entities"]["tool"] == 3 and b["entities"]["gateway"] == 1
        assert b["gateways"][0]["name"] == "t"
        assert isinstance(b["plugins"], list) and b["plugins"]
        await e.shutdown()

    run(go())


def test_performance_snapshots_accumulate(run):
    async def go():
        e = await _engine()
        perf = PerformanceService(e, max_snapshots=3)
        for _ in range(5):
            s = perf.snapshot()
        assert "rss_mb" in s or "requests" in s or s, s
        hist = perf.history()
        assert len(hist) == 3  # ring capped at max_snapshots
        await e.shutdown()

    run(go())


def test_toolops_schema_fallback_and_run(run):
    """With no LLM provider configured, generate_tests synthesizes cases
    from the input schema and run_tests executes them through the real
    dispatch path."""

    async def go():
        e = await _engine()
        ops = ToolOps(e)
        cases = await ops.generate_tests("t-convert_time", count=2)
        assert len(cases) == 2
        assert all(isinstance(c, dict) for c in cases)
        assert cases[0].get("time") == "example"  # schema-derived
        rep = await ops.run_tests("t-echo", count=2)
        assert rep["total"] == 2 and rep["passed"] == 2, rep
        with pytest.raises(KeyError):
            await ops.generate_tests("no-such-tool")
        await e.shutdown()

    run(go())
