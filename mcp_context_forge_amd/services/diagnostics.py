"""Diagnostics: support bundle + performance snapshots + toolops.

Reference analogs: services/support_bundle.py (diagnostic dump),
services/performance_service.py + PerformanceSnapshot (db.py:3160),
mcpgateway/toolops/ (LLM-assisted tool testing via ALTK).
"""

from __future__ import annotations

import json
import platform
import sys
import time
from typing import Any, Dict, List, Optional


class SupportBundle:
    """Collect a redacted diagnostic snapshot (reference: support_bundle.py)."""

    REDACT = {"jwt_secret_key", "basic_auth_password", "platform_admin_password", "auth_value", "api_key"}

    def __init__(self, engine):
        self.engine = engine

    def collect(self) -> Dict[str, Any]:
        import torch

        settings = {k: ("***" if k in self.REDACT else v)
                    for k, v in self.engine.settings.model_dump().items()}
        bundle: Dict[str, Any] = {
            "generated_at": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            "version": self.engine.version_info(),
            "platform": {"python": sys.version.split()[0], "system": platform.platform(),
                         "torch": torch.__version__, "cuda_available": torch.cuda.is_available()},
            "settings": settings,
            "entities": {k: len(self.engine.registry.list(k))
                         for k in ("tool", "gateway", "server", "resource", "prompt", "a2a_agent")},
            "gateways": [{"name": g["name"], "status": g.get("status"), "reachable": g.get("reachable")}
                         for g in self.engine.registry.list("gateway")],
            "metrics": self.engine.metrics.snapshot(),
            "sessions": self.engine.sessions.count(),
            "plugins": [{"name": p.name, "mode": p.mode.value} for p in self.engine.plugins.plugins],
        }
        if self.engine.gpu_pipeline is not None:
            bundle["gpu_pipeline"] = self.engine.gpu_pipeline.stats()
        return bundle


class PerformanceService:
    """Rolling performance snapshots (reference: performance_service.py)."""

    def __init__(self, engine, max_snapshots: int = 288):
        self.engine = engine
        self.max_snapshots = max_snapshots
        self.snapshots: List[Dict[str, Any]] = []
        self._last_counters: Dict[str, float] = {}
        self._last_ts = time.monotonic()

    def snapshot(self) -> Dict[str, Any]:
        now = time.monotonic()
        counters = dict(self.engine.metrics.counters)
        dt = max(now - self._last_ts, 1e-6)
        rps = (counters.get("tool_invocations_total", 0) -
               self._last_counters.get("tool_invocations_total", 0)) / dt
        snap = {
            "ts": time.time(),
            "interval_s": round(dt, 1),
            "tool_rps": round(rps, 2),
            "totals": counters,
            "sessions": self.engine.sessions.count(),
        }
        if self.engine.gpu_pipeline is not None:
            st = self.engine.gpu_pipeline.stats()
            snap["gpu"] = {k: st[k] for k in ("requests", "fast_path", "slow_path", "blocked", "cache_hits")}
        self._last_counters = counters
        self._last_ts = now
        self.snapshots.append(snap)
        if len(self.snapshots) > self.max_snapshots:
            self.snapshots = self.snapshots[-self.max_snapshots:]
        return snap

    def history(self, limit: int = 100) -> List[Dict[str, Any]]:
        return self.snapshots[-limit:]


class ToolOps:
    """LLM-assisted tool testing/enrichment (reference: mcpgateway/toolops/).

    Uses the configured LLM proxy to (a) propose test invocations for a tool
    from its schema and (b) draft improved descriptions. Falls back to
    schema-derived synthesis when no provider is configured.
    """

    def __init__(self, engine):
        self.engine = engine

    def _example_from_schema(self, schema: Optional[dict]) -> dict:
        if not schema or not isinstance(schema, dict):
            return {}
        out = {}
        for key, sub in (schema.get("properties") or {}).items():
            ty = sub.get("type") if isinstance(sub, dict) else None
            out[key] = {"string": "example", "integer": 1, "number": 1.5,
                        "boolean": True, "array": [], "object": {}}.get(ty, "example")
        return out

    async def generate_tests(self, tool_name: str, count: int = 3) -> List[dict]:
        tool = self.engine.registry.find("tool", tool_name)
        if tool is None:
            raise KeyError(f"tool {tool_name} not found")
        schema = tool.get("input_schema") or {}
        cases: List[dict] = []
        if self.engine.llm_proxy.registry.providers:
            try:
                out = await self.engine.llm_proxy.chat_completions({
                    "messages": [{"role": "user", "content":
                                  f"Generate {count} JSON test argument objects (one per line, raw JSON) "
                                  f"for a tool with JSON schema: {json.dumps(schema)}"}],
                    "max_tokens": 512,
                })
                text = out["choices"][0]["message"]["content"]
                for line in text.splitlines():
                    line = line.strip().strip("`")
                    if line.startswith("{"):
                        try:
                            cases.append(json.loads(line))
                        except ValueError:
                            pass
            except Exception:
                pass
        while len(cases) < count:
            cases.append(self._example_from_schema(schema))
        return cases[:count]

    async def run_tests(self, tool_name: str, count: int = 3) -> dict:
        cases = await self.generate_tests(tool_name, count)
        results = []
        for args in cases:
            try:
                res = await self.engine.tool_service.invoke_tool(tool_name, args, user="toolops")
                results.append({"arguments": args, "ok": not res.get("isError", False)})
            except Exception as exc:
                results.append({"arguments": args, "ok": False, "error": str(exc)})
        passed = sum(1 for r in results if r["ok"])
        return {"tool": tool_name, "total": len(results), "passed": passed, "results": results}
