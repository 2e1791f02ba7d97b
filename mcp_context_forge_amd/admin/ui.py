"""Minimal admin UI (reference analog: mcpgateway/admin_ui + templates/admin.html,
HTMX+Alpine — here a single self-contained page over the same admin API)."""

from __future__ import annotations

import html
import json


def render_admin_page(engine) -> str:
    ents = {k: engine.registry.list(k) for k in ("tool", "gateway", "server", "resource", "prompt", "a2a_agent")}
    plugins = [{"name": p.name, "mode": p.mode.value, "priority": p.priority} for p in engine.plugins.plugins]
    gpu = engine.gpu_pipeline.stats() if engine.gpu_pipeline else None

    def table(rows, cols):
        if not rows:
            return "<p><em>none</em></p>"
        head = "".join(f"<th>{html.escape(c)}</th>" for c in cols)
        body = "".join(
            "<tr>" + "".join(f"<td>{html.escape(str(r.get(c, '')))[:80]}</td>" for c in cols) + "</tr>"
            for r in rows[:200])
        return f"<table border=1 cellpadding=4 cellspacing=0><tr>{head}</tr>{body}</table>"

    sections = []
    sections.append(f"<h2>Gateways ({len(ents['gateway'])})</h2>" +
                    table(ents["gateway"], ["name", "url", "transport", "status", "reachable", "owner_rank"]))
    sections.append(f"<h2>Tools ({len(ents['tool'])})</h2>" +
                    table(ents["tool"], ["name", "integration_type", "enabled", "reachable", "gateway_id"]))
    sections.append(f"<h2>Virtual servers ({len(ents['server'])})</h2>" +
                    table(ents["server"], ["name", "enabled", "associated_tools"]))
    sections.append(f"<h2>Resources ({len(ents['resource'])})</h2>" + table(ents["resource"], ["uri", "name", "mime_type"]))
    sections.append(f"<h2>Prompts ({len(ents['prompt'])})</h2>" + table(ents["prompt"], ["name", "description"]))
    sections.append(f"<h2>A2A agents ({len(ents['a2a_agent'])})</h2>" +
                    table(ents["a2a_agent"], ["name", "endpoint_url", "agent_type", "enabled"]))
    sections.append(f"<h2>Plugins</h2>" + table(plugins, ["name", "mode", "priority"]))
    bindings = engine.registry.list("plugin_binding")
    if bindings:
        sections.append(f"<h2>Plugin bindings ({len(bindings)})</h2>" +
                        table(bindings, ["tool_name", "plugin_name", "mode", "enabled"]))
    tags = engine.tags.list_tags()
    if tags:
        sections.append(f"<h2>Tags ({len(tags)})</h2>" + table(tags, ["name", "count"]))
    if gpu:
        sections.append("<h2>GPU pipeline</h2><pre>" + html.escape(json.dumps(gpu, indent=2)) + "</pre>")
    metrics = engine.metrics.snapshot()
    sections.append("<h2>Metrics</h2><pre>" + html.escape(json.dumps(metrics, indent=2, default=str)) + "</pre>")
    api_links = ["stats", "traces", "audit", "logs", "metrics/rollups", "siem/export",
                 "compliance/report", "classification", "runtime", "performance",
                 "support-bundle", "plugins"]
    nav = " · ".join(f'<a href="/admin/{p}">{p}</a>' for p in api_links)
    return (
        "<!doctype html><html><head><title>MCP Context Forge AMD — Admin</title>"
        "<style>body{font-family:sans-serif;margin:2em}table{border-collapse:collapse;font-size:13px}"
        "th{background:#eee;text-align:left}</style></head><body>"
        "<h1>MCP Context Forge AMD</h1>"
        f"<p>MI355X-native gateway · sessions: {engine.sessions.count()}</p>"
        f"<p>API: {nav}</p>"
        + "".join(sections) + "</body></html>"
    )
