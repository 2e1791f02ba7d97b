"""Admin UI: server-rendered partials + a minimal hx-* loader.

Reference analog: mcpgateway/admin.py HTMX partial endpoints +
templates/admin.html + admin_ui/ (35k LoC of JS). Same architecture,
MI355X-image constraints: there is no egress to fetch the htmx library, so
a ~60-line vanilla-JS loader implements the subset the partials use
(hx-get / hx-post / hx-delete / hx-target / hx-confirm / hx-refresh forms)
and every action goes through the SAME public REST/admin API the CLI and
tests use — the UI holds no privileged endpoints.

Layout: GET /admin serves the shell (tabs + loader); each tab loads
GET /admin/ui/{partial}; action buttons call the REST API and re-render
their partial. Tested by the httpx-DOM tier (tests/test_admin_ui.py).
"""

from __future__ import annotations

import html
import json
import time
from typing import Any, Dict, List, Optional

E = html.escape

# -- the loader: the hx-* subset the partials use ---------------------------
MINI_HTMX_JS = r"""
'use strict';
function hxSwap(target, html_) {
  const el = document.querySelector(target); if (el) el.innerHTML = html_;
  bindHx(el || document);
}
async function hxDo(el) {
  const conf = el.getAttribute('hx-confirm');
  if (conf && !window.confirm(conf)) return;
  let url = el.getAttribute('hx-get') || el.getAttribute('hx-post') || el.getAttribute('hx-delete');
  const method = el.hasAttribute('hx-post') ? 'POST' : (el.hasAttribute('hx-delete') ? 'DELETE' : 'GET');
  const target = el.getAttribute('hx-target') || '#content';
  const opts = {method: method, headers: {}};
  if (el.tagName === 'FORM') {
    const data = {};
    new FormData(el).forEach((v, k) => { if (v !== '') data[k] = v; });
    if (method === 'GET') {
      url += (url.includes('?') ? '&' : '?') + new URLSearchParams(data).toString();
    } else {
      opts.body = JSON.stringify(data);
      opts.headers['Content-Type'] = 'application/json';
    }
  }
  const resp = await fetch(url, opts);
  const after = el.getAttribute('hx-after');   // partial to (re)load after an action
  if (after) {
    const r2 = await fetch(after);
    hxSwap(target, await r2.text());
  } else if (resp.headers.get('content-type') && resp.headers.get('content-type').includes('text/html')) {
    hxSwap(target, await resp.text());
  }
  if (!resp.ok) {
    const msg = document.createElement('div');
    msg.className = 'flash-error';
    msg.textContent = 'Error ' + resp.status + ': ' + (await resp.text()).slice(0, 300);
    document.querySelector(target).prepend(msg);
  }
}
function bindHx(root) {
  (root || document).querySelectorAll('[hx-get],[hx-post],[hx-delete]').forEach(el => {
    if (el._hx) return; el._hx = true;
    if (el.tagName === 'FORM') {
      el.addEventListener('submit', ev => { ev.preventDefault(); hxDo(el); });
    } else {
      el.addEventListener('click', ev => { ev.preventDefault(); hxDo(el); });
    }
  });
}
function loadTab(name) {
  document.querySelectorAll('nav a').forEach(a => a.classList.toggle('active', a.dataset.tab === name));
  fetch('/admin/ui/' + name).then(r => r.text()).then(t => hxSwap('#content', t));
  window.location.hash = name;
}
window.addEventListener('DOMContentLoaded', () => {
  bindHx(document);
  loadTab(window.location.hash ? window.location.hash.slice(1) : 'dashboard');
});
"""

CSS = """
body{font-family:system-ui,sans-serif;margin:0;background:#f6f7f9;color:#1c2733}
header{background:#13151c;color:#fff;padding:10px 20px;display:flex;align-items:baseline;gap:16px}
header h1{font-size:17px;margin:0} header .sub{color:#8ea0b5;font-size:12px}
nav{background:#20232e;padding:0 12px;display:flex;flex-wrap:wrap}
nav a{color:#aab8c8;text-decoration:none;padding:9px 11px;font-size:13px;cursor:pointer}
nav a.active,nav a:hover{color:#fff;border-bottom:2px solid #e8512f}
#content{padding:18px 22px;max-width:1280px}
table{border-collapse:collapse;width:100%;background:#fff;font-size:13px;box-shadow:0 1px 2px #0001}
th{background:#eef1f5;text-align:left;padding:6px 9px;border-bottom:2px solid #d6dde6;font-size:12px}
td{padding:5px 9px;border-bottom:1px solid #edf0f4;vertical-align:top;max-width:360px;overflow:hidden;
   text-overflow:ellipsis;white-space:nowrap}
tr:hover td{background:#f7fafc}
.pill{display:inline-block;padding:1px 8px;border-radius:9px;font-size:11px}
.ok{background:#d9f2e3;color:#136c34}.bad{background:#fde3e0;color:#a1271b}.warn{background:#fdf0d4;color:#8a6207}
button,.btn{font-size:11px;padding:2px 8px;margin:0 2px;border:1px solid #c6cfd9;border-radius:4px;
  background:#fff;cursor:pointer} button:hover{background:#eef3f8}
.danger{color:#a1271b;border-color:#e2b4ae}
.cards{display:flex;gap:14px;flex-wrap:wrap;margin-bottom:18px}
.card{background:#fff;border-radius:8px;padding:12px 18px;box-shadow:0 1px 3px #0002;min-width:130px}
.card .num{font-size:26px;font-weight:700}.card .lbl{font-size:12px;color:#61728a}
form.inline{background:#fff;padding:12px;border-radius:8px;margin:12px 0;box-shadow:0 1px 3px #0001}
form.inline input,form.inline select{margin:3px 6px 3px 0;padding:4px 6px;font-size:13px;
  border:1px solid #c6cfd9;border-radius:4px}
.flash-error{background:#fde3e0;color:#a1271b;padding:6px 10px;border-radius:5px;margin-bottom:10px;font-size:12px}
pre{background:#fff;padding:12px;border-radius:8px;overflow:auto;font-size:12px;box-shadow:0 1px 3px #0001}
h2{font-size:15px;margin:18px 0 8px}
.muted{color:#7587a0;font-size:12px}
"""

TABS = ["dashboard", "gateways", "tools", "servers", "resources", "prompts", "a2a",
        "plugins", "bindings", "metrics", "logs", "traces", "audit", "tokens", "runtime"]


def render_admin_page(engine) -> str:
    """The shell: tabs + loader; content arrives as partials."""
    nav = "".join(f'<a data-tab="{t}" onclick="loadTab(\'{t}\')">{t.capitalize()}</a>' for t in TABS)
    return (
        "<!doctype html><html><head><title>MCP Context Forge AMD — Admin</title>"
        f"<style>{CSS}</style><script>{MINI_HTMX_JS}</script></head><body>"
        "<header><h1>MCP Context Forge AMD</h1>"
        f"<span class='sub'>MI355X-native gateway · rank {engine.rank}/{engine.world_size}"
        f" · GPU {'on' if engine.gpu_pipeline else 'off'}</span></header>"
        f"<nav>{nav}</nav><div id='content'><p class='muted'>loading…</p></div>"
        "</body></html>"
    )


# ---------------------------------------------------------------- helpers


def _table(rows: List[Dict[str, Any]], cols: List[str], actions=None, limit: int = 500) -> str:
    if not rows:
        return "<p class='muted'>none</p>"
    head = "".join(f"<th>{E(c)}</th>" for c in cols) + ("<th>actions</th>" if actions else "")
    out = [f"<table><tr>{head}</tr>"]
    for r in rows[:limit]:
        tds = []
        for c in cols:
            v = r.get(c, "")
            if c in ("enabled", "reachable"):
                v = f"<span class='pill {'ok' if v else 'bad'}'>{'yes' if v else 'no'}</span>"
            elif c == "status":
                cls = {"active": "ok", "failed": "bad", "unreachable": "bad"}.get(str(v), "warn")
                v = f"<span class='pill {cls}'>{E(str(v))}</span>"
            else:
                v = E(str(v))[:160]
            tds.append(f"<td>{v}</td>")
        if actions:
            tds.append(f"<td>{actions(r)}</td>")
        out.append("<tr>" + "".join(tds) + "</tr>")
    out.append("</table>")
    if len(rows) > limit:
        out.append(f"<p class='muted'>showing {limit} of {len(rows)}</p>")
    return "".join(out)


def _act(label: str, method: str, url: str, partial: str, confirm: str = "",
         danger: bool = False) -> str:
    attr = {"GET": "hx-get", "POST": "hx-post", "DELETE": "hx-delete"}[method]
    conf = f" hx-confirm=\"{E(confirm)}\"" if confirm else ""
    cls = "btn danger" if danger else "btn"
    return (f"<button class='{cls}' {attr}='{E(url)}' hx-after='/admin/ui/{partial}'"
            f" hx-target='#content'{conf}>{E(label)}</button>")


# ---------------------------------------------------------------- partials


def partial_dashboard(engine) -> str:
    counts = {k: len(engine.registry.list(k, include_disabled=True))
              for k in ("tool", "gateway", "server", "resource", "prompt", "a2a_agent")}
    m = engine.metrics.snapshot()
    cards = "".join(
        f"<div class='card'><div class='num'>{v}</div><div class='lbl'>{E(k)}s</div></div>"
        for k, v in counts.items())
    cards += (f"<div class='card'><div class='num'>{engine.sessions.count()}</div>"
              f"<div class='lbl'>live sessions</div></div>")
    total = sum(row.get("count", 0) for row in m.get("top_tools", []))
    cards += (f"<div class='card'><div class='num'>{total}</div>"
              f"<div class='lbl'>tool calls (buffered)</div></div>")
    gpu = ""
    if engine.gpu_pipeline:
        st = engine.gpu_pipeline.stats()
        gpu = ("<h2>GPU pipeline</h2><div class='cards'>" + "".join(
            f"<div class='card'><div class='num'>{st.get(k, 0)}</div><div class='lbl'>{k}</div></div>"
            for k in ("batches", "requests", "fast_path", "slow_path", "blocked", "cache_hits"))
            + "</div>")
    up = round(time.time() - engine.started_at, 1)
    return (f"<div class='cards'>{cards}</div>{gpu}"
            f"<p class='muted'>uptime {up}s · protocol {E(engine.settings.protocol_version)}"
            f" · world {engine.world_size}</p>")


_ENTITY_COLS = {
    "gateway": ["name", "url", "transport", "status", "reachable", "retry_count",
                "failure_class", "owner_rank"],
    "tool": ["name", "integration_type", "enabled", "reachable", "gateway_id"],
    "server": ["name", "enabled", "associated_tools"],
    "resource": ["uri", "name", "mime_type", "enabled"],
    "prompt": ["name", "description", "enabled"],
    "a2a_agent": ["name", "endpoint_url", "agent_type", "enabled"],
}
_ENTITY_PLURAL = {"gateway": "gateways", "tool": "tools", "server": "servers",
                  "resource": "resources", "prompt": "prompts", "a2a_agent": "a2a"}


def partial_entities(engine, kind: str) -> str:
    plural = _ENTITY_PLURAL[kind]
    rows = engine.registry.list(kind, include_disabled=True)

    def actions(r):
        eid = r.get("id", "")
        on = r.get("enabled", True)
        acts = [_act("disable" if on else "enable", "POST",
                     f"/{plural}/{eid}/toggle?activate={'false' if on else 'true'}", plural)]
        if kind == "gateway":
            acts.append(_act("refresh", "POST", f"/gateways/{eid}/refresh", plural))
            if r.get("status") == "failed":
                acts.append(_act("retry", "POST", f"/gateways/{eid}/retry", plural))
        acts.append(_act("delete", "DELETE", f"/{plural}/{eid}", plural,
                         confirm=f"Delete {r.get('name', r.get('uri', eid))}?", danger=True))
        return "".join(acts)

    create = ""
    if kind == "gateway":
        create = (
            f"<form class='inline' hx-post='/gateways' hx-after='/admin/ui/{plural}' hx-target='#content'>"
            "<b>Register gateway</b><br>"
            "<input name='name' placeholder='name' required>"
            "<input name='url' placeholder='http://host:port/mcp' size='34'>"
            "<select name='transport'><option>streamablehttp</option><option>sse</option></select>"
            "<label><input type='checkbox' name='defer' value='true'> async (pending lifecycle)</label>"
            "<button type='submit'>register</button></form>")
    elif kind == "tool":
        create = (
            f"<form class='inline' hx-post='/tools' hx-after='/admin/ui/{plural}' hx-target='#content'>"
            "<b>Create REST tool</b><br>"
            "<input name='name' placeholder='name' required>"
            "<input name='url' placeholder='https://api…' size='34'>"
            "<input name='description' placeholder='description' size='28'>"
            "<button type='submit'>create</button></form>")
    title = f"<h2>{plural.capitalize()} ({len(rows)})</h2>"
    return title + create + _table(rows, _ENTITY_COLS[kind], actions)


def partial_plugins(engine) -> str:
    rows = []
    for p in sorted(engine.plugins.plugins, key=lambda x: x.priority):
        rows.append({"name": p.name, "mode": p.mode.value, "priority": p.priority,
                     "hooks": ", ".join(h.value for h in p.hooks)})

    def actions(r):
        out = []
        for mode in ("enforce", "permissive", "disabled"):
            if mode != r["mode"]:
                out.append(_act(mode, "POST", f"/admin/plugins/{r['name']}/mode?mode={mode}",
                                "plugins"))
        return "".join(out)

    return f"<h2>Plugins ({len(rows)})</h2>" + _table(rows, ["name", "mode", "priority", "hooks"], actions)


def partial_bindings(engine) -> str:
    rows = engine.registry.list("plugin_binding")

    def actions(r):
        return _act("unbind", "DELETE",
                    f"/tools/{r['tool_name']}/plugin-bindings/{r['plugin_name']}", "bindings",
                    confirm="Remove binding?", danger=True)

    form = ("<form class='inline' id='bindform' "
            "hx-post='/admin/ui/bind' hx-after='/admin/ui/bindings' hx-target='#content'>"
            "<b>Bind plugin to tool</b><br>"
            "<input name='tool_name' placeholder='tool name' required>"
            "<input name='plugin_name' placeholder='plugin name' required>"
            "<select name='mode'><option value=''>keep mode</option><option>enforce</option>"
            "<option>permissive</option><option>disabled</option></select>"
            "<button type='submit'>bind</button></form>")
    return (f"<h2>Plugin bindings ({len(rows)})</h2>" + form +
            _table(rows, ["tool_name", "plugin_name", "mode", "enabled"], actions))


def partial_metrics(engine) -> str:
    snap = engine.metrics.snapshot()
    tools = [{**t, "avg_ms": round(t.get("avg_ms", 0), 3)} for t in snap.get("top_tools", [])]
    # hourly rollups (same path as GET /admin/metrics/rollups)
    from sqlalchemy import select

    from ..db.models import DbMetricRollup
    from ..services.metrics import rollup_hourly

    engine.metrics.flush()
    rollup_hourly(engine.db)
    with engine.db.session() as s_:
        rows = s_.execute(select(DbMetricRollup).order_by(DbMetricRollup.hour.desc())
                          .limit(48)).scalars().all()
        roll = [{"hour": str(r.hour), "entity_id": r.entity_id, "count": r.count,
                 "errors": r.error_count,
                 "avg_ms": round((r.total_ms / r.count) if r.count else 0, 3)} for r in rows]
    return ("<h2>Live buffer (top tools)</h2>" + _table(tools, ["tool_id", "count", "avg_ms"]) +
            f"<p class='muted'>counters: {E(json.dumps(snap.get('counters', {})))}"
            f" · pending rows: {snap.get('pending_rows', 0)}</p>" +
            "<h2>Hourly rollups</h2>" + _table(roll, ["hour", "entity_id", "count", "errors", "avg_ms"]))


def partial_logs(engine, q: str = "", level: str = "", limit: int = 100) -> str:
    from sqlalchemy import select

    from ..db.models import DbStructuredLog

    stmt = select(DbStructuredLog).order_by(DbStructuredLog.id.desc()).limit(min(limit, 1000))
    if level:
        stmt = stmt.where(DbStructuredLog.level == level.upper())
    if q:
        stmt = stmt.where(DbStructuredLog.message.like(f"%{q}%"))
    with engine.db.session() as s_:
        rows = [{"timestamp": r.timestamp.isoformat(), "level": r.level,
                 "logger": r.logger, "message": r.message}
                for r in s_.execute(stmt).scalars()]
    form = ("<form class='inline' hx-get='/admin/ui/logs' hx-target='#content'>"
            f"<b>Log search</b><br><input name='q' placeholder='substring' value='{E(q)}'>"
            "<select name='level'><option value=''>any level</option><option>INFO</option>"
            "<option>WARNING</option><option>ERROR</option></select>"
            "<button type='submit'>search</button></form>")
    return "<h2>Structured logs</h2>" + form + _table(rows, ["timestamp", "level", "logger", "message"])


def partial_traces(engine) -> str:
    engine.observability.flush()
    spans = engine.observability.query_traces(200)
    spans = [{**sp, "duration_ms": round(sp.get("duration_ms", 0), 3),
              "attributes": json.dumps(sp.get("attributes") or {})[:120]} for sp in spans]
    return ("<h2>Traces (self-hosted spans)</h2>" +
            _table(spans, ["trace_id", "name", "duration_ms", "status", "attributes"]))


def partial_audit(engine) -> str:
    rows = engine.audit.query(limit=200)
    return ("<h2>Audit trail</h2>" +
            _table(rows, ["timestamp", "actor", "action", "entity_type", "entity_id"]))


def partial_tokens(engine, auth) -> str:
    rows = auth.list_api_tokens(engine.settings.platform_admin_email) if auth else []

    def actions(r):
        if r.get("revoked"):
            return "<span class='pill bad'>revoked</span>"
        return _act("revoke", "DELETE", f"/tokens/{r['id']}", "tokens",
                    confirm="Revoke this token?", danger=True)

    return ("<h2>API tokens (platform admin)</h2>" +
            _table(rows, ["id", "name", "server_id", "scopes"], actions))


def partial_runtime(engine, collector=None, edge_stats: Optional[dict] = None) -> str:
    s = engine.settings
    form = ("<form class='inline' hx-post='/admin/ui/runtime' hx-after='/admin/ui/runtime'"
            " hx-target='#content'><b>Live runtime knobs</b><br>"
            f"<label>batch window µs <input name='window_us' size=8"
            f" value='{int(collector.window_s * 1e6) if collector else s.gpu_batch_window_us}'></label>"
            f"<label>max batch <input name='max_batch' size=8"
            f" value='{collector.max_batch if collector else s.gpu_batch_max_requests}'></label>"
            "<button type='submit'>apply</button></form>")
    info = {"gpu": bool(engine.gpu_pipeline), "world_size": engine.world_size,
            "rank": engine.rank,
            "collector": None if collector is None else
            {"window_us": int(collector.window_s * 1e6), "max_batch": collector.max_batch,
             "batches": collector.batches, "max_seen": collector.max_seen},
            "native_edge": edge_stats,
            "plugins_enabled": s.plugins_enabled,
            "federation_enabled": s.federation_enabled,
            "leader": engine.leader_elector.is_leader if engine.leader_elector else "rank0-by-construction"}
    return "<h2>Runtime</h2>" + form + "<pre>" + E(json.dumps(info, indent=2, default=str)) + "</pre>"
