"""Plugin framework.

Reference analog: the external `cpex` package + mcpgateway/plugins glue
(mcpgateway/plugins/__init__.py — hook set and modes at
plugins/README.md:40-66; TenantPluginManager.invoke_hook call sites e.g.
services/tool_service.py:5530). Hook names and mode semantics match the
reference so plugin configs port 1:1.

MI355X-native addition: every plugin may implement a **batched** interface
(`batch_tool_pre`, `batch_tool_post`) operating on a staged
:class:`~mcp_context_forge_amd.gpu.batch.RequestBatch`. The GPU pipeline
invokes the batched form over the whole micro-batch (HIP kernels under the
hood); the per-request async form is the CPU/reference path and the parity
oracle for tests (SURVEY.md §4 plugin-parity gate).
"""

from __future__ import annotations

import asyncio
import enum
import fnmatch
import logging
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence

logger = logging.getLogger(__name__)


class HookType(str, enum.Enum):
    TOOL_PRE_INVOKE = "tool_pre_invoke"
    TOOL_POST_INVOKE = "tool_post_invoke"
    PROMPT_PRE_FETCH = "prompt_pre_fetch"
    PROMPT_POST_FETCH = "prompt_post_fetch"
    RESOURCE_PRE_FETCH = "resource_pre_fetch"
    RESOURCE_POST_FETCH = "resource_post_fetch"
    AGENT_PRE_INVOKE = "agent_pre_invoke"
    AGENT_POST_INVOKE = "agent_post_invoke"
    HTTP_PRE_REQUEST = "http_pre_request"
    HTTP_POST_REQUEST = "http_post_request"


class PluginMode(str, enum.Enum):
    """Reference modes (mcpgateway/plugins/__init__.py:71-81)."""

    ENFORCE = "enforce"
    ENFORCE_IGNORE_ERROR = "enforce_ignore_error"
    PERMISSIVE = "permissive"
    DISABLED = "disabled"


class PluginViolationError(Exception):
    """Raised when an enforce-mode plugin blocks the request."""

    def __init__(self, plugin: str, reason: str, code: str = "policy_violation", details: Any = None):
        self.plugin = plugin
        self.reason = reason
        self.code = code
        self.details = details
        super().__init__(f"{plugin}: {reason}")


@dataclass
class PluginResult:
    """Outcome of one hook invocation."""

    continue_processing: bool = True
    modified_payload: Optional[Any] = None  # replaces args (pre) / result (post) when set
    violation: Optional[str] = None
    violation_code: str = "policy_violation"
    metadata: Dict[str, Any] = field(default_factory=dict)

    @classmethod
    def ok(cls, payload: Optional[Any] = None, **meta: Any) -> "PluginResult":
        return cls(continue_processing=True, modified_payload=payload, metadata=meta)

    @classmethod
    def block(cls, reason: str, code: str = "policy_violation", **meta: Any) -> "PluginResult":
        return cls(continue_processing=False, violation=reason, violation_code=code, metadata=meta)


@dataclass
class PluginContext:
    """Per-invocation context handed to hooks."""

    hook: HookType
    name: str = ""                 # tool / prompt / resource / agent name
    args: Any = None               # pre hooks: arguments; post hooks: result payload
    user: Optional[str] = None
    server_id: Optional[str] = None
    headers: Dict[str, str] = field(default_factory=dict)
    state: Dict[str, Any] = field(default_factory=dict)   # cross-hook scratch (per request)
    global_state: Dict[str, Any] = field(default_factory=dict)


class Plugin:
    """Base plugin. Subclasses set `hooks` and override the hook methods.

    Batched GPU interface: override `batch_tool_pre(batch, pipeline)` /
    `batch_tool_post(batch, pipeline)` to participate in the staged GPU
    chain. Default falls back to per-request hooks applied serially.
    """

    name: str = "plugin"
    hooks: Sequence[HookType] = ()
    priority: int = 100
    mode: PluginMode = PluginMode.ENFORCE
    gpu_capable: bool = False

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        self.config = config or {}
        self.priority = int(self.config.get("priority", self.priority))
        self.mode = PluginMode(self.config.get("mode", self.mode.value))
        self.conditions = self.config.get("conditions") or {}

    # -- per-request (reference-compatible) hooks --
    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def prompt_pre_fetch(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def prompt_post_fetch(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def resource_pre_fetch(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def resource_post_fetch(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def agent_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def agent_post_invoke(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def http_pre_request(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def http_post_request(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok()

    async def shutdown(self) -> None:
        pass

    # -- condition matching (reference: plugins/config.yaml `conditions`) --
    def applies_to(self, ctx: PluginContext) -> bool:
        conds = self.conditions
        if not conds:
            return True
        tools = conds.get("tools")
        if tools and ctx.name and not any(fnmatch.fnmatch(ctx.name, p) for p in tools):
            return False
        servers = conds.get("server_ids")
        if servers and ctx.server_id and ctx.server_id not in servers:
            return False
        users = conds.get("user_patterns")
        if users and ctx.user and not any(fnmatch.fnmatch(ctx.user, p) for p in users):
            return False
        return True


_HOOK_METHOD = {h: h.value for h in HookType}


class PluginManager:
    """Ordered hook dispatcher (reference: TenantPluginManager.invoke_hook)."""

    def __init__(self, plugins: Optional[List[Plugin]] = None, enabled: bool = True, timeout_s: float = 5.0):
        self.plugins: List[Plugin] = sorted(plugins or [], key=lambda p: p.priority)
        self.enabled = enabled
        self.timeout_s = timeout_s
        self.version = 0  # bumped on runtime plugin changes (GPU pipeline re-syncs)
        self.global_state: Dict[str, Any] = {}
        self._stats: Dict[str, Dict[str, Any]] = {}
        # per-tool bindings (reference: tool_plugin_bindings db.py:6875):
        # tool name -> plugin name -> {"mode": str|None, "config": dict|None}
        self.bindings: Dict[str, Dict[str, Dict[str, Any]]] = {}
        self._derived: Dict[tuple, Plugin] = {}  # (plugin, tool) -> config-override instance

    def add(self, plugin: Plugin) -> None:
        self.plugins.append(plugin)
        self.plugins.sort(key=lambda p: p.priority)
        self.version += 1

    def bump(self) -> None:
        """Signal runtime plugin-config change (mode flips etc.)."""
        self.version += 1

    def get(self, name: str) -> Optional[Plugin]:
        for p in self.plugins:
            if p.name == name:
                return p
        return None

    def for_hook(self, hook: HookType) -> List[Plugin]:
        return [p for p in self.plugins if hook in p.hooks and p.mode != PluginMode.DISABLED]

    # -- per-tool bindings -------------------------------------------------
    def set_bindings(self, bindings: Dict[str, Dict[str, Dict[str, Any]]]) -> None:
        """Replace the whole binding map (loaded from the registry)."""
        self.bindings = bindings
        self._derived.clear()
        self.version += 1

    def bindings_for_tool(self, tool: Optional[str]) -> Dict[str, Dict[str, Any]]:
        return self.bindings.get(tool, {}) if tool else {}

    def effective_mode(self, plugin: Plugin, tool: Optional[str]) -> PluginMode:
        b = self.bindings_for_tool(tool).get(plugin.name)
        if b and b.get("mode"):
            return PluginMode(b["mode"])
        return plugin.mode

    def _effective_plugin(self, plugin: Plugin, tool: Optional[str]) -> tuple:
        """(instance, mode) for this tool: a binding may flip the mode or
        derive a config-override instance (cached per tool)."""
        b = self.bindings_for_tool(tool).get(plugin.name)
        if not b:
            return plugin, plugin.mode
        mode = PluginMode(b["mode"]) if b.get("mode") else plugin.mode
        if b.get("config"):
            key = (plugin.name, tool)
            inst = self._derived.get(key)
            if inst is None:
                cfg = {**plugin.config, **b["config"], "mode": mode.value, "priority": plugin.priority}
                inst = type(plugin)(cfg)
                inst.name = plugin.name
                self._derived[key] = inst
            return inst, mode
        return plugin, mode

    def _record(self, plugin: Plugin, hook: HookType, ok: bool, blocked: bool, ms: float) -> None:
        st = self._stats.setdefault(plugin.name, {"calls": 0, "errors": 0, "blocked": 0, "total_ms": 0.0})
        st["calls"] += 1
        st["total_ms"] += ms
        if not ok:
            st["errors"] += 1
        if blocked:
            st["blocked"] += 1

    def stats(self) -> Dict[str, Dict[str, Any]]:
        return self._stats

    async def invoke_hook(self, hook: HookType, ctx: PluginContext) -> PluginContext:
        """Run all plugins registered for `hook` in priority order.

        Mode semantics (reference plugins/README.md): enforce → violation
        raises; permissive → violation logged, continue; enforce_ignore_error
        → plugin *exceptions* ignored but violations enforced.
        """
        if not self.enabled:
            return ctx
        ctx.global_state = self.global_state
        import time as _t

        tool = ctx.name or None
        for plugin in self.plugins:
            if hook not in plugin.hooks:
                continue
            eff, mode = self._effective_plugin(plugin, tool) if self.bindings else (plugin, plugin.mode)
            if mode == PluginMode.DISABLED:
                continue
            if not eff.applies_to(ctx):
                continue
            t0 = _t.monotonic()
            try:
                method = getattr(eff, _HOOK_METHOD[hook])
                result: PluginResult = await asyncio.wait_for(method(ctx), timeout=self.timeout_s)
                ms = (_t.monotonic() - t0) * 1000.0
                blocked = not result.continue_processing
                self._record(plugin, hook, True, blocked, ms)
                if blocked:
                    if mode == PluginMode.PERMISSIVE:
                        logger.warning("plugin %s would block (%s) - permissive", plugin.name, result.violation)
                        continue
                    raise PluginViolationError(plugin.name, result.violation or "blocked", result.violation_code, result.metadata)
                if result.modified_payload is not None:
                    ctx.args = result.modified_payload
            except PluginViolationError:
                raise
            except Exception as exc:
                ms = (_t.monotonic() - t0) * 1000.0
                self._record(plugin, hook, False, False, ms)
                if mode in (PluginMode.ENFORCE_IGNORE_ERROR, PluginMode.PERMISSIVE):
                    logger.warning("plugin %s error ignored (%s mode): %s", plugin.name, mode.value, exc)
                    continue
                raise
        return ctx

    async def shutdown(self) -> None:
        for p in self.plugins:
            try:
                await p.shutdown()
            except Exception:
                pass
