"""Integration plugins: external-service and LLM-backed hooks.

Reference analogs: plugins/summarizer (LLM post-summarization),
plugins/virus_total_checker (URL/hash reputation), plugins/vault (OAuth
token injection from stored credentials), plugins/unified_pdp (policy
decision point with pluggable engines — OPA/cedar in the reference; a
local rule engine here, with the same allow/deny decision contract).

External HTTP calls go through an injectable async client so the plugins
are testable offline (this image has no egress).
"""

from __future__ import annotations

import fnmatch
import hashlib
import re
from typing import Any, Dict, List, Optional

from .builtin import _text_of, _walk_strings
from .framework import HookType, Plugin, PluginContext, PluginResult


class SummarizerPlugin(Plugin):
    """Summarize oversized tool results through a configured LLM provider
    (reference: plugins/summarizer — tool_post hook, threshold-gated)."""

    name = "summarizer"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 905

    def __init__(self, config=None):
        super().__init__(config)
        self.threshold = int(self.config.get("threshold_chars", 4000))
        self.max_tokens = int(self.config.get("max_tokens", 256))
        self.prompt = self.config.get(
            "prompt", "Summarize the following tool output concisely:")
        self.llm = self.config.get("llm")  # injected LLMProxyService (or None)
        self.summarized = 0

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        if self.llm is None:
            return PluginResult.ok()
        text = _text_of(ctx.args)
        if len(text) < self.threshold:
            return PluginResult.ok()
        try:
            out = await self.llm.chat_completions({
                "messages": [{"role": "user", "content": f"{self.prompt}\n\n{text[:20000]}"}],
                "max_tokens": self.max_tokens,
            })
            summary = (out.get("choices") or [{}])[0].get("message", {}).get("content", "")
        except Exception:
            return PluginResult.ok()  # fail open: keep the original result
        if not summary:
            return PluginResult.ok()
        self.summarized += 1
        return PluginResult.ok({"content": [{"type": "text", "text": summary}],
                                "isError": False,
                                "_summarized_from_chars": len(text)})


_URL_RE = re.compile(r"https?://[^\s\"'<>]+")


class VirusTotalCheckerPlugin(Plugin):
    """URL / file-hash reputation checks (reference: plugins/virus_total_checker).

    Scans argument strings for URLs and hash-looking tokens, queries the
    reputation client (injectable; VirusTotal-compatible verdict dict), and
    blocks when malicious counts exceed the threshold. Verdicts are cached."""

    name = "virus_total_checker"
    hooks = (HookType.TOOL_PRE_INVOKE,)
    priority = 34

    def __init__(self, config=None):
        super().__init__(config)
        self.client = self.config.get("client")   # async callable(kind, value) -> dict
        self.max_malicious = int(self.config.get("max_malicious", 0))
        self._cache: Dict[str, dict] = {}
        self.checked = 0

    async def _verdict(self, kind: str, value: str) -> dict:
        v = self._cache.get(value)
        if v is None and self.client is not None:
            try:
                v = await self.client(kind, value)
            except Exception:
                v = {"malicious": 0, "error": True}  # fail open
            if len(self._cache) > 4096:
                self._cache.clear()
            self._cache[value] = v
            self.checked += 1
        return v or {"malicious": 0}

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        if self.client is None:
            return PluginResult.ok()
        text = _text_of(ctx.args)
        targets = [("url", u) for u in _URL_RE.findall(text)[:16]]
        targets += [("hash", h) for h in re.findall(r"\b[a-fA-F0-9]{64}\b", text)[:16]]
        for kind, val in targets:
            v = await self._verdict(kind, val)
            if int(v.get("malicious", 0)) > self.max_malicious:
                return PluginResult.block(
                    f"virus_total: {kind} {val[:80]!r} flagged malicious "
                    f"({v.get('malicious')} engines)", code="malicious_content")
        return PluginResult.ok()


class VaultPlugin(Plugin):
    """Inject stored OAuth/bearer credentials into outbound headers by
    upstream pattern (reference: plugins/vault — token injection from the
    encrypted token store)."""

    name = "vault"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.HTTP_PRE_REQUEST)
    priority = 68

    def __init__(self, config=None):
        super().__init__(config)
        # [{"match": "github-*", "header": "Authorization", "token": "..."} ...]
        # `token_provider` (callable name->token) may be injected for
        # integration with auth/oauth.py client-credentials flows.
        self.rules: List[dict] = list(self.config.get("rules") or [])
        self.token_provider = self.config.get("token_provider")
        self.injected = 0

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        for rule in self.rules:
            if not fnmatch.fnmatch(ctx.name or "", rule.get("match", "*")):
                continue
            header = rule.get("header", "Authorization")
            if header in ctx.headers:
                continue  # never clobber caller credentials
            token = rule.get("token")
            if token is None and self.token_provider is not None:
                token = await self.token_provider(rule.get("token_name", ctx.name))
            if token:
                scheme = rule.get("scheme", "Bearer")
                ctx.headers[header] = f"{scheme} {token}" if scheme else token
                self.injected += 1
        return PluginResult.ok()

    http_pre_request = tool_pre_invoke


class UnifiedPdpPlugin(Plugin):
    """Policy decision point (reference: plugins/unified_pdp with OPA/cedar
    engines). Local rule engine: ordered rules matched on tool/user/argument
    content; first match decides. An external engine can be injected as
    `engine` (async callable(input_dict) -> {"allow": bool, "reason": str})."""

    name = "unified_pdp"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.AGENT_PRE_INVOKE)
    priority = 12

    def __init__(self, config=None):
        super().__init__(config)
        # rule: {"effect": "deny"|"allow", "tools": ["pat"], "users": ["pat"],
        #        "contains": "substr"} — reference policy shapes, flattened
        self.rules: List[dict] = list(self.config.get("rules") or [])
        self.default = self.config.get("default", "allow")
        self.engine = self.config.get("engine")
        self.decisions = 0

    def _matches(self, rule: dict, ctx: PluginContext, text: str) -> bool:
        tools = rule.get("tools")
        if tools and not any(fnmatch.fnmatch(ctx.name or "", p) for p in tools):
            return False
        users = rule.get("users")
        if users and not any(fnmatch.fnmatch(ctx.user or "", p) for p in users):
            return False
        contains = rule.get("contains")
        if contains and contains.lower() not in text.lower():
            return False
        return True

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        self.decisions += 1
        if self.engine is not None:
            try:
                out = await self.engine({"tool": ctx.name, "user": ctx.user, "args": ctx.args})
            except Exception:
                out = {"allow": self.default == "allow", "reason": "pdp engine error"}
            if not out.get("allow", False):
                return PluginResult.block(f"unified_pdp: {out.get('reason', 'denied by policy')}",
                                          code="policy_denied")
            return PluginResult.ok()
        text = _text_of(ctx.args)
        for rule in self.rules:
            if self._matches(rule, ctx, text):
                if rule.get("effect", "deny") == "deny":
                    return PluginResult.block(
                        f"unified_pdp: denied by rule {rule.get('id', '?')}", code="policy_denied")
                return PluginResult.ok()
        if self.default == "deny":
            return PluginResult.block("unified_pdp: no rule matched (default deny)",
                                      code="policy_denied")
        return PluginResult.ok()

    agent_pre_invoke = tool_pre_invoke


INTEGRATION_PLUGINS = {
    p.name: p for p in (SummarizerPlugin, VirusTotalCheckerPlugin, VaultPlugin, UnifiedPdpPlugin)
}
