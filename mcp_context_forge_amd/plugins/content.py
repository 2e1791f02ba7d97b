"""Content-transform plugin set (second tier of the reference inventory).

Reference analogs (behavioral ports): plugins/header_filter,
plugins/json_repair, plugins/markdown_cleaner, plugins/html_to_markdown,
plugins/safe_html_sanitizer, plugins/file_type_allowlist,
plugins/resource_filter, plugins/watchdog, plugins/webhook_notification,
plugins/code_formatter, plugins/ai_artifacts_normalizer,
plugins/privacy_notice_injector, plugins/timezone_translator,
plugins/license_header_injector, plugins/robots_license_guard.
"""

from __future__ import annotations

import asyncio
import html as html_mod
import html.parser
import json
import re
import time
from typing import Any, Dict, List, Optional, Tuple

from .builtin import _walk_strings
from .framework import HookType, Plugin, PluginContext, PluginResult


class HeaderFilterPlugin(Plugin):
    """Strip/allow outbound headers (reference: plugins/header_filter)."""

    name = "header_filter"
    hooks = (HookType.HTTP_PRE_REQUEST, HookType.TOOL_PRE_INVOKE)
    priority = 72

    def __init__(self, config=None):
        super().__init__(config)
        self.deny = {h.lower() for h in (self.config.get("deny") or ["cookie", "x-internal-secret"])}
        self.allow = {h.lower() for h in (self.config.get("allow") or [])}

    async def http_pre_request(self, ctx: PluginContext) -> PluginResult:
        for k in list(ctx.headers):
            kl = k.lower()
            if kl in self.deny or (self.allow and kl not in self.allow):
                del ctx.headers[k]
        return PluginResult.ok()

    tool_pre_invoke = http_pre_request


class JsonRepairPlugin(Plugin):
    """Repair near-JSON text results (reference: plugins/json_repair)."""

    name = "json_repair"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 880

    _trailing_comma = re.compile(r",\s*([}\]])")
    _single_quotes = re.compile(r"(?<=[{,\s])'([^']*)'(\s*:)")

    def repair(self, text: str) -> Optional[Any]:
        bases = [text]
        m = re.search(r"```(?:json)?\s*(.*?)```", text, re.DOTALL)
        if m:
            bases.insert(0, m.group(1))  # prefer fenced payloads
        for b in bases:
            t = self._trailing_comma.sub(r"\1", b)
            t = self._single_quotes.sub(r'"\1"\2', t)
            t = t.replace("True", "true").replace("False", "false").replace("None", "null")
            for c in (b, t):
                try:
                    return json.loads(c)
                except ValueError:
                    continue
        return None

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict) or result.get("structuredContent") is not None:
            return PluginResult.ok()
        for c in result.get("content", []):
            if isinstance(c, dict) and isinstance(c.get("text"), str):
                fixed = self.repair(c["text"])
                if fixed is not None and not isinstance(fixed, str):
                    new = dict(result)
                    new["structuredContent"] = fixed
                    return PluginResult.ok(new)
        return PluginResult.ok()


class MarkdownCleanerPlugin(Plugin):
    """Normalize markdown whitespace/artifacts (reference: plugins/markdown_cleaner)."""

    name = "markdown_cleaner"
    hooks = (HookType.TOOL_POST_INVOKE, HookType.PROMPT_POST_FETCH)
    priority = 870

    def clean(self, text: str) -> str:
        text = re.sub(r"\n{3,}", "\n\n", text)
        text = re.sub(r"[ \t]+$", "", text, flags=re.MULTILINE)
        text = re.sub(r"^(#{1,6})([^#\s])", r"\1 \2", text, flags=re.MULTILINE)
        return text.strip() + ("\n" if text.endswith("\n") else "")

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict):
            return PluginResult.ok()
        changed = False
        content = []
        for c in result.get("content", []):
            if isinstance(c, dict) and isinstance(c.get("text"), str):
                cleaned = self.clean(c["text"])
                changed = changed or cleaned != c["text"]
                content.append({**c, "text": cleaned})
            else:
                content.append(c)
        if changed:
            return PluginResult.ok({**result, "content": content})
        return PluginResult.ok()

    prompt_post_fetch = tool_post_invoke


class _HTMLToText(html.parser.HTMLParser):
    SKIP = {"script", "style"}

    def __init__(self):
        super().__init__()
        self.out: List[str] = []
        self._skip = 0
        self._href: Optional[str] = None

    def handle_starttag(self, tag, attrs):
        if tag in self.SKIP:
            self._skip += 1
        elif tag in ("p", "br", "div", "li", "tr"):
            self.out.append("\n")
        elif tag in ("h1", "h2", "h3", "h4", "h5", "h6"):
            self.out.append("\n" + "#" * int(tag[1]) + " ")
        elif tag == "a":
            self._href = dict(attrs).get("href")
        elif tag in ("strong", "b"):
            self.out.append("**")
        elif tag in ("em", "i"):
            self.out.append("*")
        elif tag == "code":
            self.out.append("`")

    def handle_endtag(self, tag):
        if tag in self.SKIP:
            self._skip = max(0, self._skip - 1)
        elif tag == "a" and self._href:
            self.out.append(f"]({self._href})")
            self._href = None
        elif tag in ("strong", "b"):
            self.out.append("**")
        elif tag in ("em", "i"):
            self.out.append("*")
        elif tag == "code":
            self.out.append("`")

    def handle_data(self, data):
        if not self._skip:
            if self._href is not None and (not self.out or not self.out[-1].endswith("[")):
                self.out.append("[")
            self.out.append(data)


class HtmlToMarkdownPlugin(Plugin):
    """HTML results → markdown (reference: plugins/html_to_markdown)."""

    name = "html_to_markdown"
    hooks = (HookType.TOOL_POST_INVOKE, HookType.RESOURCE_POST_FETCH)
    priority = 860

    def convert(self, text: str) -> str:
        p = _HTMLToText()
        p.feed(text)
        out = "".join(p.out)
        return re.sub(r"\n{3,}", "\n\n", out).strip()

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict):
            return PluginResult.ok()
        content = []
        changed = False
        for c in result.get("content", []):
            text = c.get("text") if isinstance(c, dict) else None
            if isinstance(text, str) and ("<html" in text.lower() or "<body" in text.lower() or
                                          re.search(r"<(p|div|h[1-6]|a)\b", text)):
                content.append({**c, "text": self.convert(text)})
                changed = True
            else:
                content.append(c)
        return PluginResult.ok({**result, "content": content}) if changed else PluginResult.ok()

    resource_post_fetch = tool_post_invoke


class SafeHtmlSanitizerPlugin(Plugin):
    """Escape/strip dangerous HTML (reference: plugins/safe_html_sanitizer)."""

    name = "safe_html_sanitizer"
    hooks = (HookType.TOOL_POST_INVOKE, HookType.RESOURCE_POST_FETCH)
    priority = 855

    _danger = re.compile(r"<\s*(script|iframe|object|embed|form)[^>]*>.*?<\s*/\s*\1\s*>|"
                         r"<\s*(script|iframe|object|embed|form)[^>]*/?>|on\w+\s*=\s*\"[^\"]*\"|"
                         r"javascript:", re.IGNORECASE | re.DOTALL)

    def sanitize(self, text: str) -> Tuple[str, bool]:
        new = self._danger.sub("", text)
        return new, new != text

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict):
            return PluginResult.ok()
        changed = False

        def fn(s: str) -> str:
            nonlocal changed
            out, ch = self.sanitize(s)
            changed = changed or ch
            return out

        new = _walk_strings(result, fn)
        return PluginResult.ok(new) if changed else PluginResult.ok()

    resource_post_fetch = tool_post_invoke


class FileTypeAllowlistPlugin(Plugin):
    """Restrict resource fetches by extension/MIME (reference: plugins/file_type_allowlist)."""

    name = "file_type_allowlist"
    hooks = (HookType.RESOURCE_PRE_FETCH,)
    priority = 45

    def __init__(self, config=None):
        super().__init__(config)
        self.allowed = set(self.config.get("extensions") or
                           [".txt", ".md", ".json", ".yaml", ".yml", ".csv", ".py", ".html"])

    async def resource_pre_fetch(self, ctx: PluginContext) -> PluginResult:
        uri = (ctx.args or {}).get("uri", "") if isinstance(ctx.args, dict) else str(ctx.args)
        dot = uri.rfind(".")
        if dot > 0 and "/" not in uri[dot:]:
            ext = uri[dot:].lower()
            if ext not in self.allowed:
                return PluginResult.block(f"file type {ext} not allowed", code="file_type")
        return PluginResult.ok()


class ResourceFilterPlugin(Plugin):
    """URI deny/allow patterns for resources (reference: plugins/resource_filter)."""

    name = "resource_filter"
    hooks = (HookType.RESOURCE_PRE_FETCH,)
    priority = 44

    def __init__(self, config=None):
        super().__init__(config)
        self.deny_patterns = [re.compile(p) for p in (self.config.get("deny") or [r"^file:///etc/", r"\.\."])]

    async def resource_pre_fetch(self, ctx: PluginContext) -> PluginResult:
        uri = (ctx.args or {}).get("uri", "") if isinstance(ctx.args, dict) else str(ctx.args)
        for p in self.deny_patterns:
            if p.search(uri):
                return PluginResult.block(f"resource uri blocked by pattern {p.pattern!r}", code="resource_filter")
        return PluginResult.ok()


class WatchdogPlugin(Plugin):
    """Flag slow tools (reference: plugins/watchdog)."""

    name = "watchdog"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE)
    priority = 7

    def __init__(self, config=None):
        super().__init__(config)
        self.max_ms = float(self.config.get("max_ms", 5000.0))
        self.slow: Dict[str, int] = {}

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        ctx.state["watchdog_t0"] = time.monotonic()
        return PluginResult.ok()

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        t0 = ctx.state.get("watchdog_t0")
        if t0 is not None:
            ms = (time.monotonic() - t0) * 1000.0
            if ms > self.max_ms:
                self.slow[ctx.name] = self.slow.get(ctx.name, 0) + 1
        return PluginResult.ok()


class WebhookNotificationPlugin(Plugin):
    """POST events to a webhook (reference: plugins/webhook_notification)."""

    name = "webhook_notification"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 990

    def __init__(self, config=None):
        super().__init__(config)
        self.url = self.config.get("url")
        self.events: List[dict] = []  # buffered when no URL (test/inspection mode)
        self._client = None

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        event = {"type": "tool_invoked", "tool": ctx.name, "user": ctx.user, "ts": time.time(),
                 "is_error": bool(isinstance(ctx.args, dict) and ctx.args.get("isError"))}
        if not self.url:
            self.events.append(event)
            if len(self.events) > 1000:
                self.events = self.events[-1000:]
            return PluginResult.ok()
        import httpx

        if self._client is None:
            self._client = httpx.AsyncClient(timeout=5.0)
        try:
            await self._client.post(self.url, json=event)
        except httpx.HTTPError:
            pass  # fire-and-forget (reference mode: fire_and_forget)
        return PluginResult.ok()

    async def shutdown(self) -> None:
        if self._client is not None:
            await self._client.aclose()


class CodeFormatterPlugin(Plugin):
    """Light code formatting of text results (reference: plugins/code_formatter)."""

    name = "code_formatter"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 865

    def format_text(self, text: str) -> str:
        text = text.replace("\t", "    ")
        text = re.sub(r"[ \t]+$", "", text, flags=re.MULTILINE)
        if text and not text.endswith("\n"):
            text += "\n"
        return text

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict):
            return PluginResult.ok()
        changed = False
        content = []
        for c in result.get("content", []):
            if isinstance(c, dict) and isinstance(c.get("text"), str) and "```" in c["text"]:
                f = self.format_text(c["text"])
                changed = changed or f != c["text"]
                content.append({**c, "text": f})
            else:
                content.append(c)
        return PluginResult.ok({**result, "content": content}) if changed else PluginResult.ok()


class AiArtifactsNormalizerPlugin(Plugin):
    """Strip LLM artifacts (reference: plugins/ai_artifacts_normalizer)."""

    name = "ai_artifacts_normalizer"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 868

    _artifacts = re.compile(
        r"^\s*(As an AI( language)? model,?\s*|I'm sorry, but\s*|Sure! Here('s| is)\s*)", re.IGNORECASE)

    def normalize(self, text: str) -> str:
        text = self._artifacts.sub("", text)
        text = text.replace("​", "").replace("﻿", "")
        return text

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict):
            return PluginResult.ok()
        new = _walk_strings(result, self.normalize)
        return PluginResult.ok(new if new != result else None)


class PrivacyNoticeInjectorPlugin(Plugin):
    """Append a privacy notice to results (reference: plugins/privacy_notice_injector)."""

    name = "privacy_notice_injector"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 960

    def __init__(self, config=None):
        super().__init__(config)
        self.notice = self.config.get("notice", "This response may contain processed personal data.")

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict):
            return PluginResult.ok()
        new = dict(result)
        new["content"] = list(result.get("content", [])) + [{"type": "text", "text": f"\n---\n{self.notice}"}]
        return PluginResult.ok(new)


class TimezoneTranslatorPlugin(Plugin):
    """Annotate ISO timestamps with a target timezone offset
    (reference: plugins/timezone_translator)."""

    name = "timezone_translator"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 875

    _iso = re.compile(r"\b(\d{4}-\d{2}-\d{2}T\d{2}:\d{2}:\d{2})Z\b")

    def __init__(self, config=None):
        super().__init__(config)
        self.offset_hours = int(self.config.get("offset_hours", 0))

    def translate(self, text: str) -> str:
        if self.offset_hours == 0:
            return text
        import datetime

        def sub(m):
            dt = datetime.datetime.fromisoformat(m.group(1)) + datetime.timedelta(hours=self.offset_hours)
            sign = "+" if self.offset_hours >= 0 else "-"
            return f"{dt.isoformat()}{sign}{abs(self.offset_hours):02d}:00"

        return self._iso.sub(sub, text)

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict) or self.offset_hours == 0:
            return PluginResult.ok()
        return PluginResult.ok(_walk_strings(result, self.translate))


class LicenseHeaderInjectorPlugin(Plugin):
    """Prepend license headers to code blocks (reference: plugins/license_header_injector)."""

    name = "license_header_injector"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 940

    def __init__(self, config=None):
        super().__init__(config)
        self.header = self.config.get("header", "# SPDX-License-Identifier: Apache-2.0")

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict):
            return PluginResult.ok()
        changed = False
        content = []
        for c in result.get("content", []):
            text = c.get("text") if isinstance(c, dict) else None
            if isinstance(text, str) and text.startswith("```") and self.header not in text:
                lines = text.split("\n")
                lines.insert(1, self.header)
                content.append({**c, "text": "\n".join(lines)})
                changed = True
            else:
                content.append(c)
        return PluginResult.ok({**result, "content": content}) if changed else PluginResult.ok()


class RobotsLicenseGuardPlugin(Plugin):
    """Respect robots/noai markers in fetched resources
    (reference: plugins/robots_license_guard)."""

    name = "robots_license_guard"
    hooks = (HookType.RESOURCE_POST_FETCH,)
    priority = 850

    _markers = ("noai", "noimageai", "X-Robots-Tag: noindex")

    async def resource_post_fetch(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        text = json.dumps(result, default=str) if not isinstance(result, str) else result
        for m in self._markers:
            if m in text:
                return PluginResult.block(f"resource carries {m!r} marker", code="robots_license")
        return PluginResult.ok()


EXTRA_PLUGINS = {
    p.name: p
    for p in (
        HeaderFilterPlugin, JsonRepairPlugin, MarkdownCleanerPlugin, HtmlToMarkdownPlugin,
        SafeHtmlSanitizerPlugin, FileTypeAllowlistPlugin, ResourceFilterPlugin, WatchdogPlugin,
        WebhookNotificationPlugin, CodeFormatterPlugin, AiArtifactsNormalizerPlugin,
        PrivacyNoticeInjectorPlugin, TimezoneTranslatorPlugin, LicenseHeaderInjectorPlugin,
        RobotsLicenseGuardPlugin,
    )
}
