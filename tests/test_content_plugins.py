"""Second-tier content plugins (reference analogs: plugins/{json_repair,
markdown_cleaner, html_to_markdown, safe_html_sanitizer, file_type_allowlist,
resource_filter, watchdog, webhook_notification, ...})."""

import pytest

from mcp_context_forge_amd.plugins.content import (
    AiArtifactsNormalizerPlugin,
    CodeFormatterPlugin,
    FileTypeAllowlistPlugin,
    HeaderFilterPlugin,
    HtmlToMarkdownPlugin,
    JsonRepairPlugin,
    LicenseHeaderInjectorPlugin,
    MarkdownCleanerPlugin,
    PrivacyNoticeInjectorPlugin,
    ResourceFilterPlugin,
    RobotsLicenseGuardPlugin,
    SafeHtmlSanitizerPlugin,
    TimezoneTranslatorPlugin,
    WatchdogPlugin,
    WebhookNotificationPlugin,
)
from mcp_context_forge_amd.plugins.framework import (HookType, PluginContext, PluginManager,
                                                     PluginViolationError)


def ctx(args, name="t", **kw):
    return PluginContext(hook=HookType.TOOL_POST_INVOKE, name=name, args=args, **kw)


def result(text):
    return {"content": [{"type": "text", "text": text}], "isError": False}


def test_header_filter(run):
    p = HeaderFilterPlugin({"deny": ["cookie"]})
    c = ctx({})
    c.headers = {"Cookie": "secret", "X-Ok": "1"}
    run(p.http_pre_request(c))
    assert "Cookie" not in c.headers and "X-Ok" in c.headers


def test_json_repair(run):
    p = JsonRepairPlugin()
    res = run(p.tool_post_invoke(ctx(result("```json\n{\"a\": 1,}\n```"))))
    assert res.modified_payload["structuredContent"] == {"a": 1}
    res = run(p.tool_post_invoke(ctx(result("{'key': True, 'x': None}"))))
    assert res.modified_payload["structuredContent"] == {"key": True, "x": None}
    res = run(p.tool_post_invoke(ctx(result("not json at all"))))
    assert res.modified_payload is None


def test_markdown_cleaner(run):
    p = MarkdownCleanerPlugin()
    res = run(p.tool_post_invoke(ctx(result("#Title\n\n\n\n\ntext   \nmore"))))
    assert res.modified_payload["content"][0]["text"] == "# Title\n\ntext\nmore"


def test_html_to_markdown(run):
    p = HtmlToMarkdownPlugin()
    res = run(p.tool_post_invoke(ctx(result(
        "<html><body><h1>Hi</h1><p>Some <strong>bold</strong> and <a href='http://x'>link</a></p>"
        "<script>evil()</script></body></html>"))))
    text = res.modified_payload["content"][0]["text"]
    assert "# Hi" in text and "**bold**" in text and "](http://x)" in text and "evil" not in text


def test_safe_html_sanitizer(run):
    p = SafeHtmlSanitizerPlugin()
    res = run(p.tool_post_invoke(ctx(result('<div onclick="evil()">x</div><script>bad()</script>'))))
    text = res.modified_payload["content"][0]["text"]
    assert "script" not in text and "onclick" not in text


def test_file_type_allowlist(run):
    p = FileTypeAllowlistPlugin()
    c = PluginContext(hook=HookType.RESOURCE_PRE_FETCH, name="r", args={"uri": "file://x/run.exe"})
    assert not run(p.resource_pre_fetch(c)).continue_processing
    c2 = PluginContext(hook=HookType.RESOURCE_PRE_FETCH, name="r", args={"uri": "file://x/notes.md"})
    assert run(p.resource_pre_fetch(c2)).continue_processing


def test_resource_filter(run):
    p = ResourceFilterPlugin()
    c = PluginContext(hook=HookType.RESOURCE_PRE_FETCH, name="r", args={"uri": "file:///etc/passwd"})
    assert not run(p.resource_pre_fetch(c)).continue_processing


def test_watchdog(run):
    p = WatchdogPlugin({"max_ms": 0.0})
    c = ctx({}, name="slow")
    run(p.tool_pre_invoke(c))
    run(p.tool_post_invoke(c))
    assert p.slow.get("slow", 0) >= 1


def test_webhook_buffer(run):
    p = WebhookNotificationPlugin()
    run(p.tool_post_invoke(ctx(result("x"), name="t1")))
    assert p.events and p.events[0]["tool"] == "t1"


def test_code_formatter(run):
    p = CodeFormatterPlugin()
    res = run(p.tool_post_invoke(ctx(result("```py\n\tx = 1   \n```"))))
    assert res.modified_payload["content"][0]["text"] == "```py\n    x = 1\n```\n"


def test_ai_artifacts(run):
    p = AiArtifactsNormalizerPlugin()
    res = run(p.tool_post_invoke(ctx(result("As an AI language model, I think yes"))))
    assert res.modified_payload["content"][0]["text"] == "I think yes"


def test_privacy_notice(run):
    p = PrivacyNoticeInjectorPlugin({"notice": "NOTICE"})
    res = run(p.tool_post_invoke(ctx(result("data"))))
    assert "NOTICE" in res.modified_payload["content"][-1]["text"]


def test_timezone_translator(run):
    p = TimezoneTranslatorPlugin({"offset_hours": 2})
    res = run(p.tool_post_invoke(ctx(result("at 2026-01-01T10:00:00Z sharp"))))
    assert "2026-01-01T12:00:00+02:00" in res.modified_payload["content"][0]["text"]


def test_license_header(run):
    p = LicenseHeaderInjectorPlugin()
    res = run(p.tool_post_invoke(ctx(result("```python\nprint(1)\n```"))))
    assert "SPDX-License-Identifier" in res.modified_payload["content"][0]["text"]


def test_robots_guard(run):
    p = RobotsLicenseGuardPlugin()
    c = PluginContext(hook=HookType.RESOURCE_POST_FETCH, name="r",
                      args={"contents": [{"text": "<meta name=robots content=noai>"}]})
    assert not run(p.resource_post_fetch(c)).continue_processing


def test_loader_resolves_extras():
    from mcp_context_forge_amd.plugins.loader import build_plugin

    p = build_plugin({"name": "json_repair"})
    assert p.name == "json_repair"


# -- integration plugins (reference: summarizer / virus_total / vault / unified_pdp)

def test_virus_total_blocks_malicious_url(run):
    from mcp_context_forge_amd.plugins.integrations import VirusTotalCheckerPlugin

    async def fake_client(kind, value):
        return {"malicious": 5 if "evil" in value else 0}

    p = VirusTotalCheckerPlugin({"client": fake_client})
    mgr = PluginManager([p])
    with pytest.raises(PluginViolationError):
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE,
                            ctx({"url": "visit http://evil.example/payload now"})))
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"url": "see http://good.example/x"})))
    assert out.args["url"].startswith("see")
    # verdicts cached: same URL → no second client call
    n = p.checked
    run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"url": "see http://good.example/x"})))
    assert p.checked == n


def test_vault_injects_token_without_clobbering(run):
    from mcp_context_forge_amd.plugins.integrations import VaultPlugin

    p = VaultPlugin({"rules": [{"match": "github-*", "header": "Authorization", "token": "tok123"}]})
    c = ctx({"q": 1}, name="github-search")
    run(PluginManager([p]).invoke_hook(HookType.TOOL_PRE_INVOKE, c))
    assert c.headers["Authorization"] == "Bearer tok123"
    # caller-supplied credentials are never replaced
    c2 = ctx({"q": 1}, name="github-search")
    c2.headers["Authorization"] = "Bearer mine"
    run(PluginManager([p]).invoke_hook(HookType.TOOL_PRE_INVOKE, c2))
    assert c2.headers["Authorization"] == "Bearer mine"
    # non-matching tool untouched
    c3 = ctx({}, name="other")
    run(PluginManager([p]).invoke_hook(HookType.TOOL_PRE_INVOKE, c3))
    assert "Authorization" not in c3.headers


def test_unified_pdp_rules_and_default(run):
    from mcp_context_forge_amd.plugins.integrations import UnifiedPdpPlugin

    p = UnifiedPdpPlugin({"rules": [
        {"id": "no-prod-writes", "effect": "deny", "tools": ["prod-*"], "contains": "delete"},
        {"id": "allow-admin", "effect": "allow", "users": ["admin"]},
    ], "default": "deny"})
    mgr = PluginManager([p])
    with pytest.raises(PluginViolationError):
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE,
                            ctx({"op": "delete row 5"}, name="prod-db")))
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": 1}, name="x", user="admin")))
    assert out is not None
    with pytest.raises(PluginViolationError):  # default deny for unmatched
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": 1}, name="x", user="guest")))


def test_unified_pdp_external_engine(run):
    from mcp_context_forge_amd.plugins.integrations import UnifiedPdpPlugin

    async def engine(inp):
        return {"allow": inp["tool"] != "blocked-tool", "reason": "policy says no"}

    p = UnifiedPdpPlugin({"engine": engine})
    mgr = PluginManager([p])
    with pytest.raises(PluginViolationError):
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({}, name="blocked-tool")))
    run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({}, name="fine-tool")))


def test_summarizer_uses_llm(run):
    from mcp_context_forge_amd.plugins.integrations import SummarizerPlugin

    class FakeLLM:
        async def chat_completions(self, body):
            return {"choices": [{"message": {"content": "short summary"}}]}

    p = SummarizerPlugin({"llm": FakeLLM(), "threshold_chars": 100})
    c = PluginContext(hook=HookType.TOOL_POST_INVOKE, name="t",
                      args={"content": [{"type": "text", "text": "x" * 500}]})
    out = run(PluginManager([p]).invoke_hook(HookType.TOOL_POST_INVOKE, c))
    assert out.args["content"][0]["text"] == "short summary"
    assert p.summarized == 1
    # below threshold: untouched
    c2 = PluginContext(hook=HookType.TOOL_POST_INVOKE, name="t",
                       args={"content": [{"type": "text", "text": "tiny"}]})
    out2 = run(PluginManager([p]).invoke_hook(HookType.TOOL_POST_INVOKE, c2))
    assert out2.args["content"][0]["text"] == "tiny"


# -- prompt template security (reference: prompt_service Jinja sandbox) ----

def test_prompt_template_sandbox_blocks_escapes(run):
    """Template injection attempts must not reach Python internals."""
    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine

    async def go():
        e = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                                   auth_required=False, plugins_enabled=False))
        e.registry.create("prompt", name="leaky",
                          template="{{ args.__class__.__mro__ }}")
        try:
            await e.prompt_service.get_prompt("leaky", {"args": "x"})
            leaked = True
        except Exception:
            leaked = False
        assert not leaked, "sandbox allowed dunder access"
        # benign templates still render with strict-undefined semantics
        e.registry.create("prompt", name="greet", template="hello {{ name }}")
        out = await e.prompt_service.get_prompt("greet", {"name": "ada"})
        text = str(out)
        assert "hello ada" in text
        try:
            await e.prompt_service.get_prompt("greet", {})
            missing_ok = True
        except Exception:
            missing_ok = False
        assert not missing_ok, "StrictUndefined should reject missing args"
        await e.shutdown()

    run(go())
