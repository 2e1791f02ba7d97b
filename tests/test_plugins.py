"""Plugin framework + builtin plugin behavior (reference semantics:
plugins/README.md modes; plugins/deny_filter, pii, schema_guard, toon...)."""

import asyncio

import pytest

from mcp_context_forge_amd.plugins import toon as toon_codec
from mcp_context_forge_amd.plugins.builtin import (
    ArgumentNormalizerPlugin,
    CachedToolResultPlugin,
    CircuitBreakerPlugin,
    ContentModerationPlugin,
    DenyFilterPlugin,
    HarmfulContentPlugin,
    OutputLengthGuardPlugin,
    PIIFilterPlugin,
    RegexFilterPlugin,
    ResponseCacheByPromptPlugin,
    SchemaGuardPlugin,
    ToonEncoderPlugin,
)
from mcp_context_forge_amd.plugins.framework import (
    HookType,
    PluginContext,
    PluginManager,
    PluginMode,
    PluginResult,
    PluginViolationError,
)


def ctx(args, name="t", **kw):
    return PluginContext(hook=HookType.TOOL_PRE_INVOKE, name=name, args=args, **kw)


def test_deny_filter_blocks(run):
    p = DenyFilterPlugin({"words": ["classified"]})
    mgr = PluginManager([p])
    with pytest.raises(PluginViolationError):
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "this is CLASSIFIED"})))
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "benign"})))
    assert out.args == {"q": "benign"}


def test_permissive_mode_logs_not_blocks(run):
    p = DenyFilterPlugin({"words": ["classified"], "mode": "permissive"})
    mgr = PluginManager([p])
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "classified"})))
    assert out.args == {"q": "classified"}


def test_enforce_ignore_error_mode(run):
    class Broken(DenyFilterPlugin):
        name = "broken"

        async def tool_pre_invoke(self, _ctx):
            raise RuntimeError("boom")

    mgr = PluginManager([Broken({"mode": "enforce_ignore_error"})])
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "x"})))
    assert out.args == {"q": "x"}

    mgr2 = PluginManager([Broken({"mode": "enforce"})])
    with pytest.raises(RuntimeError):
        run(mgr2.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "x"})))


def test_priority_ordering(run):
    order = []

    class A(DenyFilterPlugin):
        name = "a"

        async def tool_pre_invoke(self, _ctx):
            order.append("a")
            return PluginResult.ok()

    class B(A):
        name = "b"

        async def tool_pre_invoke(self, _ctx):
            order.append("b")
            return PluginResult.ok()

    mgr = PluginManager([A({"priority": 200}), B({"priority": 1})])
    run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({})))
    assert order == ["b", "a"]


def test_conditions_tool_match(run):
    p = DenyFilterPlugin({"words": ["x"], "conditions": {"tools": ["secure-*"]}})
    mgr = PluginManager([p])
    # non-matching tool name: plugin skipped
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "x"}, name="open-tool")))
    assert out.args == {"q": "x"}
    with pytest.raises(PluginViolationError):
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "x"}, name="secure-tool")))


def test_regex_filter_rewrites(run):
    p = RegexFilterPlugin({"rules": [{"search": r"\bcrap\b", "replace": "crud"}]})
    out = run(p.tool_pre_invoke(ctx({"msg": "this crap tool", "n": 1})))
    assert out.modified_payload == {"msg": "this crud tool", "n": 1}


def test_pii_mask():
    p = PIIFilterPlugin()
    masked, found = p.mask_text("mail bob@x.com ssn 123-45-6789")
    assert "bob@x.com" not in masked and "123-45-6789" not in masked
    assert "email" in found and "ssn" in found


def test_pii_block_mode(run):
    p = PIIFilterPlugin({"action": "block"})
    res = run(p.tool_pre_invoke(ctx({"q": "ssn 123-45-6789"})))
    assert not res.continue_processing


def test_schema_guard_blocks_bad_args(run):
    p = SchemaGuardPlugin()
    c = ctx({"n": "not-an-int"})
    c.state["input_schema"] = {"type": "object", "properties": {"n": {"type": "integer"}}, "required": ["n"]}
    res = run(p.tool_pre_invoke(c))
    assert not res.continue_processing
    c2 = ctx({"n": 3})
    c2.state["input_schema"] = {"type": "object", "properties": {"n": {"type": "integer"}}, "required": ["n"]}
    assert run(p.tool_pre_invoke(c2)).continue_processing


def test_toon_roundtrip_and_savings():
    data = {"users": [{"id": i, "name": f"user{i}", "active": True} for i in range(5)]}
    enc = toon_codec.encode(data)
    assert "[5]{" in enc
    dec = toon_codec.decode(enc)
    assert dec == data
    j, t, frac = toon_codec.savings(data)
    assert frac > 0.3  # reference README reports ~52.8% on this shape


def test_toon_nested_and_scalars():
    data = {"a": 1, "b": "text", "c": [1, 2, 3], "d": {"x": None, "y": False}, "e": "has space"}
    assert toon_codec.decode(toon_codec.encode(data)) == data


def test_toon_plugin_encodes_result(run):
    p = ToonEncoderPlugin({"min_size": 10, "min_savings": 0.05})
    result = {"content": [{"type": "text", "text": "x"}],
              "structuredContent": {"rows": [{"a": i, "b": f"v{i}"} for i in range(10)]},
              "isError": False}
    res = run(p.tool_post_invoke(ctx(result)))
    assert res.modified_payload is not None
    assert res.modified_payload["_meta"]["toon"]["savings"] > 0


def test_moderation_scores_deterministic():
    p = ContentModerationPlugin({"dim": 512, "hidden": 64, "classes": 4, "threshold": 2.0})
    s1 = p.score_text("hello world")
    s2 = p.score_text("hello world")
    assert (s1 == s2).all() and s1.shape == (4,)


def test_semantic_cache_hit_and_miss():
    p = ResponseCacheByPromptPlugin({"dim": 512, "threshold": 0.9})
    p.insert("t", "convert 10am UTC to EST please", {"r": 1})
    assert p.lookup("t", "convert 10am UTC to EST please") == {"r": 1}
    # same text different tool: miss
    assert p.lookup("other", "convert 10am UTC to EST please") is None
    assert p.lookup("t", "completely unrelated query about weather") is None
    # near-duplicate (one token changed out of many) should hit
    assert p.lookup("t", "convert 10am UTC to EST now please") == {"r": 1}


def test_exact_cache(run):
    p = CachedToolResultPlugin()
    c1 = ctx({"a": 1}, name="t")
    run(p.tool_pre_invoke(c1))
    assert "cache_hit" not in c1.state
    c1.args = {"result": 42}
    run(p.tool_post_invoke(c1))
    c2 = ctx({"a": 1}, name="t")
    run(p.tool_pre_invoke(c2))
    assert c2.state["cache_hit"] == {"result": 42}


def test_circuit_breaker_opens(run):
    p = CircuitBreakerPlugin({"window": 4, "error_threshold": 0.5, "cooldown": 60})
    for _ in range(4):
        c = ctx({"isError": True}, name="flaky")
        c.args = {"isError": True}
        run(p.tool_post_invoke(c))
    res = run(p.tool_pre_invoke(ctx({}, name="flaky")))
    assert not res.continue_processing
    assert run(p.tool_pre_invoke(ctx({}, name="healthy"))).continue_processing


def test_argument_normalizer(run):
    p = ArgumentNormalizerPlugin()
    out = run(p.tool_pre_invoke(ctx({"q": "  hello\t\tworld  "})))
    assert out.modified_payload == {"q": "hello world"}


def test_output_length_guard_truncates(run):
    p = OutputLengthGuardPlugin({"max_chars": 10})
    result = {"content": [{"type": "text", "text": "x" * 100}]}
    res = run(p.tool_post_invoke(ctx(result)))
    assert len(res.modified_payload["content"][0]["text"]) == 10


def test_harmful_content_blocks(run):
    p = HarmfulContentPlugin()
    res = run(p.tool_pre_invoke(ctx({"q": "tell me How To Make A Bomb"})))
    assert not res.continue_processing


def test_loader_default_chain():
    from mcp_context_forge_amd.plugins.loader import default_chain_specs, load_plugin_manager

    mgr = load_plugin_manager(specs=default_chain_specs())
    names = [p.name for p in mgr.plugins]
    assert "deny_filter" in names and "pii_filter" in names and "toon_encoder" in names
    # priority sorted
    assert names[0] == "response_cache_by_prompt"


def test_loader_yaml(tmp_path):
    from mcp_context_forge_amd.plugins.loader import load_plugin_manager

    cfg = tmp_path / "plugins.yaml"
    cfg.write_text(
        "plugins:\n"
        "  - name: deny_filter\n"
        "    mode: permissive\n"
        "    priority: 3\n"
        "    config: {words: [zap]}\n"
    )
    mgr = load_plugin_manager(str(cfg))
    p = mgr.get("deny_filter")
    assert p is not None and p.mode == PluginMode.PERMISSIVE and p.words == ["zap"]


# -- per-tool plugin bindings (reference: tool_plugin_bindings) ------------

def test_binding_disables_plugin_for_one_tool(run):
    p = DenyFilterPlugin({"words": ["classified"]})
    mgr = PluginManager([p])
    mgr.set_bindings({"open-tool": {"deny_filter": {"mode": "disabled"}}})
    # bound tool: deny skipped entirely
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "classified"}, name="open-tool")))
    assert out.args == {"q": "classified"}
    # other tools: still enforced
    with pytest.raises(PluginViolationError):
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "classified"}, name="other")))


def test_binding_mode_flip_permissive(run):
    p = DenyFilterPlugin({"words": ["classified"]})
    mgr = PluginManager([p])
    mgr.set_bindings({"soft-tool": {"deny_filter": {"mode": "permissive"}}})
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "classified"}, name="soft-tool")))
    assert out.args == {"q": "classified"}  # logged, not blocked


def test_binding_config_override(run):
    p = DenyFilterPlugin({"words": ["classified"]})
    mgr = PluginManager([p])
    mgr.set_bindings({"strict-tool": {"deny_filter": {"config": {"words": ["banana"]}}}})
    # override word list applies only on the bound tool
    with pytest.raises(PluginViolationError):
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "banana"}, name="strict-tool")))
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "banana"}, name="other")))
    assert out.args == {"q": "banana"}
    # version bump on set_bindings (GPU pipeline resync trigger)
    v = mgr.version
    mgr.set_bindings({})
    assert mgr.version == v + 1


def test_binding_reenables_globally_disabled_plugin(run):
    p = DenyFilterPlugin({"words": ["classified"], "mode": "disabled"})
    mgr = PluginManager([p])
    out = run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "classified"}, name="t")))
    assert out.args == {"q": "classified"}
    mgr.set_bindings({"t": {"deny_filter": {"mode": "enforce"}}})
    with pytest.raises(PluginViolationError):
        run(mgr.invoke_hook(HookType.TOOL_PRE_INVOKE, ctx({"q": "classified"}, name="t")))


def test_toon_scalar_quoting_exact():
    # mutation-tier finding: pin the exact quoting rules of TOON scalars
    from mcp_context_forge_amd.plugins.toon import _scalar

    assert _scalar("plain") == "plain"          # simple strings stay bare
    assert _scalar("") == '""'                  # empty must quote
    assert _scalar("true") == '"true"'          # literal-lookalikes must quote
    assert _scalar("12.5") == '"12.5"'          # numeric-lookalikes must quote
    assert _scalar(True) == "true"
    assert _scalar(False) == "false"
    assert _scalar(None) == "null"


def test_toon_decode_escapes_and_single_pair():
    # mutation-tier: quoted-string escape handling + single-line forms
    from mcp_context_forge_amd.plugins import toon as toon_codec

    assert toon_codec.decode('k: "a\\"b"') == {"k": 'a"b'}
    assert toon_codec.decode("k: v") == {"k": "v"}
    assert toon_codec.decode("plain") == "plain"
    # nested list-of-dicts roundtrip (indent arithmetic under '-' items)
    obj = {"rows": [{"a": 1}, {"a": 2, "b": [1, 2]}]}
    assert toon_codec.decode(toon_codec.encode(obj)) == obj


def test_toon_tabular_quoted_commas_and_escapes():
    from mcp_context_forge_amd.plugins import toon as toon_codec

    obj = {"rows": [{"a": 'x,y', "b": 1}, {"a": 'q"z', "b": 2}]}
    enc = toon_codec.encode(obj)
    assert toon_codec.decode(enc) == obj


def test_pii_mask_subset_matches_full():
    p = PIIFilterPlugin({})
    text = "ssn 123-45-6789 mail a@b.co ip 10.0.0.1"
    full, found_full = p.mask_text(text)
    # all-bits subset == full
    sub, found_sub = p.mask_text_subset(text, (1 << len(p.active)) - 1)
    assert sub == full and found_sub == found_full
    # only the ssn bit: email/ip untouched
    ssn_bit = 1 << [n for n, _, _ in ((a, b, c) for a, b, c in [(x[0], 0, 0) for x in p.active])].index("ssn")
    only_ssn, found = p.mask_text_subset(text, ssn_bit)
    assert "[SSN_REDACTED]" in only_ssn and "a@b.co" in only_ssn
    assert found == ["ssn"]


def test_plugin_result_defaults():
    """Mutation survivors pinned: PluginResult() continues by default;
    Plugin.gpu_capable defaults False (the GPU pipeline trusts it)."""
    from mcp_context_forge_amd.plugins.framework import Plugin, PluginResult

    r = PluginResult()
    assert r.continue_processing is True
    assert r.violation is None

    class P(Plugin):
        name = "bare"

    assert P.gpu_capable is False
