"""Adversarial GPU↔CPU parity fuzzing (VERDICT round-1 item 4).

The round-1 parity gate used well-formed compact payloads; these
generators TARGET the raw-bytes-vs-decoded-text containment argument
(gpu/pipeline.py module docstring): escape-hidden deny/harm words,
boundary-spanning fragments, non-ASCII, schema-key lookalikes embedded in
values, pathological schemas, arbitrary whitespace/key-order wire forms.

Tier 1 (CPU, always runs): generator sanity + the specific divergence the
fuzzer found in round 2 (escape-hidden deny words) asserted against the
CPU chain alone.
Tier 2 (@gpu): ≥1200 generated cases through BOTH runtimes, outcome-class
equality per request.
"""

import asyncio
import json
import random
import string

import pytest

DENY_WORDS = ["forbidden", "blocked_word"]
HARM_PHRASES = ["how to make a bomb"]  # default harmful_content phrases overlap


def _esc_variants(word: str, rng: random.Random):
    """Ways to hide `word` from a raw-byte scanner that decoded text reveals."""
    i = rng.randrange(1, len(word) - 1)

    def u(c):
        return "\\u%04x" % ord(c)

    yield word[:i] + u(word[i]) + word[i + 1:]                # one \u escape
    yield "".join(u(c) for c in word)                          # fully escaped
    yield word[:i] + u(word[i]) + u(word[i + 1]) + word[i + 2:] if i + 2 <= len(word) else word


def gen_adversarial(rng: random.Random, tool_names):
    """One adversarial tools/call request as RAW bytes (hand-assembled so
    formatting is NOT canonical: random whitespace, unsorted keys)."""
    kind = rng.randrange(18)
    name = rng.choice(tool_names)
    args = {}
    if kind == 12:   # result-shaping via echo passthrough: the args BECOME
        # the tool result (content key present) → drives the post chain
        # (toon encode, _meta handling, isError truthiness) on both paths
        name = "fast-time-echo"
        sc: dict = {"rows": [{"id": i, "v": f"r{i}"} for i in range(rng.randrange(2, 25))]}
        if rng.random() < 0.3:
            sc["mixed"] = [1, {"деep": "x"} if rng.random() < 0.2 else {"d": "x"}, [2]]
        args = {"content": [{"type": "text", "text": "orig " + "pad" * rng.randrange(0, 30)}],
                "structuredContent": sc}
        r = rng.random()
        if r < 0.25:
            args["_meta"] = "not-a-dict" if rng.random() < 0.5 else {"prior": 1}
        if r < 0.5:
            args["isError"] = rng.choice([True, False, 0, 1, "", "x", None])
        return _assemble(name, args, rng)
    if kind == 13:   # deep nesting crossing the C-lane depth envelope (32)
        d = rng.randrange(8, 45)
        inner: object = "leaf"
        for _ in range(d):
            inner = [inner] if rng.random() < 0.5 else {"n": inner}
        return _assemble(name, {"deep": inner}, rng)
    if kind == 14:   # surrogate pairs + astral chars in \u escapes
        payload = rng.choice(['"\\ud83d\\ude00 emoji pair"', '"text \\u00e9\\u4e2d end"',
                              '"mix \\ud83c\\udf89 and ascii"'])
        return _assemble(name, {}, rng, raw_args='{"msg":%s}' % payload)
    if kind == 15:   # number forms: big ints, -0, floats, exponents
        raw_args = rng.choice([
            '{"n":123456789012345678901234567890}', '{"n":-0}', '{"n":0}',
            '{"f":1.5,"g":-2.25}', '{"e":1e5,"t":2E-3}',
            '{"x":0.30000000000000004,"y":-17}',
        ])
        return _assemble(name, {}, rng, raw_args=raw_args)
    if kind == 16:   # multi-KB PII density (masking volume + correctness)
        frag = rng.choice(["a%d@ex%d.co " % (rng.randrange(99), rng.randrange(99)),
                           "123-45-6789 ", "4111 1111 1111 1111 ", "10.0.0.%d " % rng.randrange(255)])
        args = {"blob": frag * rng.randrange(40, 120), "tail": "x"}
        return _assemble(name, args, rng)
    if kind == 17:   # deny/harm words as KEYS and across sorted-adjacency
        w = rng.choice(DENY_WORDS + HARM_PHRASES)
        r = rng.random()
        if r < 0.4:
            return _assemble(name, {}, rng, raw_args=json.dumps({w: "value"}))
        if r < 0.7:
            cut = rng.randrange(1, len(w) - 1)
            return _assemble(name, {}, rng,
                             raw_args=json.dumps({"a": w[:cut], "b": w[cut:]}))
        return _assemble(name, {}, rng, raw_args=json.dumps({"k": w.upper()}))
    if kind == 0:    # escape-hidden deny word
        word = rng.choice(DENY_WORDS)
        hidden = rng.choice(list(_esc_variants(word, rng)))
        args = {"msg": f"text with {hidden} inside"}
        return _assemble(name, {"msg_raw": None}, rng, raw_args='{"msg":"text with %s inside"}' % hidden)
    if kind == 1:    # boundary-spanning fragments (must NOT block)
        word = rng.choice(DENY_WORDS)
        cut = rng.randrange(1, len(word) - 1)
        args = {"a": "x " + word[:cut], "b": word[cut:] + " y"}
    elif kind == 2:  # literal deny/harm (must block on both paths)
        args = {"msg": f"this is {rng.choice(DENY_WORDS + HARM_PHRASES)} content"}
    elif kind == 3:  # non-ASCII raw UTF-8 + unicode escapes
        args = {"msg": rng.choice(["café ☕ naïve", "\\u00e9l\\u00e8ve text", "日本語テキスト"]),
                "note": "ünïcode"}
    elif kind == 4:  # schema-key lookalike inside a value (escaped quotes)
        args = {"msg": 'fake \\"time\\": \\"2026\\" marker'}
        return _assemble(name, args, rng,
                         raw_args='{"msg":"fake \\"time\\": \\"2026\\" marker"}')
    elif kind == 5:  # PII, split and escaped
        args = {"msg": rng.choice(["mail a\\u0040b.com now", "ssn 123-45-6789", "a@b.co + c@d.io"])}
        return _assemble(name, args, rng,
                         raw_args=json.dumps(args).replace("\\\\u", "\\u"))
    elif kind == 6:  # whitespace-heavy values (normalizer)
        args = {"msg": "a  b\\tc", "x": "  lead", "y": "trail  "}
        return _assemble(name, args, rng, raw_args='{"msg":"a  b\\tc","x":"  lead","y":"trail  "}')
    elif kind == 7:  # nested / arrays (schema nest triggers)
        args = {"obj": {"time": "x", "deep": [1, {"k": "v"}]}, "arr": [1, 2, 3]}
    elif kind == 8:  # long strings + odd keys
        args = {"k" * rng.randrange(1, 30): "z" * rng.randrange(100, 2000),
                "under_score": rng.randrange(1000)}
    elif kind == 10:  # duplicate keys: banned word in the DISCARDED value
        word = rng.choice(DENY_WORDS + HARM_PHRASES)
        return _assemble(name, {}, rng,
                         raw_args='{"msg":"%s","msg":"benign wins"}' % word)
    elif kind == 11:  # duplicate keys: banned word in the KEPT value
        word = rng.choice(DENY_WORDS + HARM_PHRASES)
        return _assemble(name, {}, rng,
                         raw_args='{"msg":"benign loses","msg":"has %s here"}' % word)
    else:            # benign mixed types
        args = {"n": rng.randrange(100), "f": rng.random(), "b": rng.random() < 0.5,
                "s": "".join(rng.choice(string.printable[:80]) for _ in range(rng.randrange(1, 40)))}
    return _assemble(name, args, rng)


def _assemble(name, args, rng: random.Random, raw_args=None) -> bytes:
    """Wire form with random (valid) formatting: spaces around separators,
    insertion-order keys."""
    ws = lambda: rng.choice(["", " ", "  ", "\n"])  # noqa: E731
    if raw_args is None:
        parts = []
        for k, v in args.items():
            parts.append(json.dumps(k) + ws() + ":" + ws() + json.dumps(v, ensure_ascii=rng.random() < 0.5))
        raw_args = "{" + ("," + ws()).join(parts) + "}"
    rid = rng.randrange(1 << 20)
    return (('{"jsonrpc"' + ws() + ':' + ws() + '"2.0","id":' + str(rid)
             + ',"method"' + ws() + ':' + ws() + '"tools/call","params"' + ws() + ':'
             + ws() + '{"name":' + json.dumps(name) + ',"arguments"' + ws() + ':' + ws()
             + raw_args + "}}").encode("utf-8"))


async def build_pair(gpu_mod: bool, moderation: bool):
    from mcp_context_forge_amd.config import Settings
    from mcp_context_forge_amd.engine import GatewayEngine
    from mcp_context_forge_amd.plugins.loader import default_chain_specs, load_plugin_manager
    from mcp_context_forge_amd.services.upstream import NativeInProcUpstream, make_fake_time_upstream

    specs = [s for s in default_chain_specs()
             if moderation or s["name"] not in ("content_moderation",)]
    e = GatewayEngine(Settings(database_url="sqlite://", federation_enabled=False,
                               auth_required=False, gpu_enabled=gpu_mod,
                               gpu_semcache_capacity=1024),
                      plugin_manager=load_plugin_manager(specs=specs))
    await e.gateway_service.register_gateway(name="fast-time", url="inproc://t",
                                             client=make_fake_time_upstream())
    await e.gateway_service.register_gateway(name="native-time", url="inproc://n",
                                             client=NativeInProcUpstream())
    if gpu_mod:
        assert e.enable_gpu()
    return e


TOOLS = ["fast-time-echo", "fast-time-convert_time", "fast-time-get_system_time",
         "native-time-echo", "native-time-convert_time", "no-such-tool"]


def _outcome(resp):
    """Outcome class of a response: (kind, code-or-result). Wall-clock
    timestamps (get_system_time) are normalized — the CPU and GPU batches
    run ~a second apart and their 'current time' may differ."""
    import re as _re

    if resp is None:
        return ("none", None)
    o = json.loads(resp)
    if "error" in o:
        return ("error", o["error"]["code"])
    text = json.dumps(o["result"], sort_keys=True)
    text = _re.sub(r"\d{4}-\d{2}-\d{2}T\d{2}:\d{2}:\d{2}Z", "TS", text)
    return ("result", text)


# ---------------------------------------------------------------- tier 1 (CPU)


def test_generator_produces_all_categories():
    rng = random.Random(7)
    raws = [gen_adversarial(rng, TOOLS) for _ in range(300)]
    assert all(isinstance(r, bytes) for r in raws)
    # every raw form is valid JSON on the wire
    for r in raws:
        obj = json.loads(r)
        assert obj["method"] == "tools/call"
    # escape-hidden deny words are present in the corpus
    assert any(b"\\u0062" in r or b"\\u006f" in r or b"\\u0072" in r for r in raws)


def test_cpu_chain_blocks_escape_hidden_deny(run):
    """The decoded text contains the deny word even when the wire hides it
    behind \\u escapes — the CPU oracle must block (this is the behavior
    the GPU rewrite path's deny recheck mirrors)."""

    async def go():
        e = await build_pair(gpu_mod=False, moderation=False)
        raw = ('{"jsonrpc":"2.0","id":1,"method":"tools/call","params":'
               '{"name":"fast-time-echo","arguments":{"msg":"has forbi\\u0064den word"}}}').encode()
        out = (await e.process_rpc_batch([raw]))[0]
        o = json.loads(out)
        assert "error" in o and o["error"]["code"] == -32003, o
        await e.shutdown()

    run(go())


# ---------------------------------------------------------------- tier 2 (GPU)


@pytest.mark.gpu
@pytest.mark.parametrize("moderation,n_cases,seed", [
    (False, 900, 1234),   # arbitrary wire forms, full scan/rewrite/schema surface
    (True, 400, 99),      # canonical compact payloads incl. the MFMA classifier
    (False, 800, 4242),   # second draw over the widened generator set
])
def test_adversarial_parity(moderation, n_cases, seed):
    import torch

    if not torch.cuda.is_available():
        pytest.skip("no ROCm device")

    async def go():
        cpu = await build_pair(gpu_mod=False, moderation=moderation)
        gpu = await build_pair(gpu_mod=True, moderation=moderation)
        rng = random.Random(seed)
        if moderation:
            # canonical forms: the classifier featurizes identical bytes
            raws = []
            for _ in range(n_cases):
                obj = json.loads(gen_adversarial(rng, TOOLS))
                raws.append(json.dumps(obj, separators=(",", ":"), sort_keys=True,
                                       ensure_ascii=True).encode())
        else:
            raws = [gen_adversarial(rng, TOOLS) for _ in range(n_cases)]
        # several batch sizes: single, small, large
        divergences = []
        for lo, hi in [(0, 1), (1, 65), (65, n_cases)]:
            chunk = raws[lo:hi]
            if not chunk:
                continue
            c_out = await cpu.process_rpc_batch(list(chunk))
            g_out = await gpu.process_rpc_batch(list(chunk))
            for i, (c, g) in enumerate(zip(c_out, g_out)):
                co, go_ = _outcome(c), _outcome(g)
                if co != go_:
                    divergences.append((lo + i, chunk[i][:200], co, go_))
        assert not divergences, f"{len(divergences)} divergences; first 5: {divergences[:5]}"
        await cpu.shutdown()
        await gpu.shutdown()

    asyncio.run(go())


def test_cpu_duplicate_key_semantics(run):
    """json.loads keeps the LAST duplicate key — the CPU oracle dispatches
    when the banned word lives only in the discarded value, and blocks when
    it lives in the kept one. (The GPU path confirms raw deny/harm hits on
    decoded text via the host lane for exactly this reason.)"""

    async def go():
        e = await build_pair(gpu_mod=False, moderation=False)
        raw_discarded = ('{"jsonrpc":"2.0","id":1,"method":"tools/call","params":'
                         '{"name":"fast-time-echo","arguments":'
                         '{"msg":"forbidden","msg":"benign wins"}}}').encode()
        raw_kept = ('{"jsonrpc":"2.0","id":2,"method":"tools/call","params":'
                    '{"name":"fast-time-echo","arguments":'
                    '{"msg":"benign loses","msg":"has forbidden here"}}}').encode()
        o1, o2 = await e.process_rpc_batch([raw_discarded, raw_kept])
        assert "result" in json.loads(o1), o1      # discarded value → no block
        r2 = json.loads(o2)
        assert "error" in r2 and r2["error"]["code"] == -32003, o2
        await e.shutdown()

    run(go())
