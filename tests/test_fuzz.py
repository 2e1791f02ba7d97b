"""JSON-RPC fuzz tier (reference analog: tests/fuzz/test_jsonrpc_fuzz.py):
malformed/adversarial inputs must produce protocol errors, never crashes,
through both the python engine and the C++ envelope scanner."""

import json
import random
import string

import numpy as np
import pytest

from mcp_context_forge_amd.protocol import jsonrpc


@pytest.fixture(scope="module")
def fuzz_raws():
    rng = random.Random(1234)
    raws = []
    printable = string.printable
    for i in range(300):
        kind = rng.randrange(8)
        if kind == 0:  # random bytes
            raws.append(bytes(rng.randrange(256) for _ in range(rng.randrange(0, 200))))
        elif kind == 1:  # random printable
            raws.append("".join(rng.choice(printable) for _ in range(rng.randrange(0, 200))).encode())
        elif kind == 2:  # truncated valid
            full = json.dumps({"jsonrpc": "2.0", "id": i, "method": "tools/call",
                               "params": {"name": "t", "arguments": {"a": "x" * 50}}}).encode()
            raws.append(full[: rng.randrange(len(full))])
        elif kind == 3:  # deep nesting
            d = rng.randrange(1, 120)
            raws.append((b'{"jsonrpc":"2.0","id":1,"method":"tools/call","params":{"name":"t","arguments":'
                         + b'{"a":' * d + b"1" + b"}" * d + b"}}"))
        elif kind == 4:  # wrong types everywhere
            raws.append(json.dumps({"jsonrpc": rng.choice([2.0, None, "2.0"]),
                                    "id": rng.choice([[], {}, True, 1.5]),
                                    "method": rng.choice([None, 5, "", "x"]),
                                    "params": rng.choice([1, "s", None, []])}).encode())
        elif kind == 5:  # unicode chaos (raw bytes incl. invalid utf-8 + lone surrogate escape)
            body = (b'{"jsonrpc":"2.0","id":1,"method":"tools/call","params":'
                    b'{"name":"t x","arguments":{"q":"' + (b"\xc3\xa9" * 40 if rng.random() < 0.8 else b"\xed\xa0\x80")
                    + b'"}}}')
            raws.append(body)
        elif kind == 6:  # giant numbers / strings in envelope
            raws.append(json.dumps({"jsonrpc": "2.0", "id": 10 ** rng.randrange(1, 30),
                                    "method": "x" * rng.randrange(1, 300)}).encode())
        else:  # valid but unknown methods
            raws.append(json.dumps({"jsonrpc": "2.0", "id": i,
                                    "method": rng.choice(["tools/dance", "rpc.x", "a/b/c"])}).encode())
    return raws


def test_parse_never_crashes(fuzz_raws):
    for raw in fuzz_raws:
        try:
            jsonrpc.parse_request_bytes(raw)
        except jsonrpc.JSONRPCError as exc:
            assert exc.code in (-32700, -32600)


def test_envelope_scanner_never_crashes(fuzz_raws):
    from mcp_context_forge_amd.ops import hip
    from mcp_context_forge_amd.ops.build import LIB

    if not LIB.exists():
        pytest.skip("lib not built")
    offs = np.zeros(len(fuzz_raws) + 1, dtype=np.int64)
    for i, r in enumerate(fuzz_raws):
        offs[i + 1] = offs[i] + len(r)
    blob = np.frombuffer(b"".join(fuzz_raws) or b"\0", dtype=np.uint8)
    env = hip.parse_envelopes(blob, offs)
    # spans must stay in bounds for every tools/call row
    for i, raw in enumerate(fuzz_raws):
        if env["kind"][i] == hip.ENV_TOOLS_CALL:
            lo, hi = offs[i], offs[i + 1]
            for b, e in (("name_beg", "name_end"), ("args_beg", "args_end"), ("id_beg", "id_end")):
                bb, ee = env[b][i], env[e][i]
                if bb >= 0:
                    assert lo <= bb <= ee <= hi


def test_engine_fuzz_responses(run, bare_engine, fuzz_raws):
    async def go():
        for raw in fuzz_raws[:150]:
            out = await bare_engine.handle_rpc_bytes(raw)
            if out is not None:
                obj = json.loads(out)
                assert obj["jsonrpc"] == "2.0"
                assert "error" in obj or "result" in obj

    run(go())


def test_schema_fuzz():
    from mcp_context_forge_amd.utils.jsonschema import validate

    rng = random.Random(5)

    def rand_val(depth=0):
        if depth > 3:
            return rng.choice([1, "x", True, None])
        return rng.choice([
            1, 1.5, "s", True, None, [rand_val(depth + 1) for _ in range(rng.randrange(3))],
            {f"k{i}": rand_val(depth + 1) for i in range(rng.randrange(3))},
        ])

    schemas = [
        {"type": "object", "properties": {"a": {"type": "integer", "minimum": 0}}, "required": ["a"]},
        {"anyOf": [{"type": "string"}, {"type": "number"}]},
        {"type": "array", "items": {"type": "object"}, "minItems": 1},
        {"oneOf": [{"type": "boolean"}, {"const": 5}]},
    ]
    for _ in range(400):
        validate(rand_val(), rng.choice(schemas))  # must not raise


def test_toon_fuzz_roundtrip():
    from mcp_context_forge_amd.plugins import toon

    rng = random.Random(9)

    def rand_scalar():
        return rng.choice([1, -2.5, "plain", "with space", "", True, False, None, "a,b", 'q"x'])

    def rand_val(depth=0):
        if depth > 2:
            return rand_scalar()
        k = rng.randrange(4)
        if k == 0:
            return rand_scalar()
        if k == 1:
            return [rand_scalar() for _ in range(rng.randrange(4))]
        if k == 2:
            return [{"a": rand_scalar(), "b": rand_scalar()} for _ in range(rng.randrange(1, 4))]
        return {f"k{i}": rand_val(depth + 1) for i in range(rng.randrange(4))}

    for _ in range(200):
        v = rand_val()
        assert toon.decode(toon.encode(v)) == v, v


# -- property-based fuzz for round-1 additions (hypothesis) ----------------

from hypothesis import given, settings as hsettings, strategies as st

_json_scalars = st.one_of(st.none(), st.booleans(), st.integers(-10**9, 10**9),
                          st.floats(allow_nan=False, allow_infinity=False, width=32),
                          st.text(max_size=40))
_json_values = st.recursive(
    _json_scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=5),
        st.dictionaries(st.text(min_size=1, max_size=12), children, max_size=5)),
    max_leaves=20)


@hsettings(max_examples=150, deadline=None, derandomize=True)
@given(_json_values)
def test_toon_roundtrip_property(value):
    """encode→decode is identity for JSON-shaped values (TOON's contract)."""
    from mcp_context_forge_amd.plugins import toon

    enc = toon.encode(value)
    out = toon.decode(enc)
    if isinstance(value, float):
        assert out == __import__("json").loads(__import__("json").dumps(value))
    else:
        assert out == value, (value, enc, out)


@hsettings(max_examples=200, deadline=None, derandomize=True)
@given(st.text(max_size=30))
def test_jsonpath_never_crashes_unexpectedly(expr):
    """Arbitrary expressions either evaluate or raise JSONPathError —
    never IndexError/KeyError/etc. (jsonpath_filter maps errors to None)."""
    from mcp_context_forge_amd.utils.jsonpath import JSONPathError, evaluate, jsonpath_filter

    doc = {"a": [1, {"b": "x"}], "c": {"d": None}}
    try:
        evaluate(doc, expr)
    except JSONPathError:
        pass
    jsonpath_filter(doc, expr)  # must never raise


@hsettings(max_examples=100, deadline=None, derandomize=True)
@given(st.binary(max_size=200))
def test_envelope_oracle_never_crashes(raw):
    """The Python JSON-RPC parser (CPU oracle of envelope.cpp) returns a
    clean protocol error for arbitrary bytes."""
    from mcp_context_forge_amd.protocol import jsonrpc

    try:
        jsonrpc.parse_request_bytes(raw)
    except jsonrpc.JSONRPCError:
        pass
