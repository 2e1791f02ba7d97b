"""C++ host fast-path functions (envelope scanner, native upstream) — these
run on CPU (no GPU needed): the .so's host entry points."""

import json

import numpy as np
import pytest


@pytest.fixture(scope="module")
def hip():
    from mcp_context_forge_amd.ops import hip as _hip
    from mcp_context_forge_amd.ops.build import LIB

    if not LIB.exists():
        pytest.skip("libforge_hip.so not built")
    _hip._load()
    return _hip


def _env(hip, raws):
    offs = np.zeros(len(raws) + 1, dtype=np.int64)
    for i, r in enumerate(raws):
        offs[i + 1] = offs[i] + len(r)
    blob = np.frombuffer(b"".join(raws) or b"\0", dtype=np.uint8)
    return blob, hip.parse_envelopes(blob, offs)


def test_envelope_kinds(hip):
    raws = [
        b'{"jsonrpc":"2.0","id":1,"method":"tools/call","params":{"name":"t","arguments":{"a":1}}}',
        b'{"jsonrpc":"2.0","id":"s","method":"ping"}',
        b'{broken',
        b'{"jsonrpc":"2.0","method":"tools/call","params":{"name":"t","arguments":{}}}',
        b'{"jsonrpc":"2.0","id":5,"method":"tools/call","params":{"arguments":{}}}',
        b'{"jsonrpc":"1.0","id":6,"method":"x"}',
        b'{"jsonrpc":"2.0","id":7,"method":"tools/call","params":{"name":"t"}}',
        b'  {"jsonrpc":"2.0", "id": 8, "method": "tools/call", "params": {"name": "t", "arguments": {"k": "v"}}}  ',
    ]
    blob, env = _env(hip, raws)
    k = env["kind"]
    assert k[0] == hip.ENV_TOOLS_CALL
    assert k[1] == hip.ENV_OTHER
    assert k[2] == hip.ENV_PARSE_ERR
    assert k[3] == hip.ENV_TOOLS_CALL and env["id_beg"][3] == -1  # notification
    assert k[4] == hip.ENV_NEEDS_PY          # missing name → python error path
    assert k[5] == hip.ENV_NEEDS_PY          # wrong version
    assert k[6] == hip.ENV_TOOLS_CALL and env["args_beg"][6] == -1  # absent args
    # whitespace-tolerant variant
    assert k[7] == hip.ENV_TOOLS_CALL
    nb, ne = env["name_beg"][7], env["name_end"][7]
    assert blob[nb:ne].tobytes() == b"t"
    ab, ae = env["args_beg"][7], env["args_end"][7]
    assert json.loads(blob[ab:ae].tobytes()) == {"k": "v"}


def test_envelope_spans_exact(hip):
    raw = b'{"jsonrpc":"2.0","id":42,"method":"tools/call","params":{"name":"x-y","arguments":{"nested":{"deep":[1,2]}}}}'
    blob, env = _env(hip, [raw])
    assert blob[env["id_beg"][0]:env["id_end"][0]].tobytes() == b"42"
    assert blob[env["name_beg"][0]:env["name_end"][0]].tobytes() == b"x-y"
    args = blob[env["args_beg"][0]:env["args_end"][0]].tobytes()
    assert json.loads(args) == {"nested": {"deep": [1, 2]}}


def test_envelope_string_ids_and_escapes(hip):
    raws = [
        b'{"jsonrpc":"2.0","id":"a\\"b","method":"tools/call","params":{"name":"t","arguments":{}}}',
        b'{"jsonrpc":"2.0","id":3,"method":"tools/call","params":{"name":"t","arguments":{"s":"brace } in string"}}}',
    ]
    blob, env = _env(hip, raws)
    assert env["kind"][0] == hip.ENV_TOOLS_CALL
    assert blob[env["id_beg"][0]:env["id_end"][0]].tobytes() == b'"a\\"b"'
    assert env["kind"][1] == hip.ENV_TOOLS_CALL
    args = blob[env["args_beg"][1]:env["args_end"][1]].tobytes()
    assert json.loads(args) == {"s": "brace } in string"}


def test_envelope_agrees_with_python_validator(hip):
    """Fuzz-ish agreement: every kind!=TOOLS_CALL raw must be handled by the
    Python path without crashing, and kind==TOOLS_CALL spans must parse."""
    import random

    from mcp_context_forge_amd.protocol import jsonrpc

    rng = random.Random(11)
    raws = []
    for i in range(200):
        shape = rng.randrange(5)
        if shape == 0:
            raws.append(json.dumps({"jsonrpc": "2.0", "id": i, "method": "tools/call",
                                    "params": {"name": f"t{i%7}", "arguments": {"v": i, "s": "x" * rng.randrange(30)}}},
                                   separators=(",", ":")).encode())
        elif shape == 1:
            raws.append(json.dumps({"jsonrpc": "2.0", "id": i, "method": rng.choice(["ping", "tools/list"])}).encode())
        elif shape == 2:
            raws.append(b'{"jsonrpc":"2.0","id":%d' % i)  # truncated
        elif shape == 3:
            raws.append(json.dumps({"jsonrpc": "2.0", "id": {"bad": 1}, "method": "x"}).encode())
        else:
            raws.append(json.dumps({"jsonrpc": "2.0", "id": i, "method": "tools/call",
                                    "params": {"name": "t", "arguments": {"a": [1, {"b": "c"}]}}}).encode())
    blob, env = _env(hip, raws)
    for i, raw in enumerate(raws):
        k = env["kind"][i]
        if k == hip.ENV_TOOLS_CALL:
            obj = json.loads(raw)
            assert obj["method"] == "tools/call"
            ab = env["args_beg"][i]
            if ab >= 0:
                assert json.loads(blob[ab:env["args_end"][i]].tobytes()) == obj["params"]["arguments"]
        elif k == hip.ENV_PARSE_ERR:
            try:
                jsonrpc.parse_request_bytes(raw)
            except jsonrpc.JSONRPCError:
                pass  # python also rejects — consistent


def test_native_upstream_matches_python_semantics(hip, run):
    from mcp_context_forge_amd.services.upstream import NativeInProcUpstream

    up = NativeInProcUpstream()
    res = run(up.call_tool("convert_time", {"time": "2026-05-05T05:05:05Z",
                                            "source_timezone": "UTC", "target_timezone": "Asia/Tokyo"}))
    sc = res["structuredContent"]
    assert sc["time"] == "2026-05-05T05:05:05Z" and sc["converted"] is True
    assert res["isError"] is False

    res = run(up.call_tool("echo", {"a": {"b": [1, 2, "x"]}, "q": 'quote " inside'}))
    assert res["structuredContent"] == {"a": {"b": [1, 2, "x"]}, "q": 'quote " inside'}
    assert json.loads(res["content"][0]["text"]) == res["structuredContent"]

    res = run(up.call_tool("get_system_time", {"timezone": "Europe/Paris"}))
    assert res["structuredContent"]["timezone"] == "Europe/Paris"

    tools = run(up.list_tools())
    assert {t["name"] for t in tools} == {"convert_time", "get_system_time", "echo"}


def test_native_upstream_batch(hip):
    args = [b'{"time":"t1","source_timezone":"a","target_timezone":"b"}', b'{"x":1}', b"{}"]
    offs = np.zeros(4, dtype=np.int64)
    for i, a in enumerate(args):
        offs[i + 1] = offs[i] + len(a)
    blob = np.frombuffer(b"".join(args), dtype=np.uint8)
    ab = offs[:-1].astype(np.int32)
    ae = offs[1:].astype(np.int32)
    kinds = np.array([0, 2, 1], dtype=np.int32)
    out, rb, re_ = hip.upstream_call_batch(blob, ab, ae, kinds, "2026-09-12T00:00:00Z")
    r0 = json.loads(out[rb[0]:re_[0]].tobytes())
    assert r0["structuredContent"]["time"] == "t1"
    r1 = json.loads(out[rb[1]:re_[1]].tobytes())
    assert r1["structuredContent"] == {"x": 1}
    r2 = json.loads(out[rb[2]:re_[2]].tobytes())
    assert r2["structuredContent"]["time"] == "2026-09-12T00:00:00Z"
