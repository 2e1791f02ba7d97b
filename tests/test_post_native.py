"""Native result post chain (ops/csrc/rewrite.cpp forge_post_rows)
cross-validated against the Python implementations it replaces: the
_host_post order pii_filter → harmful_content (sorted-dump substring) →
toon_encoder (structuredContent compression + _meta.toon record) →
wire-order serialization. Same function-level parity pattern as
tests/test_rewrite_native.py; the GPU parity fuzzer covers end-to-end."""

import json
import random
import string

import numpy as np

from mcp_context_forge_amd.ops import hip
from mcp_context_forge_amd.plugins import toon as toon_codec
from mcp_context_forge_amd.plugins.builtin import (PII_PATTERNS, PIIFilterPlugin,
                                                   ToonEncoderPlugin, _walk_strings)

PII = PIIFilterPlugin({})
ALL_MASK = (1 << len(PII_PATTERNS)) - 1
FIXED_NAMES = [n for n, _, _ in PII_PATTERNS]

HARM = ["how to make a bomb", "attack plan"]
HARM_BLOB = np.frombuffer(b"".join(p.encode() for p in HARM), dtype=np.uint8).copy()
HARM_OFF = np.zeros(len(HARM) + 1, dtype=np.int32)
for _i, _p in enumerate(HARM):
    HARM_OFF[_i + 1] = HARM_OFF[_i] + len(_p)


def run_c(raws, fl=7, pii_mode=0, toon_min_size=256, toon_min_savings=0.1):
    blob = b"".join(raws)
    beg, end, off = [], [], 0
    for r in raws:
        beg.append(off)
        off += len(r)
        end.append(off)
    return hip.post_rows(
        np.frombuffer(blob, dtype=np.uint8).copy() if blob else np.zeros(1, dtype=np.uint8),
        np.asarray(beg, dtype=np.int64), np.asarray(end, dtype=np.int64),
        np.asarray([fl] * len(raws), dtype=np.uint8), ALL_MASK, pii_mode,
        harm_blob=HARM_BLOB, harm_off=HARM_OFF,
        toon_min_size=toon_min_size, toon_min_savings=toon_min_savings)


def py_reference(raw, fl=7, pii_block=False, toon_min_size=256, toon_min_savings=0.1):
    """The _host_post chain replicated from plugins (gpu/pipeline.py
    _host_post): returns (final_bytes, blocked_msg_kind, is_err)."""
    result = json.loads(raw)
    if fl & 1:
        found = []

        def fn(s):
            masked, f = PII.mask_text(s)
            found.extend(f)
            return masked

        new = _walk_strings(result, fn)
        if found and pii_block:
            return None, ("pii", sorted(set(found))), True
        if found and not pii_block:
            result = new
    if fl & 2:
        hay = json.dumps(result, separators=(",", ":"), sort_keys=True, default=str).lower()
        for w, p in enumerate(HARM):
            if p in hay:
                return None, ("harm", w), True
    if fl & 4:
        plug = ToonEncoderPlugin({"min_size": toon_min_size, "min_savings": toon_min_savings})
        new = plug.encode_result(result)
        if new is not None:
            result = new
    is_err = bool(isinstance(result, dict) and result.get("isError"))
    return json.dumps(result, separators=(",", ":")).encode(), None, is_err


def check(raws, **kw):
    pii_mode = 1 if kw.pop("pii_block", False) else 0
    st, found, harm_hit, is_err, arena, ob, oe = run_c(raws, pii_mode=pii_mode, **kw)
    n_done = 0
    for i, raw in enumerate(raws):
        want_b, blocked, want_err = py_reference(raw, pii_block=(pii_mode == 1), **kw)
        if blocked is None:
            assert st[i] == hip.RW_DONE, (i, raw, st[i])
            assert arena[ob[i]:oe[i]].tobytes() == want_b, (i, raw[:120])
            assert bool(is_err[i]) == want_err, (i, raw)
            n_done += 1
        elif blocked[0] == "pii":
            assert st[i] == hip.RW_BLOCKED, (i, raw, st[i])
            names = sorted(FIXED_NAMES[b] for b in range(len(FIXED_NAMES))
                           if (int(found[i]) >> b) & 1)
            assert names == blocked[1], (i, names, blocked[1])
        else:
            assert st[i] == hip.RW_DENY, (i, raw, st[i])
            assert int(harm_hit[i]) == blocked[1], (i, raw)
    return n_done


def test_pii_masking_over_results():
    raws = [json.dumps({"content": [{"type": "text", "text": t}]},
                       separators=(",", ":")).encode()
            for t in ["mail a@b.co now", "ssn 123-45-6789", "clean text",
                      "cards 4111 1111 1111 1111 and 192.168.0.1"]]
    assert check(raws, fl=1) == 4  # mask mode: every row completes natively
    # block mode: pii rows report the found-category names
    check(raws, fl=1, pii_block=True)


def test_harm_over_sorted_dump():
    raws = [
        json.dumps({"z": "benign", "a": "how to make a bomb"}, separators=(",", ":")).encode(),
        json.dumps({"msg": "ATTACK PLAN caps"}, separators=(",", ":")).encode(),
        json.dumps({"msg": "attack", "other": "plan"}, separators=(",", ":")).encode(),  # split → no hit
        json.dumps({"msg": "safe"}, separators=(",", ":")).encode(),
    ]
    check(raws, fl=2)


def test_toon_encoding_matches_plugin():
    rows = [{"id": i, "name": f"row{i}", "ok": i % 2 == 0} for i in range(30)]
    raws = [
        json.dumps({"content": [{"type": "text", "text": "x"}],
                    "structuredContent": {"rows": rows}}, separators=(",", ":")).encode(),
        json.dumps({"structuredContent": {"rows": rows}, "_meta": {"k": 1},
                    "isError": False}, separators=(",", ":")).encode(),
        json.dumps({"structuredContent": {"small": 1}}, separators=(",", ":")).encode(),
        json.dumps({"no_sc": "here"}, separators=(",", ":")).encode(),
        json.dumps({"structuredContent": {"mixed": [1, {"x": [1, 2]}, "s"],
                    "vals": list(range(40)), "t": "long text " * 30}},
                   separators=(",", ":")).encode(),
    ]
    assert check(raws, fl=4) == len(raws)


def test_toon_meta_nondict_punts():
    raw = json.dumps({"structuredContent": {"rows": [{"a": 1}] * 50},
                      "_meta": "not a dict"}, separators=(",", ":")).encode()
    st = run_c([raw], toon_min_size=10)[0]
    assert st[0] == hip.RW_PUNT
    # and the python plugin must handle it without raising (latent-crash fix)
    plug = ToonEncoderPlugin({"min_size": 10})
    new = plug.encode_result(json.loads(raw))
    assert new is not None and new["_meta"] == "not a dict"


def test_is_err_truthiness():
    cases = [({"isError": True}, True), ({"isError": False}, False),
             ({"isError": 0}, False), ({"isError": 1}, True),
             ({"isError": ""}, False), ({"isError": "x"}, True),
             ({"isError": None}, False), ({"isError": []}, False),
             ({"isError": {"a": 1}}, True), ({}, False)]
    raws = [json.dumps(c, separators=(",", ":")).encode() for c, _ in cases]
    st, found, hh, is_err, *_ = run_c(raws, fl=0)
    for i, (_, want) in enumerate(cases):
        assert st[i] == hip.RW_DONE and bool(is_err[i]) == want, (i, cases[i])


def test_punt_envelope():
    raws = ["café non-ascii".encode(), b'{"f":1.5}', b'{broken', b'']
    st = run_c(raws)[0]
    assert all(s != hip.RW_DONE for s in st)


def test_randomized_cross_validation():
    rng = random.Random(0xBEEF)
    frags = ["a@b.co", "123-45-6789", "plain words", "attack", "plan",
             "how to make a bomb".upper(), "10.1.2.3", "text " * 10, ""]
    raws = []
    for _ in range(300):
        def rnd_str():
            return " ".join(rng.choice(frags) for _ in range(rng.randrange(1, 3)))

        obj = {}
        if rng.random() < 0.5:
            obj["content"] = [{"type": "text", "text": rnd_str()}]
        if rng.random() < 0.6:
            rowsn = rng.randrange(1, 20)
            keys = ["".join(rng.choice(string.ascii_lowercase) for _ in range(3))
                    for _ in range(rng.randrange(1, 4))]
            obj["structuredContent"] = {
                "rows": [{k: (rng.randrange(100) if rng.random() < 0.5 else rnd_str())
                          for k in keys} for _ in range(rowsn)]}
        if rng.random() < 0.3:
            obj["isError"] = rng.random() < 0.5
        if rng.random() < 0.2:
            obj["_meta"] = {"prior": rng.randrange(10)}
        obj["extra"] = rnd_str()
        raws.append(json.dumps(obj, separators=(",", ":")).encode())
    n_done = check(raws, toon_min_size=64)
    assert n_done >= 1  # the rest blocked by pii/harm per the oracle — still verified


def test_toon_codec_roundtrip_of_c_output():
    """The C-encoded TOON text must decode back to the original
    structuredContent via the python decoder (losslessness holds)."""
    sc = {"rows": [{"id": i, "name": f"n{i}"} for i in range(20)], "total": 20}
    raw = json.dumps({"structuredContent": sc}, separators=(",", ":")).encode()
    st, found, hh, ie, arena, ob, oe = run_c([raw], fl=4, toon_min_size=10)
    assert st[0] == hip.RW_DONE
    out = json.loads(arena[ob[0]:oe[0]].tobytes())
    text = out["content"][0]["text"]
    assert text == toon_codec.encode(sc)
    assert toon_codec.decode(text) == sc


def test_threaded_range_stitching():
    """n >= 512 engages the 8-thread row split in forge_post_rows."""
    raws = []
    for i in range(700):
        raws.append(json.dumps(
            {"content": [{"type": "text", "text": f"r{i} a@b{i % 5}.co"}],
             "structuredContent": {"rows": [{"a": i, "b": f"v{i}"}] * 12},
             "isError": i % 13 == 0}, separators=(",", ":")).encode())
    st, found, hh, ie, arena, ob, oe = run_c(raws, fl=5, toon_min_size=64)
    n_done = 0
    for i, raw in enumerate(raws):
        want_b, blocked, want_err = py_reference(raw, fl=5, toon_min_size=64)
        assert blocked is None and st[i] == hip.RW_DONE, (i, st[i])
        assert arena[ob[i]:oe[i]].tobytes() == want_b, i
        assert bool(ie[i]) == want_err, i
        n_done += 1
    assert n_done == 700
